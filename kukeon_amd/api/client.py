"""Client interfaces to the kukeon control plane.

Three implementations, mirroring the reference's pkg/api/kukeonv1:
  * UnixClient — JSON-RPC over the daemon's unix socket (kuke's default),
  * LocalClient — in-process over a Controller (kuke's --local / init path),
  * FakeClient — every method raises UnexpectedCall; tests embed+override.
"""
from __future__ import annotations

import json
import socket
import threading
from typing import Any, Dict, List, Optional

from kukeon_amd.api import errors

DEFAULT_SOCKET = "/run/kukeon/kukeond.sock"


class Client:
    """Dynamic method surface: client.CreateCell(doc=...) style calls."""

    def call(self, method: str, **params) -> Any:
        raise NotImplementedError

    def __getattr__(self, name: str):
        if name.startswith("_") or not name[0].isupper():
            raise AttributeError(name)

        def _invoke(**params):
            return self.call(name, **params)
        return _invoke


class UnixClient(Client):
    def __init__(self, socket_path: str = DEFAULT_SOCKET,
                 timeout: float = 30.0):
        self.socket_path = socket_path
        self.timeout = timeout
        self._lock = threading.Lock()
        self._sock: Optional[socket.socket] = None
        self._rfile = None
        self._next_id = 0

    def _connect(self):
        s = socket.socket(socket.AF_UNIX, socket.SOCK_STREAM)
        s.settimeout(self.timeout)
        s.connect(self.socket_path)
        self._sock = s
        self._rfile = s.makefile("rb")

    def call(self, method: str, **params) -> Any:
        with self._lock:
            if self._sock is None:
                self._connect()
            self._next_id += 1
            req = {"id": self._next_id, "method": method, "params": params}
            try:
                self._sock.sendall((json.dumps(req) + "\n").encode())
                line = self._rfile.readline()
            except OSError:
                # one reconnect attempt (daemon restarted)
                self._connect()
                self._sock.sendall((json.dumps(req) + "\n").encode())
                line = self._rfile.readline()
            if not line:
                raise errors.KukeonError("daemon closed the connection")
            resp = json.loads(line)
            if "error" in resp:
                raise errors.from_wire(resp["error"])
            return resp.get("result")

    def close(self):
        if self._sock is not None:
            self._sock.close()
            self._sock = None


class LocalClient(Client):
    """In-process client over a Controller (no daemon)."""

    def __init__(self, controller):
        from kukeon_amd.daemon.server import Service
        self._service = Service(controller)

    def call(self, method: str, **params) -> Any:
        # round-trip through JSON so Local and Unix behave identically
        payload = json.loads(json.dumps(params))
        return self._service.dispatch(method, payload)


class UnexpectedCall(errors.KukeonError):
    code = "ErrUnexpectedCall"


class FakeClient(Client):
    """Test double: every method raises UnexpectedCall unless a handler is
    installed via `fake.on("CreateCell", fn)`."""

    def __init__(self):
        self.calls: List[Dict[str, Any]] = []
        self._handlers: Dict[str, Any] = {}

    def on(self, method: str, handler) -> "FakeClient":
        self._handlers[method] = handler
        return self

    def call(self, method: str, **params) -> Any:
        self.calls.append({"method": method, "params": params})
        if method in self._handlers:
            return self._handlers[method](**params)
        raise UnexpectedCall(f"unexpected RPC {method}")
