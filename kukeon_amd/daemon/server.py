"""kukeond: the daemon. Unix-socket JSON-RPC server + reconcile loops.

Wire protocol: newline-delimited JSON frames
    {"id": N, "method": "CreateCell", "params": {...}}
 -> {"id": N, "result": ...} | {"id": N, "error": {"code", "message"}}

Background loops (reference: internal/daemon/server.go:224-342 — eager
first reconcile on startup, then periodic passes; RPC service surface
mirrors rpcservice.go:38-475; instance pinning instance.go:59-141): a cell-reconcile
pass every `reconcile_interval` seconds (eager first pass on startup so
state converges after a host restart), a session-lifetime pass, and a
space-network re-assert pass; each pass is exception-guarded so one bad
resource can't kill the loop.
"""
from __future__ import annotations

import contextlib
import json
import logging
import os
import socketserver
import threading
import time
from pathlib import Path
from typing import Any, Dict, Optional

from kukeon_amd.api import errors
from kukeon_amd.api import v1beta1 as api
from kukeon_amd.controller.core import Controller

log = logging.getLogger("kukeon.daemon")

INSTANCE_FILE = ".kukeon-instance.json"


def verify_or_write_instance(run_path: Path, namespace_suffix: str = "",
                             cgroup_root: str = "kukeon") -> None:
    """Pin daemon identity so a restart with different flags fails fast."""
    p = run_path / INSTANCE_FILE
    want = {"namespaceSuffix": namespace_suffix, "cgroupRoot": cgroup_root}
    if p.exists():
        cur = json.loads(p.read_text())
        if cur != want:
            raise errors.InvalidArgument(
                f"instance mismatch at {p}: have {cur}, want {want}")
        return
    run_path.mkdir(parents=True, exist_ok=True)
    p.write_text(json.dumps(want))


class Service:
    """RPC method surface (the kukeonv1 service analog)."""

    def __init__(self, ctl: Controller, server: "Server" = None):
        self.ctl = ctl
        self.server = server

    # -- helpers -------------------------------------------------------
    @staticmethod
    def _doc(params: Dict[str, Any], cls):
        return cls.from_dict(params["doc"])

    def dispatch(self, method: str, params: Dict[str, Any]) -> Any:
        fn = getattr(self, method, None)
        if fn is None or method.startswith("_") or not callable(fn):
            raise errors.InvalidArgument(f"unknown RPC method {method!r}")
        return fn(params or {})

    # -- daemon --------------------------------------------------------
    def Ping(self, p):
        return {"ok": True, "pid": os.getpid(),
                "version": __import__("kukeon_amd").__version__}

    def Status(self, p):
        return {
            "pid": os.getpid(),
            "runPath": str(self.ctl.run_path),
            "cgroupMode": self.ctl.cgroups.mode,
            "gpus": {"devices": self.ctl.gpus.devices,
                     "free": self.ctl.gpus.free,
                     "assignments": self.ctl.gpus.assignments},
            "reconcileIntervalSeconds":
                self.server.reconcile_interval if self.server else None,
        }

    def DaemonStop(self, p):
        if self.server:
            threading.Thread(target=self.server.stop, daemon=True).start()
        return {"stopping": True}

    # -- scopes --------------------------------------------------------
    def CreateRealm(self, p):
        return self.ctl.create_realm(self._doc(p, api.RealmDoc)).to_dict()

    def GetRealm(self, p):
        return self.ctl.get_realm(p["name"]).to_dict()

    def ListRealms(self, p):
        return [d.to_dict() for d in self.ctl.list_realms()]

    def DeleteRealm(self, p):
        self.ctl.delete_realm(p["name"], p.get("cascade", False))
        return {}

    def CreateSpace(self, p):
        return self.ctl.create_space(self._doc(p, api.SpaceDoc)).to_dict()

    def GetSpace(self, p):
        return self.ctl.get_space(p["realm"], p["name"]).to_dict()

    def ListSpaces(self, p):
        return [d.to_dict() for d in self.ctl.list_spaces(p["realm"])]

    def DeleteSpace(self, p):
        self.ctl.delete_space(p["realm"], p["name"], p.get("cascade", False))
        return {}

    def CreateStack(self, p):
        return self.ctl.create_stack(self._doc(p, api.StackDoc)).to_dict()

    def GetStack(self, p):
        return self.ctl.get_stack(p["realm"], p["space"], p["name"]).to_dict()

    def ListStacks(self, p):
        return [d.to_dict()
                for d in self.ctl.list_stacks(p["realm"], p["space"])]

    def DeleteStack(self, p):
        self.ctl.delete_stack(p["realm"], p["space"], p["name"],
                              p.get("cascade", False))
        return {}

    # -- cells ---------------------------------------------------------
    def CreateCell(self, p):
        doc = self._doc(p, api.CellDoc)
        if p.get("runtimeEnv"):
            doc.spec.runtime_env = list(p["runtimeEnv"])
        return self.ctl.create_cell(doc).to_dict()

    def GetCell(self, p):
        return self.ctl.get_cell(p["realm"], p["space"], p["stack"],
                                 p["name"]).to_dict()

    def ListCells(self, p):
        return [d.to_dict() for d in
                self.ctl.list_cells(p["realm"], p["space"], p["stack"])]

    def CellMetrics(self, p):
        return self.ctl.cell_metrics(p["realm"], p["space"], p["stack"],
                                     p["name"])

    def StartCell(self, p):
        return self.ctl.start_cell(p["realm"], p["space"], p["stack"],
                                   p["name"]).to_dict()

    def StopCell(self, p):
        return self.ctl.stop_cell(p["realm"], p["space"], p["stack"],
                                  p["name"]).to_dict()

    def KillCell(self, p):
        return self.ctl.kill_cell(p["realm"], p["space"], p["stack"],
                                  p["name"]).to_dict()

    def RestartCell(self, p):
        return self.ctl.restart_cell(p["realm"], p["space"], p["stack"],
                                     p["name"]).to_dict()

    def StartContainer(self, p):
        return self.ctl.start_container(
            p["realm"], p["space"], p["stack"], p["name"],
            p["container"]).to_dict()

    def StopContainer(self, p):
        return self.ctl.stop_container(
            p["realm"], p["space"], p["stack"], p["name"],
            p["container"]).to_dict()

    def RestartContainer(self, p):
        return self.ctl.restart_container(
            p["realm"], p["space"], p["stack"], p["name"],
            p["container"]).to_dict()

    def DeleteCell(self, p):
        self.ctl.delete_cell(p["realm"], p["space"], p["stack"], p["name"],
                             p.get("force", False))
        return {}

    def PurgeCell(self, p):
        self.ctl.purge_cell(p["realm"], p["space"], p["stack"], p["name"])
        return {}

    def AttachContainer(self, p):
        path = self.ctl.attach_path(p["realm"], p["space"], p["stack"],
                                    p["name"])
        return {"hostSocketPath": path}

    def LogPath(self, p):
        return {"path": self.ctl.log_path(p["realm"], p["space"], p["stack"],
                                          p["name"], p.get("container", ""))}

    def RunFromBlueprint(self, p):
        return self.ctl.run_from_blueprint(
            p["realm"], p["space"], p["stack"], p["blueprint"],
            p.get("params", {}), p.get("env"), p.get("name")).to_dict()

    def RunFromConfig(self, p):
        return self.ctl.run_from_config(
            p["realm"], p["space"], p["stack"], p["config"],
            p.get("params", {}), p.get("name")).to_dict()

    # -- scoped docs -----------------------------------------------------
    def PutSecret(self, p):
        self.ctl.put_secret(self._doc(p, api.SecretDoc))
        return {}

    def GetSecret(self, p):
        return self.ctl.get_secret(p["realm"], p["space"],
                                   p["name"]).to_dict()

    def ListSecrets(self, p):
        return self.ctl.list_secrets(p["realm"], p["space"])

    def DeleteSecret(self, p):
        self.ctl.delete_secret(p["realm"], p["space"], p["name"])
        return {}

    def PutBlueprint(self, p):
        self.ctl.put_blueprint(self._doc(p, api.CellBlueprintDoc))
        return {}

    def GetBlueprint(self, p):
        return self.ctl.get_blueprint(p["realm"], p["space"],
                                      p["name"]).to_dict()

    def ListBlueprints(self, p):
        return self.ctl.list_blueprints(p["realm"], p["space"])

    def DeleteBlueprint(self, p):
        self.ctl.delete_blueprint(p["realm"], p["space"], p["name"])
        return {}

    def PutConfig(self, p):
        self.ctl.put_config(self._doc(p, api.CellConfigDoc))
        return {}

    def GetConfig(self, p):
        return self.ctl.get_config(p["realm"], p["space"],
                                   p["name"]).to_dict()

    def ListConfigs(self, p):
        return self.ctl.list_configs(p["realm"], p["space"])

    def DeleteConfig(self, p):
        self.ctl.delete_config(p["realm"], p["space"], p["name"])
        return {}

    def PutVolume(self, p):
        return self.ctl.put_volume(self._doc(p, api.VolumeDoc)).to_dict()

    def GetVolume(self, p):
        return self.ctl.get_volume(p["realm"], p["space"],
                                   p["name"]).to_dict()

    def DeleteVolume(self, p):
        self.ctl.delete_volume(p["realm"], p["space"], p["name"])
        return {}

    # -- sessions --------------------------------------------------------
    def CreateSession(self, p):
        return self.ctl.create_session(self._doc(p, api.SessionDoc)).to_dict()

    def GetSession(self, p):
        return self.ctl.get_session(p["realm"], p["space"], p["stack"],
                                    p["name"]).to_dict()

    def ListSessions(self, p):
        return [d.to_dict() for d in self.ctl.list_sessions()]

    def CloseSession(self, p):
        return self.ctl.close_session(
            p["realm"], p["space"], p["stack"], p["name"],
            p.get("state", api.STATE_COMPLETED)).to_dict()

    def DeleteSession(self, p):
        self.ctl.delete_session(p["realm"], p["space"], p["stack"],
                                p["name"])
        return {}

    def TouchSession(self, p):
        self.ctl.touch_session(p["realm"], p["space"], p["stack"], p["name"])
        return {}

    # -- apply / reconcile -----------------------------------------------
    def ApplyDocuments(self, p):
        results = self.ctl.apply_documents(p["yaml"], p.get("team", ""))
        return [{"kind": r.kind, "name": r.name, "action": r.action,
                 "error": r.error} for r in results]

    def ReconcileCells(self, p):
        return {"visited": self.ctl.reconcile_cells()}

    def ReconcileSessions(self, p):
        return {"visited": self.ctl.reconcile_sessions()}

    def RefreshAll(self, p):
        return self.ctl.refresh_all()


class Server:
    def __init__(self, ctl: Controller, socket_path: str,
                 reconcile_interval: float = 30.0,
                 socket_gid: Optional[int] = None):
        self.ctl = ctl
        self.socket_path = socket_path
        self.reconcile_interval = reconcile_interval
        self.socket_gid = socket_gid
        self.service = Service(ctl, self)
        self._stop = threading.Event()
        self._threads = []
        self._srv: Optional[socketserver.ThreadingUnixStreamServer] = None

    # ------------------------------------------------------------------
    def start(self) -> None:
        verify_or_write_instance(self.ctl.run_path)
        sp = Path(self.socket_path)
        sp.parent.mkdir(parents=True, exist_ok=True)
        with contextlib.suppress(FileNotFoundError):
            sp.unlink()
        service = self.service

        class Handler(socketserver.StreamRequestHandler):
            def handle(self):
                for line in self.rfile:
                    line = line.strip()
                    if not line:
                        continue
                    try:
                        req = json.loads(line)
                        result = service.dispatch(req.get("method", ""),
                                                  req.get("params"))
                        resp = {"id": req.get("id"), "result": result}
                    except Exception as e:  # noqa: BLE001
                        resp = {"id": req.get("id") if isinstance(req, dict)
                                else None,
                                "error": errors.to_wire(e)}
                        if not isinstance(e, errors.KukeonError):
                            log.exception("RPC %s failed",
                                          req.get("method", "?"))
                    try:
                        self.wfile.write(
                            (json.dumps(resp) + "\n").encode())
                        self.wfile.flush()
                    except (BrokenPipeError, OSError):
                        return

        self._srv = socketserver.ThreadingUnixStreamServer(
            str(sp), Handler, bind_and_activate=True)
        self._srv.daemon_threads = True
        os.chmod(sp, 0o660)
        if self.socket_gid is not None:
            with contextlib.suppress(OSError):
                os.chown(sp, -1, self.socket_gid)
        (self.ctl.run_path / "kukeond.pid").write_text(str(os.getpid()))
        t = threading.Thread(target=self._srv.serve_forever,
                             name="rpc", daemon=True)
        t.start()
        self._threads.append(t)
        t2 = threading.Thread(target=self._reconcile_loop,
                              name="reconcile", daemon=True)
        t2.start()
        self._threads.append(t2)
        log.info("kukeond serving on %s (reconcile every %.0fs)",
                 sp, self.reconcile_interval)

    def _reconcile_loop(self) -> None:
        # eager first pass converges stale state after a host restart
        first = True
        while not self._stop.is_set():
            if not first:
                self._stop.wait(self.reconcile_interval)
                if self._stop.is_set():
                    break
            first = False
            if self.reconcile_interval <= 0:
                # interval 0 = loop disabled (test harness contract)
                self._stop.wait(0.2)
                continue
            for pass_fn in (self.ctl.reconcile_cells,
                            self.ctl.reconcile_sessions,
                            self.ctl.reconcile_space_networks):
                try:
                    pass_fn()
                except Exception:  # noqa: BLE001
                    log.exception("reconcile pass %s failed",
                                  pass_fn.__name__)

    def stop(self) -> None:
        self._stop.set()
        if self._srv is not None:
            self._srv.shutdown()
            self._srv.server_close()
        with contextlib.suppress(FileNotFoundError):
            Path(self.socket_path).unlink()
        with contextlib.suppress(FileNotFoundError):
            (self.ctl.run_path / "kukeond.pid").unlink()

    def wait(self) -> None:
        try:
            while not self._stop.is_set():
                time.sleep(0.5)
        except KeyboardInterrupt:
            self.stop()
