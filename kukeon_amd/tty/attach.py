"""Interactive attach client: raw-mode byte pump against a cell's tty socket.

The hot loop runs entirely in the CLI process against the unix socket the
cell-side shim serves — the daemon only resolves the path (reference
contract: zero daemon involvement per byte). Detach: Ctrl-] Ctrl-] leaves
the workload running.
"""
from __future__ import annotations

import contextlib
import fcntl
import json
import os
import select
import signal
import socket
import struct
import sys
import termios
import tty as ttymod

DETACH_KEY = b"\x1d"  # Ctrl-]


def _send_winsize(sock: socket.socket) -> None:
    with contextlib.suppress(OSError):
        ws = fcntl.ioctl(1, termios.TIOCGWINSZ, struct.pack("HHHH", 0, 0, 0, 0))
        rows, cols, _, _ = struct.unpack("HHHH", ws)
        msg = b"\x00R" + json.dumps({"rows": rows, "cols": cols}).encode() + b"\n"
        sock.sendall(msg)


def _request_fd(sock: socket.socket):
    """Ask the shim for the PTY master via SCM_RIGHTS (reference
    cmd/kuketty SCM_RIGHTS contract — the byte pump is then
    CLI<->kernel with no shim relay). Returns the fd or None (another
    client already holds it, or an old shim)."""
    try:
        sock.sendall(b"\x00F\n")
        sock.settimeout(2.0)
        msg, fds, _, _ = socket.recv_fds(sock, 64, 1)
        sock.settimeout(None)
        if fds and msg.startswith(b"\x00FDOK"):
            return fds[0]
        for fd in fds:
            os.close(fd)
    except (OSError, ValueError):
        pass
    return None


def attach(socket_path: str, stdin=None, stdout=None,
           want_fd: bool = True) -> int:
    """Returns 0 on detach, 2 if the remote side closed."""
    stdin = stdin if stdin is not None else sys.stdin
    stdout = stdout if stdout is not None else sys.stdout
    sock = socket.socket(socket.AF_UNIX, socket.SOCK_STREAM)
    sock.connect(socket_path)
    # read the one-line handshake
    buf = b""
    while not buf.endswith(b"\n"):
        b1 = sock.recv(1)
        if not b1:
            return 2
        buf += b1
        if len(buf) > 4096:
            break
    mfd = _request_fd(sock) if want_fd else None
    if mfd is not None:
        return _attach_fd(sock, mfd, stdin, stdout)
    _send_winsize(sock)

    in_fd = stdin.fileno()
    interactive = os.isatty(in_fd)
    old = None
    if interactive:
        old = termios.tcgetattr(in_fd)
        ttymod.setraw(in_fd)
        signal.signal(signal.SIGWINCH, lambda *_: _send_winsize(sock))
    detach_armed = False
    rc = 2
    try:
        while True:
            rd, _, _ = select.select([in_fd, sock], [], [])
            if sock in rd:
                data = sock.recv(65536)
                if not data:
                    rc = 2
                    break
                os.write(stdout.fileno(), data)
            if in_fd in rd:
                data = os.read(in_fd, 4096)
                if not data:
                    rc = 0
                    break
                if interactive:
                    if detach_armed and DETACH_KEY in data:
                        rc = 0
                        break
                    detach_armed = data.endswith(DETACH_KEY)
                    if detach_armed and data.count(DETACH_KEY) >= 2:
                        rc = 0
                        break
                sock.sendall(data)
    finally:
        if old is not None:
            termios.tcsetattr(in_fd, termios.TCSADRAIN, old)
        sock.close()
    return rc


def ping(socket_path: str, timeout: float = 0.5) -> bool:
    """One-shot liveness check of the shim's attach socket."""
    try:
        sock = socket.socket(socket.AF_UNIX, socket.SOCK_STREAM)
        sock.settimeout(timeout)
        sock.connect(socket_path)
        data = sock.recv(256)
        sock.close()
        return bool(data)
    except OSError:
        return False


def _set_winsize_fd(mfd: int) -> None:
    with contextlib.suppress(OSError):
        ws = fcntl.ioctl(1, termios.TIOCGWINSZ,
                         struct.pack("HHHH", 0, 0, 0, 0))
        fcntl.ioctl(mfd, termios.TIOCSWINSZ, ws)


def _attach_fd(sock: socket.socket, mfd: int, stdin, stdout) -> int:
    """Direct PTY pump: bytes move stdin<->master with no shim in the
    path; the control socket stays open only so the shim can detect
    detach and resume capture."""
    in_fd = stdin.fileno()
    interactive = os.isatty(in_fd)
    old = None
    if interactive:
        old = termios.tcgetattr(in_fd)
        ttymod.setraw(in_fd)
        _set_winsize_fd(mfd)
        signal.signal(signal.SIGWINCH, lambda *_: _set_winsize_fd(mfd))
    detach_armed = False
    rc = 2
    try:
        while True:
            rd, _, _ = select.select([in_fd, mfd], [], [])
            if mfd in rd:
                try:
                    data = os.read(mfd, 65536)
                except OSError:
                    data = b""
                if not data:
                    rc = 2
                    break
                os.write(stdout.fileno(), data)
            if in_fd in rd:
                data = os.read(in_fd, 4096)
                if not data:
                    rc = 0
                    break
                if interactive:
                    if detach_armed and DETACH_KEY in data:
                        rc = 0
                        break
                    detach_armed = data.endswith(DETACH_KEY)
                    if detach_armed and data.count(DETACH_KEY) >= 2:
                        rc = 0
                        break
                os.write(mfd, data)
    finally:
        if old is not None:
            termios.tcsetattr(in_fd, termios.TCSADRAIN, old)
        os.close(mfd)
        sock.close()
    return rc
