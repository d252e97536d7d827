"""Container shim: the kuketty/kukepause analog for process cells.

One shim per container (the runtime spawns `python -m kukeon_amd.tty.shim
--dir <container_dir>` in its own session/process group; reference parity:
cmd/kuketty/main.go:57-117 PTY server + metadata contract, stages
stages.go:62-88, kukepause PID-1 semantics cmd/kukepause/main.go:17-60 via
--pause mode). It:

* reads the spawn spec (`spawn.json`) the runner rendered — argv, env, cwd,
  attachable flag, capture path — and records `runtime.json` (shim pid +
  start time, workload pid) for liveness probes that survive daemon restarts,
* supervises the workload and writes `status.json` {exitCode, finishedAt}
  durably on exit (the containerd-shim role: exit codes outlive the daemon),
* for attachable containers: owns the PTY master, serves the attach socket
  at <dir>/tty/socket (multi-client byte pump, capture file for `kuke log`,
  activity timestamps for Session idleTimeout), runs the optional tty init
  script first,
* runs container setup before the workload (the kuketty runOn:create
  role): clones/fetches declared git repos with the container-local git
  identity and records per-repo state to `setup.json`, which the
  reconcile loop surfaces into `ContainerStatus.repos`,
* `--pause` mode is the cell root: a signal-wait PID holding the cell's
  process group open (kukepause analog; exits 0 on SIGTERM/SIGINT).
"""
from __future__ import annotations

import argparse
import contextlib
import fcntl
import json
import os
import pty
import select
import signal
import socket
import sys
import time
from pathlib import Path


def now_iso() -> str:
    return time.strftime("%Y-%m-%dT%H:%M:%SZ", time.gmtime())


def write_json(path: Path, data: dict) -> None:
    tmp = path.with_suffix(".tmp")
    tmp.write_text(json.dumps(data))
    os.replace(tmp, path)


def run_pause(unshare_kinds: str = "", hostname: str = "",
              ns_record: str = "") -> int:
    """Root-container pause: owns the cell's namespaces (net/uts/ipc —
    the pod model of reference internal/ctr/spec.go:38), reaps children,
    exits 0 on TERM/INT. Namespace setup degrades gracefully on hosts
    without CAP_SYS_ADMIN and records the outcome for the runner."""
    held = []
    if unshare_kinds:
        from kukeon_amd.runtime import namespaces as nsmod
        for kind in unshare_kinds.split(","):
            fl = nsmod.NS_FLAGS.get(kind)
            if not fl or kind in ("mnt", "pid"):
                continue
            try:
                nsmod.unshare(fl)
                held.append(kind)
            except OSError as e:
                sys.stderr.write(f"unshare {kind}: {e}\n")
        if "uts" in held and hostname:
            try:
                nsmod.sethostname(hostname)
            except OSError:
                pass
        if "net" in held:
            # fresh netns: bring loopback up so in-cell localhost works
            try:
                from kukeon_amd.runtime.netlink import Rtnl
                with Rtnl() as nl:
                    nl.set_link_up_by_index(1)
            except OSError as e:
                sys.stderr.write(f"netns lo up: {e}\n")
    if ns_record:
        write_json(Path(ns_record),
                   {"held": held, "degraded": bool(unshare_kinds) and
                    set(unshare_kinds.split(",")) - set(held) != set()})
    stop = {"flag": False}

    def on_term(signum, frame):
        stop["flag"] = True

    def on_chld(signum, frame):
        try:
            while os.waitpid(-1, os.WNOHANG)[0] > 0:
                pass
        except ChildProcessError:
            pass

    signal.signal(signal.SIGTERM, on_term)
    signal.signal(signal.SIGINT, on_term)
    signal.signal(signal.SIGCHLD, on_chld)
    while not stop["flag"]:
        signal.pause()
    return 0


class Shim:
    CAP_LIMIT = 32 << 20   # capture.log rotation threshold
    def __init__(self, cdir: Path):
        self.dir = cdir
        self.spec = json.loads((cdir / "spawn.json").read_text())
        self.child_pid = 0

    def record_runtime(self, workload_pid: int) -> None:
        from kukeon_amd.runtime.proc import proc_starttime
        write_json(self.dir / "runtime.json", {
            "shimPid": os.getpid(),
            "shimStarttime": proc_starttime(os.getpid()),
            "workloadPid": workload_pid,
            "startedAt": now_iso(),
        })

    def record_status(self, rc: int) -> None:
        write_json(self.dir / "status.json",
                   {"exitCode": rc, "finishedAt": now_iso()})

    def run_setup(self) -> None:
        """Clone/fetch declared repos; write per-repo states to
        setup.json. Failures are recorded, not fatal — the workload still
        starts and the agent can inspect status.repos."""
        repos = self.spec.get("repos") or []
        if not repos:
            return
        import subprocess
        git_id = self.spec.get("git") or {}
        env = self.child_env()
        if git_id.get("name"):
            env["GIT_AUTHOR_NAME"] = env["GIT_COMMITTER_NAME"] = \
                git_id["name"]
        if git_id.get("email"):
            env["GIT_AUTHOR_EMAIL"] = env["GIT_COMMITTER_EMAIL"] = \
                git_id["email"]
        states = []
        for r in repos:
            url = r.get("url", "")
            dest = Path(r.get("path") or Path(url).name or "repo")
            if not dest.is_absolute():
                dest = self.dir / dest
            st = {"url": url, "state": "", "error": ""}
            try:
                if (dest / ".git").is_dir():
                    subprocess.run(["git", "-C", str(dest), "fetch",
                                    "--all", "--tags"], env=env, timeout=120,
                                   capture_output=True, check=True)
                    st["state"] = "fetched"
                else:
                    cmd = ["git", "clone", url, str(dest)]
                    subprocess.run(cmd, env=env, timeout=300,
                                   capture_output=True, check=True)
                    st["state"] = "cloned"
                if r.get("ref"):
                    subprocess.run(["git", "-C", str(dest), "checkout",
                                    r["ref"]], env=env, timeout=60,
                                   capture_output=True, check=True)
            except subprocess.CalledProcessError as e:
                st["state"] = "failed"
                st["error"] = (e.stderr or b"").decode(
                    "utf-8", "replace")[-500:]
            except Exception as e:  # timeout, missing git, ...
                st["state"] = "failed"
                st["error"] = str(e)[-500:]
            states.append(st)
        write_json(self.dir / "setup.json", {"repos": states})

    def enter_namespaces(self) -> None:
        """Join the cell root's net/uts/ipc namespaces and take a
        private mount namespace with rendered /etc files (and optional
        image rootfs). Runs in the shim BEFORE any fork so the PTY and
        workload inherit everything. Degrades per-kind; the outcome is
        recorded to ns.json for status derivation."""
        ns = self.spec.get("ns") or {}
        if not ns:
            return
        from kukeon_amd.runtime import namespaces as nsmod
        held: list = []
        degraded = False
        join_pid = ns.get("joinPid") or 0
        if join_pid and ns.get("join"):
            held += nsmod.join(join_pid, list(ns["join"]))
            if set(ns["join"]) - set(held):
                degraded = True
        if ns.get("mountNs"):
            try:
                nsmod.unshare(nsmod.CLONE_NEWNS)
                nsmod.make_mounts_private()
                held.append("mnt")
                rootfs = ns.get("rootfs") or ""
                if rootfs:
                    try:
                        self._enter_rootfs(nsmod, ns, rootfs)
                        held.append("rootfs")
                    except OSError as e:
                        degraded = True
                        sys.stderr.write(f"rootfs: {e}\n")
                elif ns.get("hostname"):
                    try:
                        nsmod.setup_etc(ns["hostname"],
                                        ns.get("hosts") or {},
                                        str(self.dir / "etc"))
                    except OSError as e:
                        sys.stderr.write(f"etc render: {e}\n")
                if not rootfs:
                    self._bind_volumes(nsmod, ns)
            except OSError as e:
                degraded = True
                sys.stderr.write(f"mount ns: {e}\n")
        write_json(self.dir / "ns.json",
                   {"held": held, "degraded": degraded})

    def _bind_volumes(self, nsmod, ns: dict, prefix: str = "") -> None:
        """Bind-mount declared volumes at their targets inside this
        container's private mount namespace (real mounts — reference
        internal/ctr/spec.go volume mounts; the host tree is
        unaffected)."""
        for b in ns.get("binds") or []:
            src, dst = b.get("src"), b.get("dst")
            if not src or not dst:
                continue
            if prefix:
                dst = os.path.join(prefix, dst.lstrip("/"))
            try:
                if os.path.isdir(src):
                    os.makedirs(dst, exist_ok=True)
                else:
                    os.makedirs(os.path.dirname(dst) or "/",
                                exist_ok=True)
                    if not os.path.exists(dst):
                        open(dst, "a").close()
                nsmod.mount(src, dst, "", nsmod.MS_BIND | nsmod.MS_REC)
            except OSError as e:
                sys.stderr.write(f"bind {src} -> {dst}: {e}\n")

    def _enter_rootfs(self, nsmod, ns: dict, rootfs) -> None:
        """chroot into an image rootfs keeping the shim functional: the
        container state dir, /proc and /dev are bind-mounted inside at
        their host paths so runtime/status files and the PTY keep
        working; /etc/hostname + hosts are rendered INTO the layer.
        `rootfs` is either a direct path or an overlay config
        {layers, upper, work, mnt} mounted here so it dies with this
        mount namespace."""
        import kukeon_amd.runtime.proc  # noqa: F401  pre-import (chroot)
        if isinstance(rootfs, dict):
            lowers = ":".join(reversed(rootfs["layers"]))
            for k in ("upper", "work", "mnt"):
                os.makedirs(rootfs[k], exist_ok=True)
            nsmod.mount("overlay", rootfs["mnt"], "overlay", 0,
                        f"lowerdir={lowers},upperdir={rootfs['upper']},"
                        f"workdir={rootfs['work']}")
            rootfs = rootfs["mnt"]
        for sub, src in ((str(self.dir).lstrip("/"), str(self.dir)),
                         ("proc", "/proc"), ("dev", "/dev"),
                         ("tmp", "/tmp")):
            tgt = os.path.join(rootfs, sub)
            os.makedirs(tgt, exist_ok=True)
            nsmod.mount(src, tgt, "", nsmod.MS_BIND | nsmod.MS_REC)
        etc = os.path.join(rootfs, "etc")
        os.makedirs(etc, exist_ok=True)
        hostname = ns.get("hostname") or "cell"
        with open(os.path.join(etc, "hostname"), "w") as f:
            f.write(hostname + "\n")
        with open(os.path.join(etc, "hosts"), "w") as f:
            f.write("127.0.0.1\tlocalhost\n")
            f.write(f"127.0.1.1\t{hostname}\n")
            for name, ip in sorted((ns.get("hosts") or {}).items()):
                f.write(f"{ip}\t{name}\n")
        # volumes bind BEFORE the chroot (host sources are unreachable
        # after), at rootfs-prefixed targets
        self._bind_volumes(nsmod, ns, prefix=rootfs)
        if ns.get("readOnlyRootfs"):
            # readOnlyRootFilesystem (reference OCI Root.readonly): the
            # overlay remounts read-only; the state-dir//dev//tmp and
            # volume binds above stay writable (they are separate
            # mounts). Only enforceable for image-rooted cells (a
            # host-rootfs process cell cannot make the host / read-only).
            nsmod.mount("", rootfs, "",
                        nsmod.MS_REMOUNT | nsmod.MS_BIND |
                        nsmod.MS_RDONLY)
        nsmod.enter_rootfs(rootfs)

    def child_env(self) -> dict:
        env = dict(os.environ)
        for kv in (self.spec.get("ns") or {}).get("imageEnv", []):
            k, _, v = kv.partition("=")
            env[k] = v
        for kv in self.spec.get("env", []):
            k, _, v = kv.partition("=")
            env[k] = v
        env.setdefault("TERM", "xterm-256color")
        return env

    # ------------------------------------------------------------------
    def run_plain(self) -> int:
        pid = os.fork()
        if pid == 0:
            self._exec_child()
        self.child_pid = pid
        self.record_runtime(pid)
        _, status = os.waitpid(pid, 0)
        rc = os.waitstatus_to_exitcode(status)
        self.record_status(rc if rc >= 0 else 128 - rc)
        return 0

    @staticmethod
    def _drop_privileges(user: str) -> None:
        """ContainerSpec.user (reference OCI Process.user): numeric
        "uid[:gid]" or a passwd name; the workload execs with the
        dropped identity. Root-only — a non-root shim (or unknown user)
        records the degrade on stderr and continues."""
        if not user or os.geteuid() != 0:
            return
        try:
            import pwd as _pwd
            uid_s, _, gid_s = user.partition(":")
            if uid_s.isdigit():
                uid = int(uid_s)
                gid = int(gid_s) if gid_s.isdigit() else uid
                try:
                    name = _pwd.getpwuid(uid).pw_name
                except KeyError:
                    name = None
            else:
                ent = _pwd.getpwnam(uid_s)
                uid, gid, name = ent.pw_uid, ent.pw_gid, ent.pw_name
                if gid_s.isdigit():
                    gid = int(gid_s)
            if name is not None:
                import grp as _grp  # noqa: F401  (ensure module loads)
                os.initgroups(name, gid)
            else:
                os.setgroups([gid])
            os.setgid(gid)
            os.setuid(uid)
        except (OSError, KeyError) as e:
            sys.stderr.write(f"user {user!r}: {e}; running as root\n")

    def _exec_child(self):
        spec = self.spec
        ns = spec.get("ns") or {}
        if spec.get("cwd"):
            try:
                os.chdir(spec["cwd"])
            except OSError:
                pass
        if ns.get("imageWorkdir"):
            try:
                os.chdir(ns["imageWorkdir"])
            except OSError:
                pass
        argv = spec["argv"]
        if argv == ["/bin/sh"] and ns.get("imageCmd"):
            argv = ["/bin/sh", "-c", ns["imageCmd"]]
        self._drop_privileges(spec.get("user") or "")
        try:
            os.execvpe(argv[0], argv, self.child_env())
        except OSError as e:
            sys.stderr.write(f"exec {argv[0]}: {e}\n")
            os._exit(127)

    # ------------------------------------------------------------------
    def run_attachable(self) -> int:
        tty_dir = self.dir / "tty"
        tty_dir.mkdir(parents=True, exist_ok=True)
        sock_path = tty_dir / "socket"
        try:
            sock_path.unlink()
        except FileNotFoundError:
            pass
        srv = socket.socket(socket.AF_UNIX, socket.SOCK_STREAM)
        # bind via a relative path: the absolute one can exceed SUN_PATH
        # (107 bytes); clients reach it through the daemon's short symlink
        cwd = os.getcwd()
        try:
            os.chdir(tty_dir)
            srv.bind("socket")
        finally:
            os.chdir(cwd)
        os.chmod(sock_path, 0o660)
        srv.listen(8)
        srv.setblocking(False)

        pid, master = pty.fork()
        if pid == 0:
            # child: on the PTY slave, its own session
            spec = self.spec
            if spec.get("cwd"):
                try:
                    os.chdir(spec["cwd"])
                except OSError:
                    pass
            init = spec.get("init_script")
            argv = list(spec["argv"])
            if init:
                argv = ["/bin/sh", "-c", f"{init}\nexec \"$@\"", "sh"] + argv
            try:
                os.execvpe(argv[0], argv, self.child_env())
            except OSError as e:
                sys.stderr.write(f"exec {argv[0]}: {e}\n")
                os._exit(127)

        self.child_pid = pid
        self.record_runtime(pid)
        # capture with size-capped rotation: a chatty agent must not
        # fill the state volume; one .1 generation keeps recent history
        # for `kuke log` (32 MiB cap per generation)
        cap_path = self.dir / "capture.log"
        capture = open(cap_path, "ab", buffering=0)
        cap_limit = self.CAP_LIMIT

        def cap_write(data: bytes):
            nonlocal capture
            capture.write(data)
            if capture.tell() >= cap_limit:
                capture.close()
                os.replace(cap_path, str(cap_path) + ".1")
                capture = open(cap_path, "ab", buffering=0)
        activity = self.dir / "activity"
        activity.touch()
        clients = []
        # SCM_RIGHTS fast path (reference cmd/kuketty/main.go:17-30): one
        # client may hold the PTY master fd directly, making its byte
        # pump CLI<->kernel with zero shim copies. While an fd-holder is
        # live the shim stays out of the master (reads would steal bytes
        # from the holder), so capture pauses for that span — a
        # documented deviation; it resumes on detach.
        self._fd_holder = None
        rc = 0
        fcntl.fcntl(master, fcntl.F_SETFL,
                    fcntl.fcntl(master, fcntl.F_GETFL) | os.O_NONBLOCK)
        try:
            while True:
                rl = [srv] + clients
                if self._fd_holder is None:
                    rl.append(master)
                try:
                    rd, _, _ = select.select(rl, [], [], 0.5)
                except InterruptedError:
                    rd = []
                for r in rd:
                    if r is srv:
                        try:
                            c, _ = srv.accept()
                            c.setblocking(False)
                            self._handle_handshake(c)
                            clients.append(c)
                        except OSError:
                            pass
                    elif r is master:
                        try:
                            data = os.read(master, 65536)
                        except (OSError, IOError):
                            data = b""
                        if not data:
                            raise EOFError
                        cap_write(data)
                        dead = []
                        for c in clients:
                            try:
                                c.sendall(data)
                            except OSError:
                                dead.append(c)
                        for c in dead:
                            clients.remove(c)
                            c.close()
                    else:
                        try:
                            data = r.recv(65536)
                        except OSError:
                            data = b""
                        if not data:
                            clients.remove(r)
                            if self._fd_holder is r:
                                self._fd_holder = None  # resume capture
                            r.close()
                            continue
                        activity.touch()
                        self._handle_client_data(r, master, data)
                # reap
                done, status = os.waitpid(pid, os.WNOHANG)
                if done == pid:
                    rc = os.waitstatus_to_exitcode(status)
                    break
        except EOFError:
            _, status = os.waitpid(pid, 0)
            rc = os.waitstatus_to_exitcode(status)
        finally:
            self.record_status(rc if rc >= 0 else 128 - rc)
            for c in clients:
                c.close()
            srv.close()
            capture.close()
        return 0

    def _handle_handshake(self, c: socket.socket) -> None:
        hello = json.dumps({"ok": True, "proto": "kukeon-tty/1",
                            "pid": self.child_pid}) + "\n"
        try:
            c.sendall(hello.encode())
        except OSError:
            pass

    def _handle_client_data(self, c, master: int, data: bytes) -> None:
        # control escapes: \x00F\n requests the PTY master via SCM_RIGHTS;
        # \x00R{"rows":..,"cols":..}\n resizes; raw bytes otherwise
        if data.startswith(b"\x00F"):
            nl = data.find(b"\n")
            if nl >= 0:
                data = data[nl + 1:]
            if self._fd_holder is None:
                try:
                    socket.send_fds(c, [b"\x00FDOK\n"], [master])
                    self._fd_holder = c
                except OSError:
                    pass
            else:
                with contextlib.suppress(OSError):
                    c.sendall(b"\x00FNO\n")
        if data.startswith(b"\x00R"):
            nl = data.find(b"\n")
            if nl > 0:
                try:
                    import struct
                    import termios
                    ws = json.loads(data[2:nl])
                    fcntl.ioctl(master, termios.TIOCSWINSZ,
                                struct.pack("HHHH", ws["rows"], ws["cols"],
                                            0, 0))
                except (ValueError, OSError, KeyError):
                    pass
                data = data[nl + 1:]
        if data:
            try:
                os.write(master, data)
            except OSError:
                pass


def main() -> int:
    ap = argparse.ArgumentParser()
    ap.add_argument("--dir")
    ap.add_argument("--pause", action="store_true")
    ap.add_argument("--unshare", default="")
    ap.add_argument("--hostname", default="")
    ap.add_argument("--ns-record", default="")
    args = ap.parse_args()
    if args.pause:
        return run_pause(args.unshare, args.hostname, args.ns_record)
    shim = Shim(Path(args.dir))
    shim.enter_namespaces()

    def on_term(signum, frame):
        # forward to the workload; the wait loop observes the exit
        if shim.child_pid > 0:
            try:
                os.kill(shim.child_pid, signal.SIGTERM)
            except ProcessLookupError:
                pass

    signal.signal(signal.SIGTERM, on_term)
    shim.run_setup()
    if shim.spec.get("attachable"):
        return shim.run_attachable()
    return shim.run_plain()


if __name__ == "__main__":
    sys.exit(main())
