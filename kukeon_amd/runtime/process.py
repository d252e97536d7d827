"""Process-cell runtime: the execution backend behind the runner.

The reference drives containerd; this environment has no container engine,
so cells are realized with raw Linux primitives behind the same interface
(fakes replace it in controller tests, exactly the reference's test
strategy): every container is a supervised process tree —

  shim (session leader, = containerd-shim analog)
    └── workload (PTY slave for attachables)

plus a `--pause` shim as the cell's root PID (kukepause analog) so the cell
has an anchor process group for teardown, cgroup membership under
kukeon/<realm>/<space>/<stack>/<cell>, durable runtime/status files for
daemon-restart-safe status derivation, and amdgpu device pinning via env +
device cgroups.
"""
from __future__ import annotations

import contextlib
import json
import os
import shlex
import subprocess
import sys
import time
from dataclasses import dataclass
from pathlib import Path
from typing import Dict, List, Optional

from kukeon_amd.api import v1beta1 as api
from kukeon_amd.runtime import proc
from kukeon_amd.runtime.cgroup import CgroupManager

ROOT_CONTAINER = "__root__"


@dataclass
class ContainerProbe:
    exists: bool = False
    running: bool = False
    pid: int = 0
    exit_code: Optional[int] = None
    started_at: str = ""
    finished_at: str = ""

    @property
    def state(self) -> str:
        if not self.exists:
            return api.STATE_NOT_CREATED
        if self.running:
            return api.STATE_READY
        if self.exit_code is None:
            return api.STATE_UNKNOWN
        return api.STATE_EXITED if self.exit_code == 0 else api.STATE_ERROR


class Runtime:
    """Interface the runner programs against (ProcessRuntime / FakeRuntime)."""

    def start_root(self, cell_dir: Path, cgroup_rel: str,
                   ns: Optional[Dict] = None) -> int:
        raise NotImplementedError

    def start_container(self, cdir: Path, spec: api.ContainerSpec,
                        env: List[str], cgroup_rel: str,
                        ns: Optional[Dict] = None) -> int:
        raise NotImplementedError

    def probe(self, cdir: Path) -> ContainerProbe:
        raise NotImplementedError

    def stop(self, cdir: Path, grace_seconds: float = 10.0) -> None:
        raise NotImplementedError

    def kill(self, cdir: Path) -> None:
        raise NotImplementedError


class ProcessRuntime(Runtime):
    def __init__(self, cgroups: Optional[CgroupManager] = None):
        self.cgroups = cgroups or CgroupManager(enabled=False)
        self._children: List[subprocess.Popen] = []

    def _reap(self) -> None:
        """Collect exited shim children so they don't linger as zombies."""
        self._children = [p for p in self._children if p.poll() is None]

    # ------------------------------------------------------------------
    def _spawn_shim(self, cdir: Path, args: List[str],
                    cgroup_rel: str) -> int:
        cdir.mkdir(parents=True, exist_ok=True)
        logf = open(cdir / "shim.log", "ab")
        env = dict(os.environ)
        pkg_root = str(Path(__file__).resolve().parent.parent.parent)
        env["PYTHONPATH"] = pkg_root + (
            ":" + env["PYTHONPATH"] if env.get("PYTHONPATH") else "")
        p = subprocess.Popen(
            [sys.executable, "-m", "kukeon_amd.tty.shim"] + args,
            stdin=subprocess.DEVNULL, stdout=logf, stderr=logf,
            start_new_session=True, close_fds=True, env=env,
            cwd=str(cdir))
        logf.close()
        self._children.append(p)
        self._reap()
        if cgroup_rel:
            self.cgroups.attach(cgroup_rel, p.pid)
        return p.pid

    def start_root(self, cell_dir: Path, cgroup_rel: str,
                   ns: Optional[Dict] = None) -> int:
        cdir = cell_dir / ROOT_CONTAINER
        cdir.mkdir(parents=True, exist_ok=True)
        args = ["--pause"]
        if ns and ns.get("unshare"):
            args += ["--unshare", ",".join(ns["unshare"]),
                     "--ns-record", str(cdir / "ns.json")]
            if ns.get("hostname"):
                args += ["--hostname", ns["hostname"]]
        pid = self._spawn_shim(cdir, args, cgroup_rel)
        if ns and ns.get("unshare"):
            # wait for the pause to record its namespace outcome so the
            # caller can plumb the netns before peers join
            for _ in range(300):
                if (cdir / "ns.json").exists() or not proc.alive(pid):
                    break
                time.sleep(0.01)
        st = proc.proc_starttime(pid)
        _write_json(cdir / "runtime.json", {
            "shimPid": pid, "shimStarttime": st, "workloadPid": pid,
            "startedAt": _now()})
        return pid

    def start_container(self, cdir: Path, spec: api.ContainerSpec,
                        env: List[str], cgroup_rel: str,
                        ns: Optional[Dict] = None) -> int:
        cdir.mkdir(parents=True, exist_ok=True)
        argv = self._argv(spec)
        spawn = {
            "argv": argv,
            "env": env,
            "cwd": spec.working_dir or str(cdir),
            "attachable": bool(spec.attachable),
            "init_script": (spec.tty.init_script if spec.tty else ""),
            "repos": [{"url": r.url, "path": r.path, "ref": r.ref}
                      for r in (spec.repos or [])],
            "git": ({"name": spec.git.name, "email": spec.git.email}
                    if spec.git else {}),
            "user": spec.user or "",
            "ns": ns or {},
        }
        # clear stale exit state from a previous run
        with contextlib.suppress(FileNotFoundError):
            (cdir / "status.json").unlink()
        with contextlib.suppress(FileNotFoundError):
            (cdir / "runtime.json").unlink()
        _write_json(cdir / "spawn.json", spawn)
        pid = self._spawn_shim(cdir, ["--dir", str(cdir)], cgroup_rel)
        # wait briefly for the shim to record the workload
        for _ in range(300):
            if (cdir / "runtime.json").exists():
                break
            if not proc.alive(pid) and not (cdir / "status.json").exists():
                log_tail = ""
                with contextlib.suppress(OSError):
                    log_tail = (cdir / "shim.log").read_text()[-500:]
                raise RuntimeError(
                    f"container shim died at startup: {log_tail}")
            time.sleep(0.01)
        return pid

    @staticmethod
    def _argv(spec: api.ContainerSpec) -> List[str]:
        if spec.command:
            argv = shlex.split(spec.command) + list(spec.args)
        elif spec.args:
            argv = list(spec.args)
        else:
            argv = ["/bin/sh"]
        return argv

    # ------------------------------------------------------------------
    def probe(self, cdir: Path) -> ContainerProbe:
        self._reap()
        rt = _read_json(cdir / "runtime.json")
        status = _read_json(cdir / "status.json")
        if rt is None and status is None:
            return ContainerProbe(exists=cdir.exists() and
                                  (cdir / "spawn.json").exists())
        p = ContainerProbe(exists=True)
        if rt:
            p.pid = rt.get("workloadPid", 0)
            p.started_at = rt.get("startedAt", "")
        if status:
            p.exit_code = status.get("exitCode")
            p.finished_at = status.get("finishedAt", "")
            p.running = False
            return p
        if rt and proc.alive(rt.get("shimPid", 0), rt.get("shimStarttime")):
            p.running = True
        return p

    def stop(self, cdir: Path, grace_seconds: float = 10.0) -> None:
        self._reap()
        rt = _read_json(cdir / "runtime.json")
        if not rt:
            return
        proc.terminate(rt.get("shimPid", 0), rt.get("shimStarttime"),
                       grace_seconds)

    def kill(self, cdir: Path) -> None:
        rt = _read_json(cdir / "runtime.json")
        if not rt:
            return
        proc.kill(rt.get("shimPid", 0), rt.get("shimStarttime"))


class FakeRuntime(Runtime):
    """In-memory runtime for controller/daemon unit tests (reference
    pattern: per-test ctr.Client fakes)."""

    def __init__(self):
        self.started: List[str] = []
        self.started_envs: List[List[str]] = []
        self.stopped: List[str] = []
        self.killed: List[str] = []
        self.states: Dict[str, ContainerProbe] = {}
        self.fail_on: Dict[str, Exception] = {}
        self._pid = 1000

    def _key(self, cdir: Path) -> str:
        return str(cdir)

    def start_root(self, cell_dir: Path, cgroup_rel: str, ns=None) -> int:
        return self.start_container(cell_dir / ROOT_CONTAINER,
                                    api.ContainerSpec(), [], cgroup_rel)

    def start_container(self, cdir, spec, env, cgroup_rel, ns=None) -> int:
        key = self._key(cdir)
        if key in self.fail_on:
            raise self.fail_on[key]
        self._pid += 1
        self.started.append(key)
        self.started_envs.append(list(env))
        self.states[key] = ContainerProbe(exists=True, running=True,
                                          pid=self._pid, started_at=_now())
        cdir.mkdir(parents=True, exist_ok=True)
        return self._pid

    def probe(self, cdir) -> ContainerProbe:
        return self.states.get(self._key(cdir), ContainerProbe())

    def stop(self, cdir, grace_seconds: float = 10.0) -> None:
        key = self._key(cdir)
        self.stopped.append(key)
        if key in self.states:
            st = self.states[key]
            if st.running:
                st.running = False
                st.exit_code = 0
                st.finished_at = _now()

    def kill(self, cdir) -> None:
        key = self._key(cdir)
        self.killed.append(key)
        if key in self.states:
            st = self.states[key]
            if st.running:
                st.running = False
                st.exit_code = 137
                st.finished_at = _now()

    # test helpers
    def mark_exited(self, cdir: Path, rc: int) -> None:
        st = self.states[self._key(Path(cdir))]
        st.running = False
        st.exit_code = rc
        st.finished_at = _now()


def _now() -> str:
    return time.strftime("%Y-%m-%dT%H:%M:%SZ", time.gmtime())


def _write_json(path: Path, data: dict) -> None:
    tmp = path.with_suffix(".tmp")
    tmp.write_text(json.dumps(data))
    os.replace(tmp, path)


def _read_json(path: Path) -> Optional[dict]:
    try:
        return json.loads(path.read_text())
    except (OSError, ValueError):
        return None
