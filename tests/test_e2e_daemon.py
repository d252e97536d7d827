"""Black-box e2e: real kukeond (unix-socket JSON-RPC), real process cells,
real PTY attach — BASELINE.json config 1 (single Cell + interactive attach
on a CPU-only host) as a test. Reference harness pattern: per-test daemon on
a short /tmp socket with --reconcile-interval 0 so tests don't race the
loop."""
import json
import os
import socket
import time
import uuid
from pathlib import Path

import pytest

from kukeon_amd.api import errors
from kukeon_amd.api import v1beta1 as api
from kukeon_amd.api.client import UnixClient
from kukeon_amd.controller.core import Controller
from kukeon_amd.daemon.server import Server
from kukeon_amd.runtime import proc
from kukeon_amd.tty import attach as attach_mod


@pytest.fixture
def harness(tmp_path):
    sock = f"/tmp/kuke-{uuid.uuid4().hex[:10]}.sock"
    ctl = Controller(str(tmp_path / "run"), gpu_devices=[])
    ctl.bootstrap()
    srv = Server(ctl, sock, reconcile_interval=0)
    srv.start()
    client = UnixClient(sock, timeout=15.0)
    yield ctl, srv, client
    # teardown: kill any cells still running
    for realm in ctl.store.list_children(ctl.store.data_root):
        for space in ctl.store.list_children(ctl.store.realm_dir(realm)):
            for stack in ctl.store.list_children(
                    ctl.store.space_dir(realm, space)):
                for cell in ctl.store.list_children(
                        ctl.store.stack_dir(realm, space, stack)):
                    try:
                        ctl.kill_cell(realm, space, stack, cell)
                    except Exception:
                        pass
    client.close()
    srv.stop()


CELL_YAML = """
apiVersion: v1beta1
kind: Cell
metadata: {name: busy}
spec:
  realmId: default
  spaceId: default
  stackId: default
  containers:
    - id: main
      image: busybox
      command: sleep
      args: ["60"]
"""


def test_apply_start_stop_cell(harness):
    ctl, srv, client = harness
    res = client.ApplyDocuments(yaml=CELL_YAML)
    assert res[0]["action"] == "created"
    cell = client.StartCell(realm="default", space="default",
                            stack="default", name="busy")
    assert cell["status"]["state"] == "Ready"
    pid = cell["status"]["containers"][0]["pid"]
    assert pid > 0 and proc.alive(pid)
    # the on-disk metadata tree has the doc (state format contract)
    meta = json.loads((ctl.store.cell_dir("default", "default", "default",
                                          "busy") / "metadata.json").read_text())
    assert meta["kind"] == "Cell" and meta["status"]["state"] == "Ready"
    cell = client.StopCell(realm="default", space="default", stack="default",
                           name="busy")
    assert cell["status"]["state"] == "Stopped"
    deadline = time.monotonic() + 5
    while proc.alive(pid) and time.monotonic() < deadline:
        time.sleep(0.05)
    assert not proc.alive(pid)
    client.DeleteCell(realm="default", space="default", stack="default",
                      name="busy")
    with pytest.raises(errors.CellNotFound):
        client.GetCell(realm="default", space="default", stack="default",
                       name="busy")


def test_interactive_pty_attach(harness):
    ctl, srv, client = harness
    doc = api.CellDoc(
        metadata=api.Metadata(name="shelly"),
        spec=api.CellSpec(
            realm_id="default", space_id="default", stack_id="default",
            containers=[api.ContainerSpec(
                id="term", image="busybox", command="/bin/sh",
                args=["-i"], attachable=True)]))
    client.CreateCell(doc=doc.to_dict())
    client.StartCell(realm="default", space="default", stack="default",
                     name="shelly")
    res = client.AttachContainer(realm="default", space="default",
                                 stack="default", name="shelly")
    path = res["hostSocketPath"]
    deadline = time.monotonic() + 5
    while not attach_mod.ping(path) and time.monotonic() < deadline:
        time.sleep(0.1)
    assert attach_mod.ping(path)

    # drive the PTY over the socket directly (no raw-mode tty in pytest)
    s = socket.socket(socket.AF_UNIX, socket.SOCK_STREAM)
    s.connect(path)
    s.settimeout(5.0)
    hello = s.recv(4096)
    assert b"kukeon-tty/1" in hello
    s.sendall(b"echo m-$((40+2))\n")
    buf = b""
    deadline = time.monotonic() + 5
    while b"m-42" not in buf and time.monotonic() < deadline:
        try:
            buf += s.recv(4096)
        except socket.timeout:
            break
    assert b"m-42" in buf
    s.close()
    # detach left the workload running; capture file holds the transcript
    res = client.LogPath(realm="default", space="default", stack="default",
                         name="shelly")
    assert b"m-42" in Path(res["path"]).read_bytes()
    client.KillCell(realm="default", space="default", stack="default",
                    name="shelly")


def test_restart_policy_real_process(harness):
    ctl, srv, client = harness
    doc = api.CellDoc(
        metadata=api.Metadata(name="crashy"),
        spec=api.CellSpec(
            realm_id="default", space_id="default", stack_id="default",
            containers=[api.ContainerSpec(
                id="main", image="busybox", command="/bin/sh",
                args=["-c", "exit 3"], restart_policy="on-failure",
                restart_backoff_seconds=0, restart_max_retries=2)]))
    client.CreateCell(doc=doc.to_dict())
    client.StartCell(realm="default", space="default", stack="default",
                     name="crashy")
    # let it crash, then reconcile until the retry cap trips
    deadline = time.monotonic() + 15
    state = ""
    while time.monotonic() < deadline:
        time.sleep(0.3)
        client.ReconcileCells()
        cell = client.GetCell(realm="default", space="default",
                              stack="default", name="crashy")
        state = cell["status"]["state"]
        if state == "Error":
            break
    assert state == "Error"
    assert cell["status"]["containers"][0]["restartCount"] == 2
    assert cell["status"]["containers"][0]["exitCode"] == 3


def test_clean_exit_and_autodelete(harness):
    ctl, srv, client = harness
    doc = api.CellDoc(
        metadata=api.Metadata(name="oneshot"),
        spec=api.CellSpec(
            realm_id="default", space_id="default", stack_id="default",
            auto_delete=True,
            containers=[api.ContainerSpec(
                id="main", image="busybox", command="/bin/true")]))
    client.CreateCell(doc=doc.to_dict())
    client.StartCell(realm="default", space="default", stack="default",
                     name="oneshot")
    deadline = time.monotonic() + 10
    gone = False
    while time.monotonic() < deadline:
        time.sleep(0.3)
        client.ReconcileCells()
        try:
            client.GetCell(realm="default", space="default", stack="default",
                           name="oneshot")
        except errors.CellNotFound:
            gone = True
            break
    assert gone, "AutoDelete cell was not cleaned up"


def test_state_survives_daemon_restart(harness, tmp_path):
    ctl, srv, client = harness
    client.ApplyDocuments(yaml=CELL_YAML)
    client.StartCell(realm="default", space="default", stack="default",
                     name="busy")
    pid = client.GetCell(realm="default", space="default", stack="default",
                         name="busy")["status"]["containers"][0]["pid"]
    srv.stop()
    client.close()
    # a fresh daemon over the same run path re-derives live state
    ctl2 = Controller(str(ctl.run_path), gpu_devices=[])
    sock2 = f"/tmp/kuke-{uuid.uuid4().hex[:10]}.sock"
    srv2 = Server(ctl2, sock2, reconcile_interval=0)
    srv2.start()
    c2 = UnixClient(sock2, timeout=10.0)
    try:
        c2.ReconcileCells()
        cell = c2.GetCell(realm="default", space="default", stack="default",
                          name="busy")
        assert cell["status"]["state"] == "Ready"
        assert cell["status"]["containers"][0]["pid"] == pid
        assert proc.alive(pid)
        c2.KillCell(realm="default", space="default", stack="default",
                    name="busy")
    finally:
        c2.close()
        srv2.stop()


def test_error_mapping_over_rpc(harness):
    ctl, srv, client = harness
    with pytest.raises(errors.CellNotFound):
        client.GetCell(realm="default", space="default", stack="default",
                       name="nope")
    with pytest.raises(errors.RealmNotFound):
        client.GetRealm(name="ghost")


def test_cell_metrics_live(harness):
    """`kuke top` backing verb: live /proc metrics for a running cell."""
    ctl, srv, client = harness
    client.ApplyDocuments(yaml=CELL_YAML)
    client.StartCell(realm="default", space="default", stack="default",
                     name="busy")
    m = client.CellMetrics(realm="default", space="default",
                           stack="default", name="busy")
    assert m["cell"] == "busy"
    running = [c for c in m["containers"].values() if c.get("running")]
    assert running, m
    assert all(c["rssBytes"] > 0 and c["threads"] >= 1 for c in running)
    assert m["total"]["rssBytes"] > 0
    client.KillCell(realm="default", space="default", stack="default",
                    name="busy")


def test_purge_recovers_from_corrupted_metadata(harness):
    """purge must clear residual state even when metadata.json is damaged
    (the reference's purge contract: force residual-state removal)."""
    ctl, srv, client = harness
    client.ApplyDocuments(yaml=CELL_YAML)
    cell = client.StartCell(realm="default", space="default",
                            stack="default", name="busy")
    pid = cell["status"]["containers"][0]["pid"]
    assert proc.alive(pid)
    # corrupt the cell document on disk
    mpath = ctl.store.cell_dir("default", "default", "default",
                               "busy") / "metadata.json"
    mpath.write_text("{ not json !!!")
    # normal get now fails...
    with pytest.raises(Exception):
        client.GetCell(realm="default", space="default", stack="default",
                       name="busy")
    # ...but purge clears the process and the residual tree
    client.PurgeCell(realm="default", space="default", stack="default",
                     name="busy")
    deadline = time.monotonic() + 5
    while proc.alive(pid) and time.monotonic() < deadline:
        time.sleep(0.05)
    assert not proc.alive(pid)
    assert not ctl.store.cell_dir("default", "default", "default",
                                  "busy").exists()


def test_background_reconcile_loop_restarts(tmp_path):
    """The daemon's own periodic loop (no manual ReconcileCells calls)
    must pick up a crash and restart the container."""
    import uuid as _uuid

    sock = f"/tmp/kuke-bg-{_uuid.uuid4().hex[:8]}.sock"
    ctl = Controller(str(tmp_path / "run"), gpu_devices=[])
    ctl.bootstrap()
    srv = Server(ctl, sock, reconcile_interval=0.1)
    srv.start()
    client = UnixClient(sock, timeout=15.0)
    try:
        doc = api.CellDoc(
            metadata=api.Metadata(name="bg"),
            spec=api.CellSpec(
                realm_id="default", space_id="default", stack_id="default",
                containers=[api.ContainerSpec(
                    id="main", image="busybox", command="/bin/sh",
                    args=["-c", "sleep 60"], restart_policy="always",
                    restart_backoff_seconds=0)]))
        client.CreateCell(doc=doc.to_dict())
        cell = client.StartCell(realm="default", space="default",
                                stack="default", name="bg")
        pid = cell["status"]["containers"][0]["pid"]
        os.kill(pid, 9)
        deadline = time.monotonic() + 15
        restarted = False
        while time.monotonic() < deadline:
            time.sleep(0.2)
            cur = client.GetCell(realm="default", space="default",
                                 stack="default", name="bg")
            cs = cur["status"]["containers"][0]
            if cs.get("restartCount", 0) >= 1 and cs["pid"] != pid and \
                    proc.alive(cs["pid"]):
                restarted = True
                break
        assert restarted, "background loop never restarted the container"
        client.KillCell(realm="default", space="default", stack="default",
                        name="bg")
    finally:
        client.close()
        srv.stop()


def test_container_repo_setup(harness, tmp_path):
    """The shim clones declared repos before the workload (kuketty
    runOn:create parity) and the reconcile loop surfaces per-repo state
    into ContainerStatus.repos."""
    import subprocess

    ctl, srv, client = harness
    src = tmp_path / "srcrepo"
    src.mkdir()
    subprocess.run(["git", "init", "-q", str(src)], check=True)
    (src / "hello.txt").write_text("hi")
    subprocess.run(["git", "-C", str(src), "add", "."], check=True)
    subprocess.run(["git", "-C", str(src), "-c", "user.name=t",
                    "-c", "user.email=t@x", "commit", "-qm", "init"],
                   check=True)
    doc = api.CellDoc(
        metadata=api.Metadata(name="repocell"),
        spec=api.CellSpec(
            realm_id="default", space_id="default", stack_id="default",
            containers=[api.ContainerSpec(
                id="main", image="busybox", command="sleep", args=["30"],
                repos=[api.ContainerRepo(url=str(src), path="work")],
                git=api.ContainerGit(name="agent", email="a@x"))]))
    client.CreateCell(doc=doc.to_dict())
    client.StartCell(realm="default", space="default", stack="default",
                     name="repocell")
    cdir = ctl.store.cell_dir("default", "default", "default",
                              "repocell") / "main"
    deadline = time.monotonic() + 20
    while time.monotonic() < deadline and not (cdir / "setup.json").exists():
        time.sleep(0.1)
    assert (cdir / "work" / ".git").is_dir()
    assert (cdir / "work" / "hello.txt").read_text() == "hi"
    cell = client.ReconcileCells() and client.GetCell(
        realm="default", space="default", stack="default", name="repocell")
    repos = cell["status"]["containers"][0]["repos"]
    assert repos and repos[0]["state"] == "cloned", repos
    client.KillCell(realm="default", space="default", stack="default",
                    name="repocell")


def test_scm_rights_fd_attach(harness):
    """SCM_RIGHTS fast path (reference cmd/kuketty/main.go:17-30): the
    first attacher gets the PTY master fd and pumps bytes with no shim
    relay; a concurrent second attacher falls back to the relay path;
    capture resumes once the fd holder detaches."""
    import socket as sk

    ctl, srv, client = harness
    doc = api.CellDoc(
        metadata=api.Metadata(name="fdcell"),
        spec=api.CellSpec(
            realm_id="default", space_id="default", stack_id="default",
            containers=[api.ContainerSpec(
                id="term", image="busybox", command="/bin/sh",
                args=["-i"], attachable=True)]))
    client.CreateCell(doc=doc.to_dict())
    client.StartCell(realm="default", space="default", stack="default",
                     name="fdcell")
    res = client.AttachContainer(realm="default", space="default",
                                 stack="default", name="fdcell")
    path = res["hostSocketPath"]
    deadline = time.monotonic() + 5
    while not attach_mod.ping(path) and time.monotonic() < deadline:
        time.sleep(0.1)

    s1 = sk.socket(sk.AF_UNIX, sk.SOCK_STREAM)
    s1.connect(path)
    s1.settimeout(5.0)
    assert b"kukeon-tty/1" in s1.recv(4096)
    mfd = attach_mod._request_fd(s1)
    assert mfd is not None  # got the PTY master via SCM_RIGHTS
    # the DIRECT pump: write to the master fd, read the echo back
    os.write(mfd, b"echo fd-$((40+2))\n")
    buf = b""
    deadline = time.monotonic() + 5
    while b"fd-42" not in buf and time.monotonic() < deadline:
        import select as _sel
        rd, _, _ = _sel.select([mfd], [], [], 0.5)
        if rd:
            buf += os.read(mfd, 4096)
    assert b"fd-42" in buf

    # second concurrent attacher is refused the fd (falls back to relay)
    s2 = sk.socket(sk.AF_UNIX, sk.SOCK_STREAM)
    s2.connect(path)
    s2.settimeout(5.0)
    s2.recv(4096)
    assert attach_mod._request_fd(s2) is None
    s2.close()

    # detach the fd holder; the shim resumes the relay + capture
    os.close(mfd)
    s1.close()
    time.sleep(0.3)
    s3 = sk.socket(sk.AF_UNIX, sk.SOCK_STREAM)
    s3.connect(path)
    s3.settimeout(5.0)
    s3.recv(4096)
    s3.sendall(b"echo post-$((40+3))\n")
    buf = b""
    deadline = time.monotonic() + 5
    while b"post-43" not in buf and time.monotonic() < deadline:
        try:
            buf += s3.recv(4096)
        except OSError:
            break
    assert b"post-43" in buf
    s3.close()
    # capture holds the post-detach output (recorded by the shim again)
    res = client.LogPath(realm="default", space="default", stack="default",
                         name="fdcell")
    assert b"post-43" in Path(res["path"]).read_bytes()
    client.KillCell(realm="default", space="default", stack="default",
                    name="fdcell")


def test_run_from_blueprint_rpc(harness):
    """RunFromBlueprint over the wire: materialize + start + lineage
    labels (the `kuke run -b` backing verb)."""
    ctl, srv, client = harness
    bp = {
        "apiVersion": "v1beta1", "kind": "CellBlueprint",
        "metadata": {"name": "bp-sleep"},
        "spec": {
            "realmId": "default", "spaceId": "default",
            "namePrefix": "bp",
            "params": [{"name": "SECS", "default": "30"}],
            "template": {
                "kind": "Cell",
                "spec": {"realmId": "default", "spaceId": "default",
                         "stackId": "default",
                         "containers": [{"id": "main",
                                         "command": "sleep",
                                         "args": ["${SECS}"]}]}},
        },
    }
    client.PutBlueprint(doc=bp)
    cell = client.RunFromBlueprint(realm="default", space="default",
                                   stack="default", blueprint="bp-sleep",
                                   params={"SECS": "60"}, env=[], name="")
    name = cell["metadata"]["name"]
    assert name.startswith("bp-")
    assert cell["status"]["state"] == "Ready"
    got = client.GetCell(realm="default", space="default",
                         stack="default", name=name)
    assert got["spec"]["containers"][0]["args"] == ["60"]
    assert got["metadata"]["labels"].get("kukeon.io/blueprint") == \
        "bp-sleep"
    client.KillCell(realm="default", space="default", stack="default",
                    name=name)


def test_run_from_config_rpc(harness):
    """RunFromConfig over the wire: blueprint binding + values overlay +
    provenance stamping (the `kuke run -c` backing verb)."""
    ctl, srv, client = harness
    client.PutBlueprint(doc={
        "apiVersion": "v1beta1", "kind": "CellBlueprint",
        "metadata": {"name": "bp-c"},
        "spec": {"realmId": "default", "spaceId": "default",
                 "namePrefix": "cfged",
                 "params": [{"name": "SECS", "default": "30"}],
                 "template": {"kind": "Cell",
                              "spec": {"realmId": "default",
                                       "spaceId": "default",
                                       "stackId": "default",
                                       "containers": [
                                           {"id": "main",
                                            "command": "sleep",
                                            "args": ["${SECS}"]}]}}}})
    client.PutConfig(doc={
        "apiVersion": "v1beta1", "kind": "CellConfig",
        "metadata": {"name": "cfg-fast"},
        "spec": {"realmId": "default", "spaceId": "default",
                 "blueprint": "bp-c", "values": {"SECS": "45"}}})
    cell = client.RunFromConfig(realm="default", space="default",
                                stack="default", config="cfg-fast",
                                params={}, name="")
    name = cell["metadata"]["name"]
    got = client.GetCell(realm="default", space="default",
                         stack="default", name=name)
    assert got["spec"]["containers"][0]["args"] == ["45"]
    prov = got["spec"].get("provenance", {})
    assert (prov.get("bindingKind"), prov.get("bindingRef")) == \
        ("config", "cfg-fast")
    assert prov.get("params", {}).get("SECS") == "45"
    client.KillCell(realm="default", space="default", stack="default",
                    name=name)


def test_container_user_privilege_drop(harness):
    """ContainerSpec.user: the workload execs with the dropped identity
    (root-only enforcement; reference OCI Process.user)."""
    import os as _os

    if _os.geteuid() != 0:
        pytest.skip("needs root to drop privileges")
    ctl, srv, client = harness
    import tempfile
    out = Path(tempfile.mkdtemp(prefix="kuke-uid-"))
    _os.chmod(out, 0o1777)   # the dropped user must be able to write
    doc = {
        "apiVersion": "v1beta1", "kind": "Cell",
        "metadata": {"name": "dropped"},
        "spec": {"realmId": "default", "spaceId": "default",
                 "stackId": "default",
                 "containers": [{"id": "main", "command": "sh",
                                 "user": "65534",   # nobody
                                 "args": ["-c",
                                          f"id -u > {out}/uid.txt; "
                                          "sleep 30"]}]},
    }
    client.CreateCell(doc=doc)
    client.StartCell(realm="default", space="default", stack="default",
                     name="dropped")
    deadline = time.monotonic() + 10
    while not (out / "uid.txt").exists() and time.monotonic() < deadline:
        time.sleep(0.05)
    time.sleep(0.1)
    assert (out / "uid.txt").read_text().strip() == "65534"
    client.KillCell(realm="default", space="default", stack="default",
                    name="dropped")


def test_space_defaults_user_inheritance(harness):
    """SpaceDefaults.container.user applies to containers that leave
    `user` unset (reference space.go:83-107 isolation inheritance)."""
    import os as _os

    if _os.geteuid() != 0:
        pytest.skip("needs root to drop privileges")
    ctl, srv, client = harness
    client.ApplyDocuments(yaml="""
apiVersion: v1beta1
kind: Space
metadata: {name: lowpriv}
spec:
  realmId: default
  defaults:
    container: {user: "65534"}
""")
    client.ApplyDocuments(yaml="""
apiVersion: v1beta1
kind: Stack
metadata: {name: s}
spec: {realmId: default, spaceId: lowpriv}
""")
    import tempfile
    out = Path(tempfile.mkdtemp(prefix="kuke-sdu-"))
    _os.chmod(out, 0o1777)
    client.CreateCell(doc={
        "apiVersion": "v1beta1", "kind": "Cell",
        "metadata": {"name": "inh"},
        "spec": {"realmId": "default", "spaceId": "lowpriv",
                 "stackId": "s",
                 "containers": [{"id": "main", "command": "sh",
                                 "args": ["-c",
                                          f"id -u > {out}/uid.txt; "
                                          "sleep 30"]}]}})
    client.StartCell(realm="default", space="lowpriv", stack="s",
                     name="inh")
    deadline = time.monotonic() + 10
    while not (out / "uid.txt").exists() and time.monotonic() < deadline:
        time.sleep(0.05)
    time.sleep(0.1)
    assert (out / "uid.txt").read_text().strip() == "65534"
    client.KillCell(realm="default", space="lowpriv", stack="s",
                    name="inh")
