"""E2E matrix (VERDICT r01 item 8 — reference e2e breadth:
e2e_kuke_invalid_names_test.go, e2e_kuke_apply_test.go:238-666,
e2e_kuke_cell_test.go:85-867): invalid names per kind, per-kind apply
create/update/unchanged, container-level verbs, purge/residual matrix,
run divergence warnings — against the real daemon + shim, asserting
state + process + CLI output together."""
import json
import os
import subprocess
import time
import uuid
from pathlib import Path

import pytest
import yaml

from kukeon_amd.api import errors
from kukeon_amd.api import v1beta1 as api
from kukeon_amd.api.client import UnixClient
from kukeon_amd.controller.core import Controller
from kukeon_amd.daemon.server import Server
from kukeon_amd.runtime import proc

REPO = Path(__file__).resolve().parent.parent
KUKE = str(REPO / "bin" / "kuke")


@pytest.fixture
def harness(tmp_path):
    sock = f"/tmp/kuke-{uuid.uuid4().hex[:10]}.sock"
    ctl = Controller(str(tmp_path / "run"), gpu_devices=[])
    ctl.bootstrap()
    srv = Server(ctl, sock, reconcile_interval=0)
    srv.start()
    client = UnixClient(sock, timeout=15.0)
    yield ctl, srv, client
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


def cell_doc(name, cmd_args=("30",), rp=""):
    return api.CellDoc(
        metadata=api.Metadata(name=name),
        spec=api.CellSpec(
            realm_id="default", space_id="default", stack_id="default",
            containers=[api.ContainerSpec(
                id="main", image="busybox", command="sleep",
                args=list(cmd_args), restart_policy=rp)]))


# ---------------------------------------------------------------------------
# invalid names, every kind (reference e2e_kuke_invalid_names_test.go)
# ---------------------------------------------------------------------------
BAD_NAMES = ["UPPER", "has space", "-leading", "trailing-", "dot.dot",
             "under_score", "x" * 70, ""]


def test_invalid_names_rejected_every_kind(harness):
    ctl, srv, client = harness
    for bad in BAD_NAMES:
        with pytest.raises(errors.KukeonError):
            client.CreateRealm(doc=api.RealmDoc(
                metadata=api.Metadata(name=bad)).to_dict())
        with pytest.raises(errors.KukeonError):
            client.CreateSpace(doc=api.SpaceDoc(
                metadata=api.Metadata(name=bad),
                spec=api.SpaceSpec(realm_id="default")).to_dict())
        with pytest.raises(errors.KukeonError):
            client.CreateStack(doc=api.StackDoc(
                metadata=api.Metadata(name=bad),
                spec=api.StackSpec(realm_id="default",
                                   space_id="default")).to_dict())
        with pytest.raises(errors.KukeonError):
            client.CreateCell(doc=cell_doc(bad).to_dict())
    # nothing leaked into the store
    assert ctl.store.list_children(
        ctl.store.stack_dir("default", "default", "default")) == []


def test_invalid_names_rejected_via_apply(harness):
    ctl, srv, client = harness
    text = yaml.safe_dump(cell_doc("Bad.Name").to_dict())
    res = client.ApplyDocuments(yaml=text)
    assert res[0]["action"] == "failed"
    assert "invalid" in res[0]["error"]


# ---------------------------------------------------------------------------
# per-kind apply matrix: create / update / unchanged (apply_test.go:238+)
# ---------------------------------------------------------------------------
def test_apply_matrix_document_kinds(harness):
    ctl, srv, client = harness
    docs = {
        "Secret": {"apiVersion": "v1beta1", "kind": "Secret",
                   "metadata": {"name": "tok"},
                   "spec": {"realmId": "default", "spaceId": "default",
                            "data": {"KEY": "djE="}}},
        "Volume": {"apiVersion": "v1beta1", "kind": "Volume",
                   "metadata": {"name": "scratch"},
                   "spec": {"realmId": "default", "spaceId": "default"}},
        "CellBlueprint": {
            "apiVersion": "v1beta1", "kind": "CellBlueprint",
            "metadata": {"name": "bp"},
            "spec": {"realmId": "default", "spaceId": "default",
                     "namePrefix": "bp",
                     "template": {"spec": {
                         "realmId": "default", "spaceId": "default",
                         "stackId": "default",
                         "containers": [{"id": "main", "image": "busybox",
                                         "command": "sleep",
                                         "args": ["5"]}]}}}},
    }
    for kind, d in docs.items():
        res = client.ApplyDocuments(yaml=yaml.safe_dump(d))
        acts = [r["action"] for r in res]
        assert acts in (["created"], ["updated"]), (kind, res)
    # re-apply: document kinds are upserts -> updated (never failed)
    for kind, d in docs.items():
        res = client.ApplyDocuments(yaml=yaml.safe_dump(d))
        assert res[0]["action"] in ("updated", "unchanged")
    # multi-doc apply is kind-ordered regardless of input order
    multi = "\n---\n".join(
        yaml.safe_dump(d) for d in [
            cell_doc("orderly").to_dict(),
            {"apiVersion": "v1beta1", "kind": "Stack",
             "metadata": {"name": "newstack"},
             "spec": {"realmId": "default", "spaceId": "default"}},
        ])
    res = client.ApplyDocuments(yaml=multi)
    kinds = [r["kind"] for r in res]
    assert kinds == ["Stack", "Cell"]


def test_apply_compatible_update_preserves_process(harness):
    """Per-field diff classification end-to-end: a metadata-only change
    (restart policy knob) keeps the running PID; an argv change is
    Compatible on a child container and the apply itself converges the
    running cell (spec-hash respawn -> new PID, no manual start); a
    gpus change is Breaking -> recreate."""
    ctl, srv, client = harness
    client.CreateCell(doc=cell_doc("upd").to_dict())
    client.StartCell(realm="default", space="default", stack="default",
                     name="upd")
    cdir = ctl.store.cell_dir("default", "default", "default",
                              "upd") / "main"
    pid1 = json.loads((cdir / "runtime.json").read_text())["workloadPid"]
    # metadata-only: restart policy change -> same process
    d = cell_doc("upd", rp="on-failure").to_dict()
    res = client.ApplyDocuments(yaml=yaml.safe_dump(d))
    assert res[0]["action"] == "updated"
    client.StartCell(realm="default", space="default", stack="default",
                     name="upd")
    pid2 = json.loads((cdir / "runtime.json").read_text())["workloadPid"]
    assert pid2 == pid1  # unchanged spawn spec -> same process
    # argv change: Compatible, converged by the apply itself
    d = cell_doc("upd", cmd_args=("60",)).to_dict()
    res = client.ApplyDocuments(yaml=yaml.safe_dump(d))
    assert res[0]["action"] == "updated", res
    deadline = time.monotonic() + 10
    pid3 = pid2
    while pid3 == pid2 and time.monotonic() < deadline:
        data = ctl.store.read(cdir / "runtime.json")
        pid3 = (data or {}).get("workloadPid", pid2)
        time.sleep(0.1)
    assert pid3 != pid2  # respawned with the new argv by apply
    got = ctl.get_cell("default", "default", "default", "upd")
    assert got.spec.containers[0].args == ["60"]
    # hostNetwork flip: baked into the cell namespaces -> Breaking
    d = cell_doc("upd", cmd_args=("60",)).to_dict()
    d["spec"]["containers"][0]["hostNetwork"] = True
    res = client.ApplyDocuments(yaml=yaml.safe_dump(d))
    assert res[0]["action"] == "recreated", res
    client.KillCell(realm="default", space="default", stack="default",
                    name="upd")


# ---------------------------------------------------------------------------
# container-level verbs (reference start.go:1239 StartContainer)
# ---------------------------------------------------------------------------
def test_container_level_restart(harness):
    ctl, srv, client = harness
    client.CreateCell(doc=cell_doc("duo").to_dict())
    client.StartCell(realm="default", space="default", stack="default",
                     name="duo")
    cdir = ctl.store.cell_dir("default", "default", "default",
                              "duo") / "main"
    pid1 = json.loads((cdir / "runtime.json").read_text())["shimPid"]
    assert proc.alive(pid1)
    res = client.StopContainer(realm="default", space="default",
                               stack="default", name="duo",
                               container="main")
    deadline = time.monotonic() + 10
    while proc.alive(pid1) and time.monotonic() < deadline:
        time.sleep(0.1)
    assert not proc.alive(pid1)
    res = client.StartContainer(realm="default", space="default",
                                stack="default", name="duo",
                                container="main")
    pid2 = json.loads((cdir / "runtime.json").read_text())["shimPid"]
    assert proc.alive(pid2) and pid2 != pid1
    assert res["status"]["state"] in ("Ready", "Degraded")
    with pytest.raises(errors.KukeonError):
        client.RestartContainer(realm="default", space="default",
                                stack="default", name="duo",
                                container="ghost")
    client.KillCell(realm="default", space="default", stack="default",
                    name="duo")


# ---------------------------------------------------------------------------
# purge / residual matrix (cell_test.go purge cases)
# ---------------------------------------------------------------------------
def test_purge_matrix(harness):
    ctl, srv, client = harness
    # (a) purge of a RUNNING cell kills the process tree and removes state
    client.CreateCell(doc=cell_doc("purgy").to_dict())
    client.StartCell(realm="default", space="default", stack="default",
                     name="purgy")
    cdir = ctl.store.cell_dir("default", "default", "default", "purgy")
    pid = json.loads((cdir / "main" / "runtime.json").read_text())["shimPid"]
    client.PurgeCell(realm="default", space="default", stack="default",
                     name="purgy")
    deadline = time.monotonic() + 10
    while proc.alive(pid) and time.monotonic() < deadline:
        time.sleep(0.1)
    assert not proc.alive(pid)
    assert not cdir.exists()  # no residual dirs
    with pytest.raises(errors.KukeonError):
        client.GetCell(realm="default", space="default", stack="default",
                       name="purgy")
    # (b) graceful delete of a RUNNING cell refuses nothing: it stops
    # first, then removes
    client.CreateCell(doc=cell_doc("gone").to_dict())
    client.StartCell(realm="default", space="default", stack="default",
                     name="gone")
    client.DeleteCell(realm="default", space="default", stack="default",
                      name="gone")
    assert not ctl.store.cell_dir("default", "default", "default",
                                  "gone").exists()
    # (c) delete of a non-existent cell errors; --force tolerates
    with pytest.raises(errors.KukeonError):
        client.DeleteCell(realm="default", space="default",
                          stack="default", name="never")
    client.DeleteCell(realm="default", space="default", stack="default",
                      name="never", force=True)


def test_scope_delete_guards_and_cascade(harness):
    ctl, srv, client = harness
    client.CreateStack(doc=api.StackDoc(
        metadata=api.Metadata(name="laden"),
        spec=api.StackSpec(realm_id="default",
                           space_id="default")).to_dict())
    d = cell_doc("occupant")
    d.spec.stack_id = "laden"
    client.CreateCell(doc=d.to_dict())
    with pytest.raises(errors.KukeonError):
        client.DeleteStack(realm="default", space="default", name="laden")
    client.DeleteStack(realm="default", space="default", name="laden",
                       cascade=True)
    assert not ctl.store.stack_dir("default", "default", "laden").exists()


# ---------------------------------------------------------------------------
# kuke run divergence warning (reference run.go:446-749)
# ---------------------------------------------------------------------------
def _kuke(run, *args):
    env = dict(os.environ)
    env["KUKE_CONFIG"] = str(Path(run) / "no-kuke.yaml")
    return subprocess.run([KUKE, "--run-path", run, "--local", *args],
                          capture_output=True, text=True, timeout=60,
                          env=env)


def test_run_divergence_warning_cli(tmp_path):
    run = str(tmp_path / "run")
    r = _kuke(run, "init")
    assert r.returncode == 0, r.stderr
    f = tmp_path / "cell.yaml"
    f.write_text(yaml.safe_dump(cell_doc("diva", ("30",)).to_dict()))
    r = _kuke(run, "run", "-f", str(f), "--no-attach")
    assert r.returncode == 0, r.stderr
    # same file again: reuse, no warning
    r = _kuke(run, "run", "-f", str(f), "--no-attach")
    assert "warning" not in r.stderr.lower()
    # diverged file: warn but reuse the live cell
    f.write_text(yaml.safe_dump(cell_doc("diva", ("999",)).to_dict()))
    r = _kuke(run, "run", "-f", str(f), "--no-attach")
    assert r.returncode == 0, r.stderr
    assert "diverging spec" in r.stderr
    _kuke(run, "kill", "diva")
