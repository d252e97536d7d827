"""Aux-subsystem tests mirroring the reference's injectable-runner style
(iptables paths asserted by command line, frozen clocks,
fake samplers — no root, no iptables, no amdgpu needed)."""
import logging

import pytest

from kukeon_amd.api import errors
from kukeon_amd.api import v1beta1 as api
from kukeon_amd.netpolicy import (IptablesEnforcer, MASTER_CHAIN, Policy,
                                  Rule, build_policy, space_chain)
from kukeon_amd.runtime import proc
from kukeon_amd.runtime.devices import GPUAllocator, visible_devices_env
from kukeon_amd.runtime.diskpressure import Guard, Sample


class RecordingRunner:
    """Every command 'exists' checks fail (-C/-nL return 1) so the
    enforcer takes its create path; all other commands succeed."""

    def __init__(self):
        self.cmds = []

    def __call__(self, args):
        self.cmds.append(" ".join(args))
        return 1 if args[1] in ("-C", "-nL") else 0


def test_egress_enforcer_command_lines():
    r = RecordingRunner()
    enf = IptablesEnforcer(runner=r)
    pol = Policy(default_deny=True,
                 rules=[Rule(cidr="140.82.0.0/16", ports=[443]),
                        Rule(cidr="10.0.0.5/32", ports=[])])
    enf.apply("default", "dev", "10.88.3.0/24", pol)
    chain = space_chain("default", "dev")
    joined = "\n".join(r.cmds)
    assert f"iptables -N {MASTER_CHAIN}" in joined
    assert "iptables -I FORWARD -j KUKEON-EGRESS" in joined
    assert f"iptables -N {chain}" in joined
    assert f"iptables -A {MASTER_CHAIN} -s 10.88.3.0/24 -j {chain}" in joined
    assert (f"iptables -A {chain} -d 140.82.0.0/16 -p tcp --dport 443 "
            "-j ACCEPT") in joined
    assert f"iptables -A {chain} -d 10.0.0.5/32 -j ACCEPT" in joined
    # established return traffic precedes the default drop
    est = joined.index("ESTABLISHED,RELATED -j ACCEPT")
    assert f"iptables -A {chain} -j DROP" in joined
    assert est < joined.index(f"iptables -A {chain} -j DROP")
    r2 = RecordingRunner()
    IptablesEnforcer(runner=r2).remove("default", "dev")
    assert f"iptables -X {chain}" in "\n".join(r2.cmds)


def test_build_policy_resolves_hosts():
    def fake_resolver(host, port):
        assert host == "api.example.com"
        return [(2, 1, 6, "", ("1.2.3.4", 0)), (2, 1, 6, "", ("1.2.3.4", 0)),
                (10, 1, 6, "", ("::1", 0, 0, 0))]

    eg = api.EgressPolicy(default="deny", allow=[
        api.EgressAllowRule(host="api.example.com", ports=[443]),
        api.EgressAllowRule(cidr="9.9.9.0/24")])
    p = build_policy(eg, resolver=fake_resolver)
    assert p.default_deny
    # v6 and duplicate addresses dropped; host becomes /32
    assert [r.cidr for r in p.rules] == ["1.2.3.4/32", "9.9.9.0/24"]


def test_gpu_allocator_persistence_and_exhaustion(tmp_path):
    path = str(tmp_path / "gpus.json")
    a = GPUAllocator(path, devices=[0, 1, 2, 3])
    assert a.allocate("sess-a", 2) == [0, 1]
    assert a.allocate("sess-a", 2) == [0, 1]  # idempotent per owner
    assert a.allocate("sess-b", 1) == [2]
    with pytest.raises(errors.GPUUnavailable):
        a.allocate("sess-c", 2)
    # persisted: a fresh allocator sees the same assignments
    b = GPUAllocator(path, devices=[0, 1, 2, 3])
    assert b.free == [3]
    b.release("sess-a")
    assert b.free == [0, 1, 3]
    assert GPUAllocator(path, devices=[0, 1, 2, 3]).free == [0, 1, 3]
    env = dict(e.split("=", 1) for e in visible_devices_env([2]))
    assert env["ROCR_VISIBLE_DEVICES"] == "2"


def test_disk_pressure_block_warn_and_ratelimit(caplog):
    t = [0.0]
    pct = [50.0]

    def sampler(path):
        return Sample(total_bytes=100, used_bytes=int(pct[0]))

    g = Guard("/x", warn_percent=85.0, block_percent=95.0, sampler=sampler,
              now_fn=lambda: t[0])
    g.check()  # calm
    pct[0] = 96.0
    with pytest.raises(errors.DiskPressure):
        g.check()
    g.check(ignore=True)  # ignoreDiskPressure override
    pct[0] = 90.0
    with caplog.at_level(logging.WARNING, logger="kukeon.diskpressure"):
        t[0] = 1000.0
        g.check()
        t[0] = 1060.0
        g.check()  # inside the 5-minute re-emit window: suppressed
        warns = [r for r in caplog.records if "warn threshold" in r.message]
        assert len(warns) == 1
        t[0] = 1000.0 + 301.0
        g.check()
        warns = [r for r in caplog.records if "warn threshold" in r.message]
        assert len(warns) == 2


def test_proc_metrics_self():
    import os
    m = proc.metrics(os.getpid())
    assert m is not None
    assert m["rssBytes"] > 1 << 20 and m["threads"] >= 1
    assert m["cpuSeconds"] >= 0.0
    assert proc.metrics(2 ** 22 + 12345) is None  # unlikely pid


def test_cgroup_v2_tree_against_injected_root(tmp_path):
    """CgroupManager drives a v2 tree: chain creation, subtree-controller
    delegation up the ancestors, memory.max / cpu.weight knobs, cleanup —
    asserted against an injected fs root (no root privileges)."""
    from kukeon_amd.runtime.cgroup import CgroupManager

    root = tmp_path / "cg"
    root.mkdir()
    (root / "cgroup.controllers").write_text("cpu memory io pids hugetlb")
    m = CgroupManager(root=str(root))
    assert m.mode == "v2"
    created = m.create("default/dev/stack/cell-a")
    assert created and (root / "kukeon" / "default" / "dev" / "stack" /
                        "cell-a").is_dir()
    # delegation wrote the resource subset (not hugetlb) on the ancestors
    sub = (root / "kukeon" / "default" / "dev" /
           "cgroup.subtree_control").read_text()
    assert "+cpu" in sub and "+memory" in sub and "hugetlb" not in sub
    m.set_memory_limit("default/dev/stack/cell-a", 1 << 30)
    assert (root / "kukeon" / "default" / "dev" / "stack" / "cell-a" /
            "memory.max").read_text() == str(1 << 30)
    m.set_cpu_shares("default/dev/stack/cell-a", 1024)
    w = int((root / "kukeon" / "default" / "dev" / "stack" / "cell-a" /
             "cpu.weight").read_text())
    assert 1 <= w <= 10000
    # on real cgroupfs the knob files are virtual and vanish with rmdir;
    # on the injected plain fs they must be cleared first
    leaf = root / "kukeon" / "default" / "dev" / "stack" / "cell-a"
    for f in leaf.iterdir():
        f.unlink()
    m.delete("default/dev/stack/cell-a")
    assert not leaf.exists()


def test_store_cas_multiprocess_no_lost_updates(tmp_path):
    """Hard part #1: the flock+CAS envelope under REAL
    multi-process contention — 4 processes each add their increments to
    a counter field via optimistic CAS with retry; no update may be
    lost and generations must count every successful write."""
    import subprocess
    import sys as _sys

    doc = tmp_path / "metadata.json"
    worker = tmp_path / "worker.py"
    worker.write_text(f'''
import sys
sys.path.insert(0, {str(repr(str(__import__('pathlib').Path(__file__).resolve().parent.parent)))})
from pathlib import Path
from kukeon_amd.state.store import Store
from kukeon_amd.api import errors
store = Store(sys.argv[1])
path = Path(sys.argv[2])
wid, n = int(sys.argv[3]), int(sys.argv[4])
done = 0
while done < n:
    cur = store.read(path) or {{"metadata": {{"generation": 0}},
                               "count": 0, "by": {{}}}}
    gen = cur["metadata"].get("generation", 0)
    cur["count"] = cur.get("count", 0) + 1
    by = cur.setdefault("by", {{}})
    by[str(wid)] = by.get(str(wid), 0) + 1
    try:
        store.write_cas(path, cur, expected_generation=gen)
        done += 1
    except errors.StaleResource:
        pass  # lost the race: retry with a fresh read
''')
    store_root = str(tmp_path / "run")
    from kukeon_amd.state.store import Store
    Store(store_root)  # create root
    procs = [subprocess.Popen([_sys.executable, str(worker), store_root,
                               str(doc), str(i), "40"])
             for i in range(4)]
    for p in procs:
        assert p.wait(timeout=120) == 0
    final = Store(store_root).read(doc)
    assert final["count"] == 160, final
    assert final["metadata"]["generation"] == 160
    assert sorted(final["by"].items()) == [(str(i), 40) for i in range(4)]


def test_subnet_allocator_persistence_release_exhaustion(tmp_path):
    from kukeon_amd.state.store import Store
    from kukeon_amd.controller.subnet import SubnetAllocator

    store = Store(str(tmp_path / "run"))
    a = SubnetAllocator(store, honor_host_routes=False)
    s1 = a.allocate("default", "alpha")
    s2 = a.allocate("default", "beta")
    assert s1 != s2 and s1.endswith(".0/24")
    assert a.allocate("default", "alpha") == s1   # idempotent
    # persisted: a fresh allocator over the same tree sees it
    b = SubnetAllocator(Store(str(tmp_path / "run")),
                        honor_host_routes=False)
    assert b.lookup("default", "alpha") == s1
    b.release("default", "alpha")
    assert b.lookup("default", "alpha") is None
    s3 = b.allocate("default", "gamma")
    assert s3 == s1  # the freed /24 is reusable
    # exhaustion: burn the rest of the 256-slot pool (beta + gamma held)
    for i in range(254):
        b.allocate("default", f"sp{i}")
    with pytest.raises(errors.KukeonError, match="exhausted"):
        b.allocate("default", "overflow")


def test_subnet_allocator_skips_live_host_routes(tmp_path):
    """Octets with live host routes (e.g. a crashed daemon's bridge)
    must not be re-allocated: colliding subnets blackhole replies."""
    from kukeon_amd.state.store import Store
    from kukeon_amd.controller.subnet import SubnetAllocator

    rt = tmp_path / "route"
    # fake /proc/net/route with 10.88.0.0/24 and 10.88.1.0/24 held
    rt.write_text(
        "Iface Dest Gateway Flags RefCnt Use Metric Mask MTU Win IRTT\n"
        "k-dead 0058580A 00000000 0001 0 0 0 00FFFFFF 0 0 0\n"
        .replace("0058580A", "0058580A") +
        "k-dead2 0158580A 00000000 0001 0 0 0 00FFFFFF 0 0 0\n")
    # hex little-endian: 0A58_5800 ... craft via real pack below instead
    import socket as _s, struct as _st
    def hx(ip):
        return "%08X" % _st.unpack("<I", _s.inet_aton(ip))[0]
    rt.write_text(
        "Iface Dest Gateway Flags RefCnt Use Metric Mask MTU Win IRTT\n"
        f"k-dead {hx('10.88.0.0')} 00000000 0001 0 0 0 00FFFFFF 0 0 0\n"
        f"k-dead2 {hx('10.88.1.0')} 00000000 0001 0 0 0 00FFFFFF 0 0 0\n")
    a = SubnetAllocator(Store(str(tmp_path / "run")), route_file=str(rt))
    assert a.allocate("default", "alpha") == "10.88.2.0/24"


def test_gpu_allocator_top_up_after_spec_growth(tmp_path):
    """An owner whose spec grows to more GPUs gets topped up from free,
    never a stale shorter list."""
    path = str(tmp_path / "gpus.json")
    a = GPUAllocator(path, devices=[0, 1, 2, 3])
    assert a.allocate("sess-a", 1) == [0]
    # spec change: now wants 3 — keeps its existing GPU, gains two
    assert a.allocate("sess-a", 3) == [0, 1, 2]
    # persisted
    assert GPUAllocator(path, devices=[0, 1, 2, 3]).allocate(
        "sess-a", 3) == [0, 1, 2]
    # impossible top-up raises, assignment unchanged
    a.allocate("sess-b", 1)
    with pytest.raises(errors.GPUUnavailable):
        a.allocate("sess-a", 5)
    assert a.allocate("sess-a", 3) == [0, 1, 2]


def test_store_delete_leaves_lock_tombstone(tmp_path):
    """delete() must not unlink the sidecar lock file: a process blocked
    in flock on the old inode would otherwise hold a stale lock
    concurrently with a new holder."""
    from kukeon_amd.state.store import Store

    store = Store(str(tmp_path / "run"))
    doc = tmp_path / "run" / "thing.json"
    store.create_exclusive(doc, {"spec": {}})
    lockp = tmp_path / "run" / "thing.json.lock"
    assert lockp.exists()
    assert store.delete(doc)
    assert not doc.exists()
    assert lockp.exists()  # tombstone stays
    # doc can be recreated and the same lock inode still guards it
    ino_before = lockp.stat().st_ino
    store.create_exclusive(doc, {"spec": {}})
    assert lockp.stat().st_ino == ino_before


def test_apischeme_version_translation():
    """api/scheme.py: v1alpha1 wire docs up-convert through the seam
    (scope keys, flat container shorthand, enum renames) and internal
    docs export back down with named field loss."""
    from kukeon_amd.api import scheme
    from kukeon_amd.api import v1beta1 as api

    alpha_cell = {
        "apiVersion": "v1alpha1", "kind": "Cell",
        "metadata": {"name": "old-style"},
        "spec": {"realm": "prod", "space": "web", "stack": "app",
                 "autoRemove": True, "image": "tool", "command": "sleep",
                 "args": ["5"]}}
    doc = scheme.normalize_doc(alpha_cell)
    assert doc.api_version == "v1beta1"
    assert doc.spec.realm_id == "prod" and doc.spec.stack_id == "app"
    assert doc.spec.auto_delete is True
    assert len(doc.spec.containers) == 1
    assert doc.spec.containers[0].command == "sleep"
    assert doc.spec.containers[0].id == "main"

    alpha_space = {
        "apiVersion": "kukeon.io/v1alpha1", "kind": "Space",
        "metadata": {"name": "web"},
        "spec": {"realm": "prod",
                 "network": {"egress": {"mode": "deny",
                                        "allow": [{"cidr": "10.0.0.0/8"}]}}}}
    sp = scheme.normalize_doc(alpha_space)
    assert sp.spec.network.egress.default == "deny"

    # defaulting: scope falls back to the default hierarchy
    bare = scheme.normalize_doc({"kind": "Cell",
                                 "metadata": {"name": "c"},
                                 "spec": {"containers": [
                                     {"image": "x", "command": "sleep"}]}})
    assert bare.spec.realm_id == "default"
    assert bare.spec.containers[0].id == "main"

    # downgrade export names what it drops
    beta = api.CellDoc(
        metadata=api.Metadata(name="c"),
        spec=api.CellSpec(realm_id="r", space_id="s", stack_id="t",
                          containers=[api.ContainerSpec(
                              id="main", image="x", command="sleep",
                              restart_policy="on-failure",
                              repos=[api.ContainerRepo(url="u")])]))
    wire, lost = scheme.to_wire(beta, "v1alpha1")
    assert wire["spec"]["realm"] == "r"
    assert wire["spec"]["containers"][0]["restartPolicy"] == "onFailure"
    assert any("repos" in f for f in lost)
    # unsupported version rejected
    with pytest.raises(errors.KukeonError):
        scheme.normalize({"apiVersion": "v2", "kind": "Cell",
                          "metadata": {"name": "x"}, "spec": {}})


def test_apply_accepts_v1alpha1_documents(tmp_path):
    """The whole apply pipeline accepts legacy-version YAML through the
    scheme seam."""
    from kukeon_amd.controller.core import Controller
    from kukeon_amd.runtime.process import FakeRuntime

    ctl = Controller(str(tmp_path / "run"), runtime=FakeRuntime(),
                     gpu_devices=[])
    ctl.bootstrap()
    text = """
apiVersion: v1alpha1
kind: Cell
metadata: {name: legacy}
spec:
  realm: default
  space: default
  stack: default
  image: busybox
  command: sleep
  args: ["9"]
"""
    res = ctl.apply_documents(text)
    assert res[0].action == "created", (res[0].action, res[0].error)
    doc = ctl.get_cell("default", "default", "default", "legacy")
    assert doc.spec.containers[0].args == ["9"]


def test_sysuser_group_and_ownership(tmp_path):
    """sysuser: ensure_group + chown_tree + socket perms (reference
    internal/sysuser EnsureUserGroup/ChownTreeAndChmodSkip). Uses a
    throwaway group name when running as root; asserts graceful None
    degrade otherwise."""
    import os
    from kukeon_amd.runtime import sysuser

    (tmp_path / "data").mkdir()
    f = tmp_path / "data" / "doc.json"
    f.write_text("{}")
    (tmp_path / "data" / "doc.json.lock").write_text("")
    if os.geteuid() != 0:
        assert sysuser.ensure_group("kukeon-test-nope") is None
        return
    import subprocess
    name = f"kuketest{os.getpid() % 10000}"
    try:
        gid = sysuser.ensure_group(name)
        assert gid is not None
        assert sysuser.ensure_group(name) == gid  # idempotent
        changed = sysuser.chown_tree(tmp_path, gid)
        assert changed >= 2
        assert f.stat().st_gid == gid
        # lock tombstones skipped
        assert (tmp_path / "data" / "doc.json.lock").stat().st_gid != gid \
            or gid == 0
        # dirs got setgid+group-rwx
        assert (tmp_path / "data").stat().st_mode & 0o2070 == 0o2070
        sysuser.apply_socket_group(str(f), gid)
        assert f.stat().st_mode & 0o777 == 0o660
    finally:
        subprocess.run(["groupdel", name], capture_output=True)
