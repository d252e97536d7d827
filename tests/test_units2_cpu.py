"""Unit tests for round-2 subsystems that need no root/GPU:
rtnetlink message framing, image-store content addressing, Kukefile
parsing, and the per-cell scope-lock manager.

The isolation/overlay e2e suites (test_isolation_e2e.py,
test_images_e2e.py) exercise the same code against the real kernel but
skip on hosts without CAP_NET_ADMIN/overlayfs — these tests pin the
deterministic logic everywhere.
"""
from __future__ import annotations

import struct
import threading

import pytest

from kukeon_amd.runtime import netlink
from kukeon_amd.controller.locks import ScopeLocks, cell_scope


# ---------------------------------------------------------------------------
# rtnetlink framing (fake socket: capture bytes, reply with an ACK)
# ---------------------------------------------------------------------------
class _FakeNlSock:
    def __init__(self):
        self.sent = []

    def send(self, data):
        self.sent.append(bytes(data))

    def recv(self, n):
        # NLMSG_ERROR with error==0 == ack for the last message
        _, _, _, seq, pid = struct.unpack_from("<IHHII", self.sent[-1], 0)
        body = struct.pack("<i", 0) + self.sent[-1][:16]
        return struct.pack("<IHHII", 16 + len(body), netlink.NLMSG_ERROR,
                           0, seq, pid) + body

    def close(self):
        pass


def _fake_rtnl():
    nl = netlink.Rtnl.__new__(netlink.Rtnl)
    nl.sock = _FakeNlSock()
    nl.seq = 0
    return nl


def _walk_attrs(data, off):
    """Parse a flat rtattr run -> {type: payload} (nested flag stripped)."""
    out = {}
    while off + 4 <= len(data):
        ln, typ = struct.unpack_from("<HH", data, off)
        if ln < 4:
            break
        out[typ & 0x3FFF] = data[off + 4:off + ln]
        off += (ln + 3) & ~3
    return out


def test_netlink_bridge_message_framing():
    nl = _fake_rtnl()
    nl.new_bridge("k-deadbeef")
    msg = nl.sock.sent[0]
    total, mtype, flags, seq, pid = struct.unpack_from("<IHHII", msg, 0)
    assert total == len(msg)
    assert mtype == netlink.RTM_NEWLINK
    assert flags & netlink.NLM_F_REQUEST
    assert flags & netlink.NLM_F_CREATE and flags & netlink.NLM_F_EXCL
    assert flags & netlink.NLM_F_ACK
    # ifinfomsg is 16 bytes after the 16-byte nlmsghdr
    attrs = _walk_attrs(msg, 32)
    assert attrs[netlink.IFLA_IFNAME] == b"k-deadbeef\0"
    info = _walk_attrs(attrs[netlink.IFLA_LINKINFO], 0)
    assert info[netlink.IFLA_INFO_KIND].rstrip(b"\0") == b"bridge"


def test_netlink_veth_nests_peer_name():
    nl = _fake_rtnl()
    nl.new_veth("k-ve0", "k-pe0")
    msg = nl.sock.sent[0]
    attrs = _walk_attrs(msg, 32)
    assert attrs[netlink.IFLA_IFNAME] == b"k-ve0\0"
    info = _walk_attrs(attrs[netlink.IFLA_LINKINFO], 0)
    assert info[netlink.IFLA_INFO_KIND].rstrip(b"\0") == b"veth"
    peer = _walk_attrs(info[netlink.IFLA_INFO_DATA], 0)
    peer_attrs = _walk_attrs(peer[netlink.VETH_INFO_PEER], 16)
    assert peer_attrs[netlink.IFLA_IFNAME] == b"k-pe0\0"


def test_netlink_route_message_dst_gw_scope():
    nl = _fake_rtnl()
    # on-link route (no gateway) -> RT_SCOPE_LINK
    nl.route_add("10.88.2.0/24", ifname=None, gateway=None)
    msg = nl.sock.sent[0]
    _, mtype, _, _, _ = struct.unpack_from("<IHHII", msg, 0)
    assert mtype == netlink.RTM_NEWROUTE
    fam, plen, _, _, table, proto, scope, rtype = struct.unpack_from(
        "<BBBBBBBB", msg, 16)
    assert (plen, table, scope) == (24, netlink.RT_TABLE_MAIN,
                                    netlink.RT_SCOPE_LINK)
    attrs = _walk_attrs(msg, 16 + 12)
    assert attrs[netlink.RTA_DST] == bytes([10, 88, 2, 0])
    # gateway route -> RT_SCOPE_UNIVERSE + RTA_GATEWAY
    nl.route_add("0.0.0.0/0", gateway="10.88.2.1")
    msg2 = nl.sock.sent[1]
    _, _, _, _, _, _, scope2, _ = struct.unpack_from("<BBBBBBBB", msg2, 16)
    assert scope2 == netlink.RT_SCOPE_UNIVERSE
    attrs2 = _walk_attrs(msg2, 16 + 12)
    assert attrs2[netlink.RTA_GATEWAY] == bytes([10, 88, 2, 1])


def test_netlink_error_ack_raises_with_errno():
    nl = _fake_rtnl()

    class _ErrSock(_FakeNlSock):
        def recv(self, n):
            _, _, _, seq, pid = struct.unpack_from("<IHHII",
                                                   self.sent[-1], 0)
            body = struct.pack("<i", -17) + self.sent[-1][:16]  # EEXIST
            return struct.pack("<IHHII", 16 + len(body),
                               netlink.NLMSG_ERROR, 0, seq, pid) + body

    nl.sock = _ErrSock()
    with pytest.raises(netlink.NetlinkError) as ei:
        nl.new_bridge("k-dup")
    assert ei.value.errno == 17
    # but addr_add/route_add swallow EEXIST (idempotent re-assert):
    nl2 = _fake_rtnl()
    nl2.sock = _ErrSock()
    nl2.route_add("10.0.0.0/24", gateway="10.0.0.1")  # no raise


# ---------------------------------------------------------------------------
# image store: content addressing without root
# ---------------------------------------------------------------------------
def test_image_store_content_addressed_layers(tmp_path):
    from kukeon_amd.images import ImageStore
    store = ImageStore(str(tmp_path))
    src = tmp_path / "src"
    src.mkdir()
    (src / "app.txt").write_text("hello")
    lid1 = store.add_layer_from_dir(src)
    lid2 = store.add_layer_from_dir(src)
    assert lid1 == lid2  # same content -> same id (dedupe)
    (src / "app.txt").write_text("changed")
    lid3 = store.add_layer_from_dir(src)
    assert lid3 != lid1
    assert (store.layer_root(lid1) / "app.txt").read_text() == "hello"
    assert (store.layer_root(lid3) / "app.txt").read_text() == "changed"

    store.put_manifest("app:v1", [lid1], config={"cmd": ["/app.txt"]})
    m = store.get("app:v1")
    assert m["layers"] == [lid1]
    assert m["config"]["cmd"] == ["/app.txt"]
    assert store.exists("app:v1") and not store.exists("app:v2")

    # prune drops the unreferenced layer only
    pruned = store.prune_layers()
    assert lid3 in pruned and lid1 not in pruned
    assert store.layer_root(lid1).exists()
    store.delete("app:v1")
    assert not store.exists("app:v1")
    assert lid1 in store.prune_layers()


def test_kukefile_parse_and_copy_escape(tmp_path):
    from kukeon_amd.images import Builder, BuildError, ImageStore
    steps = Builder._parse(
        "# comment\n"
        "FROM scratch\n"
        "COPY app /app\n"
        "ENV A=1 B=two\n"
        "RUN echo hi \\\n"
        "    there\n"
        "WORKDIR /srv\n"
        "CMD /app --serve\n")
    verbs = [s[0] for s in steps]
    assert verbs == ["FROM", "COPY", "ENV", "RUN", "WORKDIR", "CMD"]
    run_step = steps[3][1]
    assert "hi" in run_step and "there" in run_step  # continuation joined

    # COPY must not escape the build context
    store = ImageStore(str(tmp_path / "run"))
    ctx = tmp_path / "ctx"
    ctx.mkdir()
    b = Builder(store)
    with pytest.raises(BuildError):
        b.build(ctx, "FROM scratch\nCOPY ../secret /x\n", tag="bad")


# ---------------------------------------------------------------------------
# scope locks
# ---------------------------------------------------------------------------
def test_scope_locks_refcount_and_exclusion():
    locks = ScopeLocks()
    key = cell_scope("r", "s", "st", "cell-a")
    assert key == ("r", "s", "st", "cell-a")
    order = []
    entered = threading.Event()
    release = threading.Event()

    def holder():
        with locks.hold(key):
            order.append("a")
            entered.set()
            release.wait(5)

    t = threading.Thread(target=holder)
    t.start()
    assert entered.wait(5)
    assert locks.held_count() == 1

    def contender():
        with locks.hold(key):
            order.append("b")

    t2 = threading.Thread(target=contender)
    t2.start()
    t2.join(0.2)
    assert t2.is_alive() and order == ["a"]  # excluded while held
    release.set()
    t.join(5)
    t2.join(5)
    assert order == ["a", "b"]
    assert locks.held_count() == 0  # table does not grow with dead cells

    # re-entrant from the same thread
    with locks.hold(key):
        with locks.hold(key):
            assert locks.held_count() == 1
    assert locks.held_count() == 0


# ---------------------------------------------------------------------------
# scheme round-trips
# ---------------------------------------------------------------------------
def test_scheme_roundtrip_and_idempotency():
    from kukeon_amd.api import scheme

    beta = {
        "apiVersion": "v1beta1", "kind": "Cell",
        "metadata": {"name": "rt"},
        "spec": {"realmId": "r1", "spaceId": "s1", "stackId": "st1",
                 "autoDelete": True,
                 "containers": [{"id": "main", "image": "img",
                                 "command": "run", "args": ["-v"],
                                 "restartPolicy": "on-failure"}]},
    }
    # normalize is idempotent
    n1 = scheme.normalize(beta)
    assert scheme.normalize(n1) == n1
    # beta -> alpha -> beta preserves everything non-lossy
    wire, lost = scheme.to_wire(dict(n1), "v1alpha1")
    assert lost == []
    assert wire["spec"]["realm"] == "r1" and wire["spec"]["autoRemove"]
    assert wire["spec"]["containers"][0]["restartPolicy"] == "onFailure"
    back = scheme.normalize(wire)
    assert back["spec"]["realmId"] == "r1"
    assert back["spec"]["autoDelete"] is True
    assert back["spec"]["containers"][0]["restartPolicy"] == "on-failure"
    # alpha flat shorthand up-converts to a containers list with defaults
    alpha = {"apiVersion": "v1alpha1", "kind": "Cell",
             "metadata": {"name": "flat"},
             "spec": {"realm": "r1", "image": "busybox",
                      "command": "sh"}}
    n = scheme.normalize(alpha)
    assert n["spec"]["containers"][0]["image"] == "busybox"
    assert n["spec"]["containers"][0]["id"] == "main"
    assert n["spec"]["spaceId"] == "default"  # cross-version defaulting


def test_delete_image_refuses_while_referenced(tmp_path):
    from kukeon_amd.api import errors, v1beta1 as api
    from kukeon_amd.controller.core import Controller
    from kukeon_amd.runtime.process import FakeRuntime

    ctl = Controller(str(tmp_path / "run"), runtime=FakeRuntime())
    ctl.bootstrap()
    ctl.register_image("tool:v1", spec={})
    doc = api.CellDoc(
        metadata=api.Metadata(name="imguser"),
        spec=api.CellSpec(realm_id="default", space_id="default",
                          stack_id="default",
                          containers=[api.ContainerSpec(
                              id="main", image="tool:v1",
                              command="sleep", args=["5"])]))
    ctl.create_cell(doc)
    with pytest.raises(errors.Conflict):
        ctl.delete_image("tool:v1")
    ctl.delete_image("tool:v1", force=True)   # explicit override works
    ctl.register_image("tool:v2", spec={})
    ctl.delete_image("tool:v2")               # unreferenced: fine


def test_capture_log_rotation(tmp_path):
    """The shim's capture file rotates at its size cap instead of
    growing unbounded (one .1 generation kept for kuke log)."""
    import json
    import os
    import subprocess
    import sys

    d = tmp_path / "c"
    (d / "tty").mkdir(parents=True)
    spec = {"id": "noisy",
            "argv": ["/bin/sh", "-c",
                     "i=0; while [ $i -lt 64 ]; do "
                     "head -c 4096 /dev/zero | tr '\\0' x; i=$((i+1)); "
                     "done"],
            "env": [], "attachable": True, "capture": True}
    (d / "spawn.json").write_text(json.dumps(spec))
    code = (
        "import sys; sys.path.insert(0, %r);"
        "from kukeon_amd.tty import shim as m;"
        "m.Shim.CAP_LIMIT = 64 * 1024;"
        "s = m.Shim(__import__('pathlib').Path(%r));"
        "sys.exit(s.run_attachable())" % (os.getcwd(), str(d)))
    r = subprocess.run([sys.executable, "-c", code], timeout=60,
                       capture_output=True, text=True)
    rot = d / "capture.log.1"
    assert rot.exists(), (r.returncode, r.stdout[-300:], r.stderr[-300:])
    assert rot.stat().st_size >= 64 * 1024
    assert (d / "capture.log").stat().st_size < 1 << 20


def test_spec_hash_semantics():
    """Restart knobs and scope ids never bounce a healthy process;
    spawn-feeding fields and image layers do."""
    from kukeon_amd.api import v1beta1 as api
    from kukeon_amd.controller.spechash import spec_hash

    base = api.ContainerSpec(id="main", image="img", command="run",
                             args=["-v"], restart_policy="always")
    h = spec_hash(base)
    knobs = api.ContainerSpec(id="main", image="img", command="run",
                              args=["-v"], restart_policy="on-failure",
                              restart_max_retries=9)
    assert spec_hash(knobs) == h          # metadata-only knobs
    argv = api.ContainerSpec(id="main", image="img", command="run",
                             args=["-x"], restart_policy="always")
    assert spec_hash(argv) != h           # spawn-feeding field
    assert spec_hash(base, image_layers=["l1"]) != h
    assert spec_hash(base, image_layers=["l1"]) != \
        spec_hash(base, image_layers=["l2"])   # content drift


def test_cell_ipam_allocate_release(tmp_path):
    """Per-cell host-local IPAM inside the space /24 (ungated twin of
    the root-only isolation e2e): idempotent allocation, distinct IPs,
    release-reuse, .1 reserved for the gateway."""
    from kukeon_amd.controller.subnet import SubnetAllocator
    from kukeon_amd.state.store import Store

    alloc = SubnetAllocator(Store(str(tmp_path / "run")),
                            honor_host_routes=False)
    alloc.allocate("default", "web")
    gw = alloc.gateway("default", "web")
    assert gw and gw.endswith(".1")
    a = alloc.allocate_ip("default", "web", "cell-a")
    assert alloc.allocate_ip("default", "web", "cell-a") == a  # idempotent
    b = alloc.allocate_ip("default", "web", "cell-b")
    assert a != b and a != gw and b != gw
    assert a.rsplit(".", 1)[0] == gw.rsplit(".", 1)[0]  # same /24
    alloc.release_ip("default", "web", "cell-a")
    c = alloc.allocate_ip("default", "web", "cell-c")
    assert c == a  # released address is reusable


def test_log_formatter_and_naming():
    """ReformatFormatter emits `ts LEVEL "msg" k=v` lines (reference
    slog handler contract); naming validates and generates ids."""
    import logging

    from kukeon_amd.utils.logging import ReformatFormatter
    from kukeon_amd.controller import naming
    from kukeon_amd.api import errors as kerrors

    rec = logging.LogRecord("kukeon.cell", logging.INFO, __file__, 1,
                            'cell "started"', (), None)
    rec.kv = {"cell": "dev-a1b2c3", "pid": 412}
    line = ReformatFormatter().format(rec)
    assert "INFO  " in line and "cell 'started'" in line
    assert "cell=dev-a1b2c3 pid=412" in line
    assert line.endswith("logger=kukeon.cell")
    assert line[4] == "-" and line.split(" ")[0].endswith("Z")

    assert naming.validate_name("web-1") == "web-1"
    with pytest.raises(kerrors.ValidationError):
        naming.validate_name("Bad/Name")
    assert naming.root_container_id("s", "st", "c") == "s-st-c"
    n = naming.generate_cell_name("dev", taken=set())
    assert n.startswith("dev-") and len(n) == len("dev-") + 6


def test_scheme_session_has_no_alpha_representation():
    from kukeon_amd.api import scheme

    sess = {"apiVersion": "v1beta1", "kind": "Session",
            "metadata": {"name": "s1"},
            "spec": {"realmId": "r", "spaceId": "s", "stackId": "st"}}
    wire, lost = scheme.to_wire(sess, "v1alpha1")
    assert any("Session" in f for f in lost)


def test_instance_pinning_mismatch_fails_fast(tmp_path):
    """`.kukeon-instance.json` pins daemon identity: a restart with
    different knobs refuses to start (reference internal/instance)."""
    from kukeon_amd.api import errors as kerrors
    from kukeon_amd.daemon.server import verify_or_write_instance

    verify_or_write_instance(tmp_path, cgroup_root="kukeon")
    verify_or_write_instance(tmp_path, cgroup_root="kukeon")  # same: ok
    with pytest.raises(kerrors.InvalidArgument):
        verify_or_write_instance(tmp_path, cgroup_root="other")
