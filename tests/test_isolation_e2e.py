"""Isolation e2e: real namespaces and real packets.

These run against the real ProcessRuntime + shim with actual unshare /
setns / rtnetlink plumbing: a cell in a default-deny space must FAIL a
TCP connect outside its allowlist while an allowlisted CIDR succeeds —
asserted by observing the packets land (or not) on a live listener.
Hosts without CAP_SYS_ADMIN/CAP_NET_ADMIN skip (the controller records
the same degradation at runtime).
"""
import contextlib
import socket
import threading
import time
import uuid
from pathlib import Path

import pytest

from kukeon_amd.api import v1beta1 as api
from kukeon_amd.controller.core import Controller
from kukeon_amd.runtime import namespaces as nsmod
from kukeon_amd.runtime import netlink

CAPS = (nsmod.can_unshare(nsmod.CLONE_NEWUTS | nsmod.CLONE_NEWIPC) and
        nsmod.can_unshare(nsmod.CLONE_NEWNET) and netlink.available())

pytestmark = pytest.mark.skipif(
    not CAPS, reason="host denies namespaces/netlink (degraded mode)")


def wait_for(pred, timeout=15.0, interval=0.05):
    t0 = time.time()
    while time.time() - t0 < timeout:
        if pred():
            return True
        time.sleep(interval)
    return False


@pytest.fixture
def ctl(tmp_path):
    c = Controller(str(tmp_path / "run"), gpu_devices=[])
    c.bootstrap()
    yield c
    for realm in c.store.list_children(c.store.data_root):
        for space in c.store.list_children(c.store.realm_dir(realm)):
            for stack in c.store.list_children(
                    c.store.space_dir(realm, space)):
                for cell in c.store.list_children(
                        c.store.stack_dir(realm, space, stack)):
                    try:
                        c.delete_cell(realm, space, stack, cell, force=True)
                    except Exception:
                        pass
            try:
                c.delete_space(realm, space, cascade=True)
            except Exception:
                pass


def make_cell(name, space, cmd):
    return api.CellDoc(
        metadata=api.Metadata(name=name),
        spec=api.CellSpec(
            realm_id="default", space_id=space, stack_id="default",
            containers=[api.ContainerSpec(id="main", image="none",
                                          command="sh",
                                          args=["-c", cmd])]))


def test_uts_ipc_namespace_isolation(ctl):
    """Container sees the cell name as hostname; the host keeps its own."""
    host_hn = socket.gethostname()
    name = f"nsc-{uuid.uuid4().hex[:6]}"
    ctl.ensure_stack("default", "default", "default")
    ctl.create_cell(make_cell(name, "default",
                              "hostname > hn.txt; sleep 30"))
    ctl.start_cell("default", "default", "default", name)
    cdir = ctl.store.cell_dir("default", "default", "default", name) / "main"
    assert wait_for(lambda: (cdir / "hn.txt").exists() and
                    (cdir / "hn.txt").read_text().strip())
    assert (cdir / "hn.txt").read_text().strip() == name
    assert socket.gethostname() == host_hn  # host unaffected
    # ns.json records what was actually held
    ns = (ctl.store.read(cdir / "ns.json") or {})
    assert "uts" in ns.get("held", [])
    ctl.delete_cell("default", "default", "default", name, force=True)


def test_etc_files_rendered_in_mount_ns(ctl):
    """Private mount ns: /etc/hostname + /etc/hosts show the cell
    identity inside, while the host's files are untouched."""
    host_etc = Path("/etc/hostname").read_text() \
        if Path("/etc/hostname").exists() else None
    name = f"etc-{uuid.uuid4().hex[:6]}"
    ctl.create_cell(make_cell(
        name, "default",
        "cat /etc/hostname > hn.txt; cat /etc/hosts > hosts.txt; sleep 30"))
    ctl.start_cell("default", "default", "default", name)
    cdir = ctl.store.cell_dir("default", "default", "default", name) / "main"
    assert wait_for(lambda: (cdir / "hosts.txt").exists())
    time.sleep(0.1)
    assert (cdir / "hn.txt").read_text().strip() == name
    assert name in (cdir / "hosts.txt").read_text()
    if host_etc is not None:
        assert Path("/etc/hostname").read_text() == host_etc
    ctl.delete_cell("default", "default", "default", name, force=True)


CONNECT_SCRIPT = r"""
import socket, sys
def probe(ip, port):
    s = socket.socket()
    s.settimeout(3.0)
    try:
        s.connect((ip, port))
        s.sendall(b"hi-from-" + b"%TAG%")
        return "connected"
    except socket.timeout:
        return "timeout"
    except OSError as e:
        return f"oserr:{e.errno}"
    finally:
        s.close()
with open("net.txt", "w") as f:
    f.write("gw=" + probe("%GW%", %PORT%) + "\n")
    f.write("allowed=" + probe("%ALLOWED%", %PORT%) + "\n")
    f.write("denied=" + probe("192.0.2.9", %PORT%) + "\n")
"""


def test_egress_default_deny_blocks_real_packets(ctl):
    """The VERDICT done-criterion: a default-deny space's cell FAILS a
    TCP connect outside the allowlist; an allowlisted CIDR succeeds.
    The allowed target is a live listener on a secondary bridge address
    inside the allowed CIDR, so success is observed packet-for-packet."""
    space = f"isol-{uuid.uuid4().hex[:6]}"
    sp = api.SpaceDoc(
        metadata=api.Metadata(name=space),
        spec=api.SpaceSpec(
            realm_id="default",
            network=api.SpaceNetwork(egress=api.EgressPolicy(
                default="deny",
                allow=[api.EgressAllowRule(cidr="198.51.100.0/24")]))))
    ctl.create_space(sp)
    ctl.ensure_stack("default", space, "default")
    subnet = ctl.subnets.lookup("default", space)
    gw = ctl.subnets.gateway("default", space)
    assert subnet and gw
    from kukeon_amd.runtime.cellnet import bridge_name
    br = bridge_name("default", space)
    # secondary address on the bridge inside the ALLOWED CIDR
    with netlink.Rtnl() as nl:
        nl.addr_add(br, "198.51.100.1", 24)

    port = 0
    hits = []
    ls = socket.socket()
    ls.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
    ls.bind(("0.0.0.0", 0))
    port = ls.getsockname()[1]
    ls.listen(8)
    ls.settimeout(30)

    def serve():
        try:
            while True:
                conn, _ = ls.accept()
                data = conn.recv(256)
                hits.append(data.decode("utf-8", "replace"))
                conn.close()
        except OSError:
            pass

    t = threading.Thread(target=serve, daemon=True)
    t.start()

    name = f"net-{uuid.uuid4().hex[:6]}"
    script = (CONNECT_SCRIPT.replace("%GW%", gw)
              .replace("%ALLOWED%", "198.51.100.1")
              .replace("%PORT%", str(port))
              .replace("%TAG%", name))
    ctl.create_cell(make_cell(name, space, "python3 probe.py; sleep 30"))
    cdir_pre = ctl.store.cell_dir("default", space, "default", name) / "main"
    cdir_pre.mkdir(parents=True, exist_ok=True)
    (cdir_pre / "probe.py").write_text(script)
    ctl.start_cell("default", space, "default", name)
    cell_dir = ctl.store.cell_dir("default", space, "default", name)
    cdir = cell_dir / "main"
    # network plumbed for real (not degraded)
    netrec = ctl.store.read(cell_dir / "network.json") or {}
    assert netrec.get("mode") == "netns", netrec
    assert netrec.get("ip", "").startswith("10.88.")
    assert wait_for(lambda: (cdir / "net.txt").exists(), timeout=25)
    time.sleep(0.2)
    res = dict(line.split("=", 1) for line in
               (cdir / "net.txt").read_text().strip().splitlines())
    # in-subnet gateway: reachable (on-link route)
    assert res["gw"] == "connected", res
    # allowlisted CIDR: routed via the gateway and actually served
    assert res["allowed"] == "connected", res
    # outside the allowlist: no route -> the connect FAILS (ENETUNREACH)
    assert res["denied"].startswith("oserr:101"), res
    assert any(f"hi-from-{name}" in h for h in hits)
    ls.close()
    ctl.delete_cell("default", space, "default", name, force=True)


def test_default_allow_space_routes_everything(ctl):
    """No egress policy -> default route via the gateway exists inside
    the netns (reachability beyond the host is then the host's routing
    problem, not the cell's)."""
    space = f"open-{uuid.uuid4().hex[:6]}"
    sp = api.SpaceDoc(metadata=api.Metadata(name=space),
                      spec=api.SpaceSpec(realm_id="default",
                                         network=api.SpaceNetwork()))
    ctl.create_space(sp)
    ctl.ensure_stack("default", space, "default")
    name = f"op-{uuid.uuid4().hex[:6]}"
    ctl.create_cell(make_cell(
        name, space, "cat /proc/net/route > rt.txt; sleep 30"))
    ctl.start_cell("default", space, "default", name)
    cell_dir = ctl.store.cell_dir("default", space, "default", name)
    netrec = ctl.store.read(cell_dir / "network.json") or {}
    assert netrec.get("mode") == "netns", netrec  # not a host-ns false pass
    cdir = cell_dir / "main"
    assert wait_for(lambda: (cdir / "rt.txt").exists())
    time.sleep(0.1)
    routes = (cdir / "rt.txt").read_text()
    # a default route (destination 00000000) exists in the cell's netns
    assert any(line.split()[1] == "00000000"
               for line in routes.splitlines()[1:] if line.strip()), routes
    ctl.delete_cell("default", space, "default", name, force=True)


@pytest.mark.skipif(not nsmod.can_unshare(nsmod.CLONE_NEWNS),
                    reason="no mount ns")
def test_volumes_are_real_bind_mounts(ctl, tmp_path):
    """A declared volume appears AT ITS TARGET inside the container's
    mount namespace as a real mount; the host tree is untouched."""
    vol = tmp_path / "shared"
    vol.mkdir()
    (vol / "payload.txt").write_text("mounted-for-real\n")
    name = f"vol-{uuid.uuid4().hex[:6]}"
    doc = api.CellDoc(
        metadata=api.Metadata(name=name),
        spec=api.CellSpec(
            realm_id="default", space_id="default", stack_id="default",
            containers=[api.ContainerSpec(
                id="main", image="none", command="sh",
                args=["-c", "cat /mnt/kuke-vol/payload.txt > got.txt "
                            "2>&1; echo inner > /mnt/kuke-vol/back.txt; "
                            "sleep 30"],
                volumes=[api.VolumeMount(name="shared",
                                         source=str(vol),
                                         target="/mnt/kuke-vol")])]))
    ctl.create_cell(doc)
    ctl.start_cell("default", "default", "default", name)
    cdir = ctl.store.cell_dir("default", "default", "default", name) / "main"
    assert wait_for(lambda: (cdir / "got.txt").exists())
    time.sleep(0.1)
    assert "mounted-for-real" in (cdir / "got.txt").read_text()
    # writes through the mount land in the volume source
    assert wait_for(lambda: (vol / "back.txt").exists())
    # the MOUNT is private: the host sees only the empty mountpoint
    # directory (created on the shared fs), never the volume's content
    host_view = Path("/mnt/kuke-vol")
    assert not (host_view / "payload.txt").exists()
    ctl.delete_cell("default", "default", "default", name, force=True)
    with contextlib.suppress(OSError):
        host_view.rmdir()
