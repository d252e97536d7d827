"""Per-space bridges and per-cell veth plumbing (the CNI-bridge +
host-local analog, reference internal/cni/config.go:32-81 +
container.go:34-99, over raw rtnetlink instead of exec'd plugins).

Topology: each networked space gets a Linux bridge `k-<8hex>` holding the
subnet's .1 as gateway; each cell gets a veth pair — host end attached to
the bridge, peer moved into the cell root's netns as eth0 with the cell's
/24 address. Egress policy is enforced primarily by ROUTING inside the
netns (default-deny = no default route; each allow CIDR gets a route via
the gateway), which constrains real packets even on hosts without
iptables; the iptables Enforcer layers on top where present.
"""
from __future__ import annotations

import hashlib
import json
import logging
import os
import signal
from typing import Dict, List, Optional

from kukeon_amd.runtime import namespaces as nsmod
from kukeon_amd.runtime.netlink import NetlinkError, Rtnl

log = logging.getLogger("kukeon.cellnet")


def bridge_name(realm: str, space: str) -> str:
    h = hashlib.sha256(f"{realm}/{space}".encode()).hexdigest()[:8]
    return f"k-{h}"


def veth_names(realm: str, space: str, stack: str, cell: str):
    h = hashlib.sha256(
        f"{realm}/{space}/{stack}/{cell}".encode()).hexdigest()[:6]
    return f"kv{h}", f"kvp{h}"


def _in_netns(pid: int, fn) -> None:
    """Run fn() in a forked child joined to pid's net namespace; raises
    on child failure (error text relayed through a pipe)."""
    r, w = os.pipe()
    child = os.fork()
    if child == 0:
        os.close(r)
        try:
            fd = os.open(f"/proc/{pid}/ns/net", os.O_RDONLY)
            nsmod.setns(fd, nsmod.CLONE_NEWNET)
            os.close(fd)
            fn()
            os._exit(0)
        except BaseException as e:  # noqa: BLE001
            try:
                os.write(w, str(e).encode()[:500])
            except OSError:
                pass
            os._exit(1)
    os.close(w)
    err = b""
    try:
        while True:
            chunk = os.read(r, 4096)
            if not chunk:
                break
            err += chunk
    finally:
        os.close(r)
    _, status = os.waitpid(child, 0)
    if os.waitstatus_to_exitcode(status) != 0:
        raise NetlinkError(1, f"netns setup failed: {err.decode()!r}")


class CellNetwork:
    """Daemon-side plumbing; every method is idempotent and safe to call
    on hosts without CAP_NET_ADMIN (raises NetlinkError — callers record
    degradation)."""

    def ensure_bridge(self, realm: str, space: str, subnet: str,
                      gateway: str) -> str:
        name = bridge_name(realm, space)
        with Rtnl() as nl:
            try:
                nl.new_bridge(name)
            except NetlinkError as e:
                if e.errno != 17:  # EEXIST
                    raise
            prefix = int(subnet.rsplit("/", 1)[1])
            nl.addr_add(name, gateway, prefix)
            nl.set_link(name, up=True)
        return name

    def teardown_bridge(self, realm: str, space: str) -> None:
        with Rtnl() as nl:
            try:
                nl.del_link(bridge_name(realm, space))
            except NetlinkError:
                pass

    def attach_cell(self, realm: str, space: str, stack: str, cell: str,
                    root_pid: int, ip: str, subnet: str, gateway: str,
                    default_deny: bool,
                    allow_cidrs: List[str]) -> Dict[str, str]:
        """veth pair: host side on the space bridge, peer as eth0 inside
        the root netns with routing per egress policy."""
        host, peer = veth_names(realm, space, stack, cell)
        br = bridge_name(realm, space)
        prefix = int(subnet.rsplit("/", 1)[1])
        net = subnet  # on-link subnet route inside the cell
        with Rtnl() as nl:
            try:
                nl.new_veth(host, peer)
            except NetlinkError as e:
                if e.errno == 17:
                    # stale pair from a previous run: recreate
                    try:
                        nl.del_link(host)
                    except NetlinkError:
                        pass
                    nl.new_veth(host, peer)
                else:
                    raise
            nl.set_link(host, master=br, up=True)
            nl.set_link(peer, ns_pid=root_pid)

        def configure():
            with Rtnl() as inner:
                inner.addr_add(peer, ip, prefix)
                inner.set_link(peer, up=True)
                inner.set_link_up_by_index(1)  # lo
                # on-link route for the space subnet
                inner.route_add(net, ifname=peer)
                if default_deny:
                    # deny-by-routing: ONLY allowed CIDRs are reachable
                    for cidr in allow_cidrs:
                        try:
                            inner.route_add(cidr, ifname=peer,
                                            gateway=gateway)
                        except NetlinkError as e:
                            log.warning("route %s: %s", cidr, e)
                else:
                    inner.route_add("0.0.0.0/0", ifname=peer,
                                    gateway=gateway)

        _in_netns(root_pid, configure)
        return {"ip": ip, "hostVeth": host, "bridge": br, "device": peer}

    def detach_cell(self, realm: str, space: str, stack: str,
                    cell: str) -> None:
        host, _ = veth_names(realm, space, stack, cell)
        with Rtnl() as nl:
            try:
                nl.del_link(host)
            except NetlinkError:
                pass
