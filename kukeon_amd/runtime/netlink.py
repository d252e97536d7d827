"""Minimal rtnetlink client (pure stdlib): bridges, veth pairs, addresses,
routes, link state, namespace moves.

The reference shells out to CNI plugins + `ip link` (internal/cni); this
environment ships neither iproute2 nor CNI binaries, so the per-space
bridge + per-cell veth plumbing talks NETLINK_ROUTE directly. Scope is
exactly what the runner needs — not a general iproute2.

All operations raise NetlinkError (with errno) on kernel refusal; callers
degrade gracefully (spaces record network=degraded instead of failing the
cell).
"""
from __future__ import annotations

import os
import socket
import struct
from typing import List, Optional, Tuple

# netlink message types
RTM_NEWLINK = 16
RTM_DELLINK = 17
RTM_GETLINK = 18
RTM_NEWADDR = 20
RTM_NEWROUTE = 24
NLM_F_REQUEST = 0x1
NLM_F_ACK = 0x4
NLM_F_EXCL = 0x200
NLM_F_CREATE = 0x400
NLMSG_ERROR = 0x2
NLMSG_DONE = 0x3

# ifinfomsg / ifaddrmsg / rtmsg attribute ids
IFLA_IFNAME = 3
IFLA_MTU = 4
IFLA_MASTER = 10
IFLA_LINKINFO = 18
IFLA_NET_NS_PID = 19
IFLA_NET_NS_FD = 28
IFLA_INFO_KIND = 1
IFLA_INFO_DATA = 2
VETH_INFO_PEER = 1
IFA_ADDRESS = 1
IFA_LOCAL = 2
RTA_DST = 1
RTA_GATEWAY = 5
RTA_OIF = 4

IFF_UP = 0x1

RT_TABLE_MAIN = 254
RTPROT_STATIC = 4
RT_SCOPE_UNIVERSE = 0
RT_SCOPE_LINK = 253
RTN_UNICAST = 1

AF_UNSPEC = 0


class NetlinkError(OSError):
    pass


def _attr(kind: int, payload: bytes) -> bytes:
    ln = 4 + len(payload)
    pad = (4 - ln % 4) % 4
    return struct.pack("<HH", ln, kind) + payload + b"\0" * pad


def _attr_str(kind: int, s: str) -> bytes:
    return _attr(kind, s.encode() + b"\0")


def _attr_u32(kind: int, v: int) -> bytes:
    return _attr(kind, struct.pack("<I", v))


def _nested(kind: int, payload: bytes) -> bytes:
    return _attr(kind | 0x8000, payload)  # NLA_F_NESTED


class Rtnl:
    def __init__(self) -> None:
        self.sock = socket.socket(socket.AF_NETLINK, socket.SOCK_RAW,
                                  socket.NETLINK_ROUTE)
        self.sock.bind((0, 0))
        self.seq = 0

    def close(self) -> None:
        self.sock.close()

    def __enter__(self) -> "Rtnl":
        return self

    def __exit__(self, *a) -> None:
        self.close()

    def _roundtrip(self, mtype: int, flags: int, body: bytes) -> None:
        self.seq += 1
        hdr = struct.pack("<IHHII", 16 + len(body), mtype,
                          flags | NLM_F_REQUEST | NLM_F_ACK, self.seq,
                          os.getpid())
        self.sock.send(hdr + body)
        data = self.sock.recv(65536)
        # parse first message; expect NLMSG_ERROR with error==0 (the ack)
        ln, t, fl, seq, pid = struct.unpack_from("<IHHII", data, 0)
        if t == NLMSG_ERROR:
            err = struct.unpack_from("<i", data, 16)[0]
            if err != 0:
                raise NetlinkError(-err, os.strerror(-err))
            return
        raise NetlinkError(71, f"unexpected netlink reply type {t}")

    # -- links ---------------------------------------------------------
    def _ifinfo(self, index: int = 0, flags: int = 0,
                change: int = 0) -> bytes:
        return struct.pack("<BxHiII", AF_UNSPEC, 0, index, flags, change)

    def new_bridge(self, name: str) -> None:
        body = self._ifinfo()
        body += _attr_str(IFLA_IFNAME, name)
        body += _nested(IFLA_LINKINFO, _attr_str(IFLA_INFO_KIND, "bridge"))
        self._roundtrip(RTM_NEWLINK, NLM_F_CREATE | NLM_F_EXCL, body)

    def new_veth(self, name: str, peer: str) -> None:
        peer_body = self._ifinfo() + _attr_str(IFLA_IFNAME, peer)
        info = _attr_str(IFLA_INFO_KIND, "veth")
        info += _nested(IFLA_INFO_DATA, _nested(VETH_INFO_PEER, peer_body))
        body = self._ifinfo()
        body += _attr_str(IFLA_IFNAME, name)
        body += _nested(IFLA_LINKINFO, info)
        self._roundtrip(RTM_NEWLINK, NLM_F_CREATE | NLM_F_EXCL, body)

    def del_link(self, name: str) -> None:
        body = self._ifinfo() + _attr_str(IFLA_IFNAME, name)
        self._roundtrip(RTM_DELLINK, 0, body)

    def link_index(self, name: str) -> int:
        return socket.if_nametoindex(name)

    def set_link(self, name: str, up: Optional[bool] = None,
                 master: Optional[str] = None,
                 ns_pid: Optional[int] = None,
                 ns_fd: Optional[int] = None) -> None:
        idx = self.link_index(name)
        flags = IFF_UP if up else 0
        change = IFF_UP if up is not None else 0
        body = self._ifinfo(idx, flags, change)
        if master is not None:
            body += _attr_u32(IFLA_MASTER,
                              self.link_index(master) if master else 0)
        if ns_pid is not None:
            body += _attr_u32(IFLA_NET_NS_PID, ns_pid)
        if ns_fd is not None:
            body += _attr_u32(IFLA_NET_NS_FD, ns_fd)
        self._roundtrip(RTM_NEWLINK, 0, body)

    def set_link_up_by_index(self, idx: int) -> None:
        body = self._ifinfo(idx, IFF_UP, IFF_UP)
        self._roundtrip(RTM_NEWLINK, 0, body)

    # -- addresses -----------------------------------------------------
    def addr_add(self, ifname: str, ip: str, prefix: int) -> None:
        idx = self.link_index(ifname)
        body = struct.pack("<BBBBi", socket.AF_INET, prefix, 0,
                           RT_SCOPE_UNIVERSE, idx)
        raw = socket.inet_aton(ip)
        body += _attr(IFA_LOCAL, raw) + _attr(IFA_ADDRESS, raw)
        try:
            self._roundtrip(RTM_NEWADDR, NLM_F_CREATE | NLM_F_EXCL, body)
        except NetlinkError as e:
            if e.errno != 17:  # EEXIST is fine (idempotent re-assert)
                raise

    # -- routes --------------------------------------------------------
    def route_add(self, dst_cidr: str, ifname: Optional[str] = None,
                  gateway: Optional[str] = None) -> None:
        dst, _, plen = dst_cidr.partition("/")
        prefix = int(plen) if plen else 32
        scope = RT_SCOPE_LINK if gateway is None else RT_SCOPE_UNIVERSE
        body = struct.pack("<BBBBBBBBI", socket.AF_INET, prefix, 0, 0,
                           RT_TABLE_MAIN, RTPROT_STATIC, scope,
                           RTN_UNICAST, 0)
        body += _attr(RTA_DST, socket.inet_aton(dst))
        if gateway:
            body += _attr(RTA_GATEWAY, socket.inet_aton(gateway))
        if ifname:
            body += _attr_u32(RTA_OIF, self.link_index(ifname))
        try:
            self._roundtrip(RTM_NEWROUTE, NLM_F_CREATE | NLM_F_EXCL, body)
        except NetlinkError as e:
            if e.errno != 17:
                raise


def available() -> bool:
    """Can this process drive rtnetlink (needs CAP_NET_ADMIN)?"""
    try:
        with Rtnl():
            return os.geteuid() == 0
    except OSError:
        return False
