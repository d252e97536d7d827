"""Per-space /24 subnet allocator out of 10.88.0.0/16.

Persisted at <run>/data/<realm>/<space>/network.json exactly like the
reference (internal/cni/subnet.go); a single in-process lock arbitrates,
and the store's flock covers cross-process daemon restarts.
"""
from __future__ import annotations

import json
import threading
from typing import Dict, Optional

from kukeon_amd.api import errors
from kukeon_amd.state.store import Store

POOL_PREFIX = "10.88"


class SubnetAllocator:
    def __init__(self, store: Store, honor_host_routes: bool = True,
                 route_file: str = "/proc/net/route"):
        self.store = store
        self.honor_host_routes = honor_host_routes
        self.route_file = route_file
        self._mu = threading.Lock()

    def _network_path(self, realm: str, space: str):
        return self.store.space_dir(realm, space) / "network.json"

    def _used(self) -> Dict[int, str]:
        used = {}
        root = self.store.data_root
        if root.is_dir():
            for net in root.glob("*/*/network.json"):
                try:
                    data = json.loads(net.read_text())
                    sub = data.get("subnet", "")
                    octet = int(sub.split(".")[2])
                    used[octet] = str(net)
                except (ValueError, IndexError, OSError):
                    continue
        # also honor LIVE host routes in the pool: another instance (or a
        # crashed daemon whose store was wiped) may still hold a bridge —
        # colliding subnets blackhole reply traffic
        if not self.honor_host_routes:
            return used
        try:
            with open(self.route_file) as f:
                for line in f.read().splitlines()[1:]:
                    parts = line.split()
                    if len(parts) < 8:
                        continue
                    import socket as _s
                    import struct as _st
                    dst = _s.inet_ntoa(_st.pack("<I", int(parts[1], 16)))
                    if dst.startswith(POOL_PREFIX + "."):
                        used.setdefault(int(dst.split(".")[2]),
                                        f"host-route:{parts[0]}")
        except (OSError, ValueError):
            pass
        return used

    def allocate(self, realm: str, space: str) -> str:
        with self._mu:
            path = self._network_path(realm, space)
            cur = self.store.read(path)
            if cur and cur.get("subnet"):
                return cur["subnet"]
            used = self._used()
            for octet in range(0, 256):
                if octet not in used:
                    subnet = f"{POOL_PREFIX}.{octet}.0/24"
                    self.store.write(path, {
                        "subnet": subnet,
                        "gateway": f"{POOL_PREFIX}.{octet}.1",
                        "realm": realm, "space": space,
                    })
                    return subnet
            raise errors.KukeonError("subnet pool 10.88.0.0/16 exhausted")

    def allocate_ip(self, realm: str, space: str, cell: str) -> str:
        """host-local IPAM: per-cell address from the space's /24
        (offsets 2..254; .1 is the bridge gateway), persisted beside the
        subnet so daemon restarts keep assignments."""
        with self._mu:
            path = self._network_path(realm, space)
            cur = self.store.read(path)
            if not cur or not cur.get("subnet"):
                raise errors.KukeonError(
                    f"space {realm}/{space} has no subnet")
            ips = cur.setdefault("ips", {})
            base = cur["subnet"].rsplit(".", 1)[0].rsplit("/", 1)[0]
            if cell in ips:
                return f"{base}.{ips[cell]}"
            used = set(ips.values())
            for off in range(2, 255):
                if off not in used:
                    ips[cell] = off
                    self.store.write(path, cur)
                    return f"{base}.{off}"
            raise errors.KukeonError(f"subnet {cur['subnet']} exhausted")

    def release_ip(self, realm: str, space: str, cell: str) -> None:
        with self._mu:
            path = self._network_path(realm, space)
            cur = self.store.read(path)
            if cur and cur.get("ips", {}).pop(cell, None) is not None:
                self.store.write(path, cur)

    def gateway(self, realm: str, space: str) -> Optional[str]:
        cur = self.store.read(self._network_path(realm, space))
        return cur.get("gateway") if cur else None

    def lookup(self, realm: str, space: str) -> Optional[str]:
        cur = self.store.read(self._network_path(realm, space))
        return cur.get("subnet") if cur else None

    def release(self, realm: str, space: str) -> None:
        with self._mu:
            self.store.delete(self._network_path(realm, space))
