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
    def __init__(self, store: Store):
        self.store = store
        self._mu = threading.Lock()

    def _network_path(self, realm: str, space: str):
        return self.store.space_dir(realm, space) / "network.json"

    def _used(self) -> Dict[int, str]:
        used = {}
        root = self.store.data_root
        if not root.is_dir():
            return used
        for net in root.glob("*/*/network.json"):
            try:
                data = json.loads(net.read_text())
                sub = data.get("subnet", "")
                octet = int(sub.split(".")[2])
                used[octet] = str(net)
            except (ValueError, IndexError, OSError):
                continue
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

    def lookup(self, realm: str, space: str) -> Optional[str]:
        cur = self.store.read(self._network_path(realm, space))
        return cur.get("subnet") if cur else None

    def release(self, realm: str, space: str) -> None:
        with self._mu:
            self.store.delete(self._network_path(realm, space))
