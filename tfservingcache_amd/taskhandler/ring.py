"""Consistent-hash ring.

Same contract the reference gets from stathat.com/c/consistent
(pkg/taskhandler/cluster.go:44-130 and the behavior pinned by
cluster_test.go:145-227): deterministic key->node mapping, N distinct
replicas per key, minimal movement on membership change, and exact
reversion when membership reverts. Implementation: hash ring with 20
virtual nodes per member (crc32, like the reference's ring library) and
bisect lookup.

MI355X extension: members are (node, slot) pairs — each MI355X node
registers one member per GPU so keys map to a specific GPU's pool
(SURVEY.md §2.4 mapping table).
"""
from __future__ import annotations

import bisect
import threading
import zlib
from typing import List, Optional, Sequence

VNODES = 20


def _hash(key: str) -> int:
    return zlib.crc32(key.encode())


class ConsistentHashRing:
    def __init__(self, vnodes: int = VNODES):
        self.vnodes = vnodes
        self._lock = threading.RLock()
        self._hashes: List[int] = []
        self._members_at: dict = {}
        self._members: List[str] = []

    def set_members(self, members: Sequence[str]) -> None:
        """Full replacement (the reference re-seeds the whole ring on
        every membership update — cluster.go:104-113)."""
        with self._lock:
            hashes: List[int] = []
            at = {}
            for mem in set(members):
                for i in range(self.vnodes):
                    # '#' separator so members differing only in trailing
                    # digits can't produce colliding vnode keys
                    # (e.g. 'gpu1'+'12' vs 'gpu11'+'2')
                    h = _hash(f"{mem}#{i}")
                    at[h] = mem
                    hashes.append(h)
            hashes.sort()
            self._hashes = hashes
            self._members_at = at
            self._members = sorted(set(members))

    def members(self) -> List[str]:
        with self._lock:
            return list(self._members)

    def get(self, key: str) -> Optional[str]:
        nodes = self.get_n(key, 1)
        return nodes[0] if nodes else None

    def get_n(self, key: str, n: int) -> List[str]:
        """First n DISTINCT members clockwise of the key's hash
        (consistent.GetN semantics; n clamped to member count)."""
        with self._lock:
            if not self._hashes:
                return []
            n = max(1, min(n, len(self._members)))
            start = bisect.bisect(self._hashes, _hash(key))
            out: List[str] = []
            for i in range(len(self._hashes)):
                h = self._hashes[(start + i) % len(self._hashes)]
                mem = self._members_at[h]
                if mem not in out:
                    out.append(mem)
                    if len(out) == n:
                        break
            return out
