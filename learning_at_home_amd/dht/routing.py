"""Kademlia routing: 160-bit node ids and k-bucket routing table
(SURVEY.md §2 "DHT node" — ids, k-buckets, XOR metric).
"""
from __future__ import annotations

import hashlib
import os
import time
from typing import Any, Dict, List, Optional, Tuple

ID_BITS = 160


class DHTID(int):
    """160-bit identifier in XOR-metric key space."""

    MIN, MAX = 0, 2**ID_BITS - 1

    @classmethod
    def generate(cls, source: Optional[Any] = None) -> "DHTID":
        if source is None:
            source = os.urandom(20)
        if isinstance(source, str):
            source = source.encode()
        if not isinstance(source, bytes):
            source = repr(source).encode()
        return cls(int.from_bytes(hashlib.sha1(source).digest(), "big"))

    def xor_distance(self, other: int) -> int:
        return int(self) ^ int(other)

    def to_bytes(self) -> bytes:  # type: ignore[override]
        return int(self).to_bytes(20, "big")

    @classmethod
    def from_bytes(cls, raw: bytes) -> "DHTID":  # type: ignore[override]
        return cls(int.from_bytes(raw, "big"))

    def __repr__(self) -> str:
        return f"DHTID({hex(int(self))[:12]}…)"


class KBucket:
    """One k-bucket: up to `size` (node_id → endpoint) pairs, LRU by last-seen.
    Nodes are evicted only when stale (Kademlia prefers old, live contacts)."""

    def __init__(self, lower: int, upper: int, size: int):
        self.lower, self.upper, self.size = lower, upper, size
        self.nodes: Dict[int, Any] = {}  # node_id -> endpoint
        self.last_seen: Dict[int, float] = {}
        self.replacement: Dict[int, Any] = {}
        self.last_updated = time.monotonic()

    def covers(self, node_id: int) -> bool:
        return self.lower <= node_id < self.upper

    def add_or_update(self, node_id: int, endpoint: Any) -> bool:
        self.last_updated = time.monotonic()
        if node_id in self.nodes:
            self.nodes[node_id] = endpoint
            self.last_seen[node_id] = time.monotonic()
            return True
        if len(self.nodes) < self.size:
            self.nodes[node_id] = endpoint
            self.last_seen[node_id] = time.monotonic()
            return True
        self.replacement[node_id] = endpoint  # bucket full: remember as replacement
        return False

    def remove(self, node_id: int) -> None:
        self.nodes.pop(node_id, None)
        self.last_seen.pop(node_id, None)
        if self.replacement:
            rid, rep = self.replacement.popitem()
            self.nodes[rid] = rep
            self.last_seen[rid] = time.monotonic()

    def split(self) -> Tuple["KBucket", "KBucket"]:
        mid = (self.lower + self.upper) // 2
        left = KBucket(self.lower, mid, self.size)
        right = KBucket(mid, self.upper, self.size)
        for nid, ep in self.nodes.items():
            dst = left if left.covers(nid) else right
            dst.nodes[nid] = ep
            dst.last_seen[nid] = self.last_seen.get(nid, time.monotonic())
        for nid, ep in self.replacement.items():
            (left if left.covers(nid) else right).replacement[nid] = ep
        return left, right


class RoutingTable:
    """Binary-trie-flattened list of k-buckets; only the bucket containing our
    own id splits (standard Kademlia)."""

    def __init__(self, node_id: DHTID, bucket_size: int = 20):
        self.node_id = node_id
        self.bucket_size = bucket_size
        self.buckets: List[KBucket] = [KBucket(0, 2**ID_BITS, bucket_size)]

    def _bucket_index(self, node_id: int) -> int:
        lo, hi = 0, len(self.buckets)
        while lo < hi:
            mid = (lo + hi) // 2
            if self.buckets[mid].upper <= node_id:
                lo = mid + 1
            else:
                hi = mid
        return lo

    def add_or_update_node(self, node_id: int, endpoint: Any) -> None:
        if node_id == self.node_id:
            return
        idx = self._bucket_index(node_id)
        bucket = self.buckets[idx]
        if bucket.add_or_update(node_id, endpoint):
            return
        if bucket.covers(int(self.node_id)):
            left, right = bucket.split()
            self.buckets[idx : idx + 1] = [left, right]
            self.add_or_update_node(node_id, endpoint)

    def remove_node(self, node_id: int) -> None:
        self.buckets[self._bucket_index(node_id)].remove(node_id)

    def get_endpoint(self, node_id: int) -> Optional[Any]:
        return self.buckets[self._bucket_index(node_id)].nodes.get(node_id)

    def get_nearest_neighbors(self, query_id: int, k: int, exclude: Tuple[int, ...] = ()) -> List[Tuple[int, Any]]:
        candidates: List[Tuple[int, int, Any]] = []
        excluded = set(exclude)
        for bucket in self.buckets:
            for nid, ep in bucket.nodes.items():
                if nid not in excluded:
                    candidates.append((query_id ^ nid, nid, ep))
        candidates.sort(key=lambda t: t[0])
        return [(nid, ep) for _, nid, ep in candidates[:k]]

    def __len__(self) -> int:
        return sum(len(b.nodes) for b in self.buckets)
