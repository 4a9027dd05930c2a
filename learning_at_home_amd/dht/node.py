"""DHTNode — one Kademlia peer: iterative lookups with α parallelism,
value expiration, response caching (SURVEY.md §2 "DHT node", §3.3 call stack).
"""
from __future__ import annotations

import asyncio
import heapq
from typing import Any, Dict, List, Optional, Sequence, Tuple

from ..utils import get_dht_time, get_logger
from .protocol import DHTProtocol
from .routing import DHTID

logger = get_logger(__name__)


class DHTNode:
    """Create with `await DHTNode.create(...)`. All methods are coroutines and
    must run on the loop that created the node (the DHT process façade in
    dht/__init__.py handles that for synchronous callers)."""

    def __init__(self) -> None:
        self.node_id: DHTID = None  # set by create()
        self.protocol: DHTProtocol = None
        self.port: int = None
        self.alpha = 3
        self.k = 20
        self.replication = 8  # peers a value is stored on (<= k)
        # neighbor cache for periodic re-announces: key_id -> (expiry, peers).
        # A server re-declaring hundreds of expert uids every announce period
        # would otherwise pay a full iterative traverse per key per period.
        self._store_neighbors: dict = {}
        self._store_neighbors_ttl = 60.0

    @classmethod
    async def create(
        cls,
        listen_on: Tuple[str, int] = ("127.0.0.1", 0),
        initial_peers: Sequence[Tuple[str, int]] = (),
        bucket_size: int = 20,
        alpha: int = 3,
        rpc_timeout: float = 3.0,
        cache_size: int = 10000,
    ) -> "DHTNode":
        self = cls()
        self.node_id = DHTID.generate()
        self.alpha, self.k = alpha, bucket_size
        loop = asyncio.get_event_loop()
        self.protocol = DHTProtocol(self.node_id, bucket_size, cache_size, rpc_timeout)
        transport, _ = await loop.create_datagram_endpoint(lambda: self.protocol, local_addr=listen_on)
        self.port = transport.get_extra_info("sockname")[1]
        if initial_peers:
            await self.bootstrap(initial_peers)
        return self

    async def bootstrap(self, initial_peers: Sequence[Tuple[str, int]]) -> None:
        pongs = await asyncio.gather(*(self.protocol.call_ping(tuple(p)) for p in initial_peers))
        alive = False
        for peer, node_id in zip(initial_peers, pongs):
            if node_id is not None:
                self.protocol.update_routing(node_id, tuple(peer))
                alive = True
        if alive:
            await self.find_nearest_nodes(self.node_id)  # populate buckets around our own id
        else:
            logger.warning("DHTNode.bootstrap: no initial peer responded; starting isolated")

    # ------------------------------------------------------------------
    async def _traverse(self, key_id: int, k: int, find_value: bool,
                        sufficient_expiration: float = float("inf")):
        """Iterative Kademlia lookup with α CONCURRENT workers: each worker
        repeatedly pops the closest unvisited candidate and queries it, so one
        dead/slow peer never stalls the other α-1 query lanes (the reference's
        α-parallelism; batched-gather variants stall a whole round on one
        timeout, which breaks tight liveness leases)."""
        key_int = int(key_id)
        candidates: Dict[int, Tuple[str, int]] = dict(
            self.protocol.routing_table.get_nearest_neighbors(key_int, self.k * 2)
        )
        visited: set = set()
        in_flight: set = set()
        best = {"value": None, "expiration": -float("inf"), "holder": None}
        done_event = asyncio.Event()

        def next_target() -> Optional[int]:
            unvisited = [n for n in candidates if n not in visited and n not in in_flight]
            if not unvisited:
                return None
            return min(unvisited, key=lambda n: n ^ key_int)

        def finished() -> bool:
            if find_value and best["expiration"] >= sufficient_expiration:
                return True
            closest_k = heapq.nsmallest(k, candidates, key=lambda n: n ^ key_int)
            return all(n in visited for n in closest_k)

        async def worker() -> None:
            while not done_event.is_set():
                target = next_target()
                if target is None:
                    if not in_flight:
                        done_event.set()
                        return
                    await asyncio.sleep(0.01)
                    continue
                in_flight.add(target)
                endpoint = candidates[target]
                if self.protocol.is_blacklisted(endpoint):
                    in_flight.discard(target)
                    visited.add(target)
                    if finished():
                        done_event.set()
                        return
                    continue
                if find_value:
                    reply = await self.protocol.call_find_value(endpoint, DHTID(key_int))
                else:
                    reply = await self.protocol.call_find_node(endpoint, DHTID(key_int))
                in_flight.discard(target)
                visited.add(target)
                if reply is None:
                    self.protocol.routing_table.remove_node(target)
                else:
                    if find_value:
                        value, expiration, nearest = reply
                        if value is not None and expiration is not None and expiration > best["expiration"]:
                            best["value"], best["expiration"] = value, expiration
                            best["holder"] = target
                    else:
                        nearest = reply
                    for found_id, ep in nearest:
                        if found_id != int(self.node_id) and found_id not in candidates:
                            candidates[found_id] = ep
                            self.protocol.update_routing(found_id, ep)
                if finished():
                    done_event.set()
                    return

        workers = [asyncio.ensure_future(worker()) for _ in range(self.alpha)]
        await done_event.wait()
        for w in workers:
            w.cancel()
        live = {n: ep for n, ep in candidates.items() if not self.protocol.is_blacklisted(ep)}
        nearest = heapq.nsmallest(k, live, key=lambda n: n ^ key_int)
        return ([(nid, live[nid]) for nid in nearest], best["value"], best["expiration"],
                best["holder"])

    async def find_nearest_nodes(
        self, key_id: int, k: Optional[int] = None
    ) -> List[Tuple[int, Tuple[str, int]]]:
        nearest, _, _, _ = await self._traverse(int(key_id), k or self.k, find_value=False)
        return nearest

    # ------------------------------------------------------------------
    async def store(self, key: Any, value: Any, expiration_time: float,
                    cache_neighbors: bool = False) -> bool:
        """Store (key → value, expiring) on the `replication` nearest peers
        (incl. self). With cache_neighbors=True the traversal result is
        reused for `_store_neighbors_ttl` seconds — the re-announce fast path
        (liveness leases re-store the same keys every announce period)."""
        import time as _time

        key_id = DHTID.generate(key)
        nearest = None
        if cache_neighbors:
            hit = self._store_neighbors.get(int(key_id))
            if hit is not None and hit[0] > _time.monotonic():
                nearest = hit[1]
        if nearest is None:
            nearest = await self.find_nearest_nodes(key_id, k=self.replication)
            if cache_neighbors:
                self._store_neighbors[int(key_id)] = (
                    _time.monotonic() + self._store_neighbors_ttl, nearest
                )
        stored_locally = self.protocol.storage.store(key_id, value, expiration_time)
        if not nearest:
            return stored_locally
        results = await asyncio.gather(
            *(self.protocol.call_store(ep, key_id, value, expiration_time) for _, ep in nearest)
        )
        ok = any(bool(r) for r in results) or stored_locally
        if cache_neighbors and not ok:
            self._store_neighbors.pop(int(key_id), None)  # stale set: re-traverse next time
        return ok

    async def store_many(self, items, cache_neighbors: bool = True) -> bool:
        """Store many (key, value, expiration_time) records efficiently: one
        traversal per key only on cache miss, then ONE batched store RPC per
        destination peer instead of one RPC per (key, replica). This is the
        announce path for servers declaring hundreds of expert uids."""
        import time as _time

        targets: Dict[Any, list] = {}  # endpoint -> [[key_b, value, exp], ...]
        ok_local = True
        for key, value, expiration_time in items:
            key_id = DHTID.generate(key)
            nearest = None
            if cache_neighbors:
                hit = self._store_neighbors.get(int(key_id))
                if hit is not None and hit[0] > _time.monotonic():
                    nearest = hit[1]
            if nearest is None:
                nearest = await self.find_nearest_nodes(key_id, k=self.replication)
                if cache_neighbors:
                    self._store_neighbors[int(key_id)] = (
                        _time.monotonic() + self._store_neighbors_ttl, nearest
                    )
            ok_local &= self.protocol.storage.store(key_id, value, expiration_time)
            for _, ep in nearest:
                targets.setdefault(tuple(ep), []).append(
                    [key_id.to_bytes(), value, expiration_time]
                )

        async def send_batches(ep, batch):
            results = []
            for s0 in range(0, len(batch), 64):  # keep datagrams small
                res = await self.protocol.call_store_many(ep, batch[s0 : s0 + 64])
                results.append(res is not None and all(res))
            return all(results)

        replies = await asyncio.gather(*(send_batches(ep, b) for ep, b in targets.items()))
        return (any(replies) if replies else False) or ok_local

    async def get(self, key: Any, sufficient_expiration: Optional[float] = None) -> Tuple[Optional[Any], Optional[float]]:
        """Iterative find_value; returns (value, expiration) or (None, None).
        Checks local storage and cache first; caches fetched values."""
        key_id = DHTID.generate(key)
        sufficient_expiration = sufficient_expiration or get_dht_time()
        for store in (self.protocol.storage, self.protocol.cache):
            value, expiration = store.get(key_id)
            if value is not None and expiration >= sufficient_expiration:
                return value, expiration

        nearest, best_value, best_expiration, holder = await self._traverse(
            int(key_id), self.k, find_value=True, sufficient_expiration=sufficient_expiration
        )
        if best_value is not None and best_expiration >= get_dht_time():
            self.protocol.cache.store(key_id, best_value, best_expiration)
            # Kademlia read-repair (§2.3 caching): replicate the fetched value
            # at the closest observed node that did NOT return it, so a key's
            # replica set heals toward the ideal one after churn instead of
            # relying solely on the owner's re-announce
            for nid, ep in nearest:
                if nid != holder and nid != int(self.node_id):
                    asyncio.ensure_future(
                        self.protocol.call_store(ep, key_id, best_value, best_expiration))
                    break
            return best_value, best_expiration
        return None, None

    def shutdown(self) -> None:
        if self.protocol is not None and self.protocol.transport is not None:
            self.protocol.transport.close()
