"""DHTProtocol — Kademlia RPC over asyncio UDP (SURVEY.md §2 "DHT node":
rpc ping / store / find_node / find_value with msgpack wire encoding).

Wire format (one datagram per message):
  request:  {"q": nonce, "rpc": name, "sender": id_bytes, "args": [...]}
  response: {"r": nonce, "sender": id_bytes, "value": ...}
Values are small control-plane records (endpoints + expirations) — a UDP
datagram always suffices; tensors never travel through the DHT.
"""
from __future__ import annotations

import asyncio
import os
from typing import Any, Dict, List, Optional, Tuple

from ..utils import MSGPackSerializer, TimedStorage, get_logger
from .routing import DHTID, RoutingTable

logger = get_logger(__name__)

DEFAULT_RPC_TIMEOUT = 3.0


class DHTProtocol(asyncio.DatagramProtocol):
    def __init__(self, node_id: DHTID, bucket_size: int = 20, cache_size: int = 10000,
                 rpc_timeout: float = DEFAULT_RPC_TIMEOUT):
        self.node_id = node_id
        self.routing_table = RoutingTable(node_id, bucket_size)
        self.storage = TimedStorage()
        self.cache = TimedStorage(maxsize=cache_size)
        self.rpc_timeout = rpc_timeout
        self.transport: Optional[asyncio.DatagramTransport] = None
        self._pending: Dict[bytes, asyncio.Future] = {}
        self.bucket_size = bucket_size
        # failure blacklist: endpoints that timed out recently are skipped by
        # lookups (dead peers keep being re-learned from neighbors' routing
        # tables, and paying rpc_timeout per lookup per dead peer starves
        # announce leases — SURVEY.md §5 failure-detection semantics)
        self._blacklist: Dict[Tuple[str, int], float] = {}
        self.blacklist_ttl = 15.0

    # ------------------------------------------------------------------
    def connection_made(self, transport: asyncio.DatagramTransport) -> None:
        self.transport = transport

    def datagram_received(self, data: bytes, addr: Tuple[str, int]) -> None:
        try:
            msg = MSGPackSerializer.loads(data)
        except Exception:
            return
        self._blacklist.pop(tuple(addr), None)  # endpoint is provably alive
        if "q" in msg:
            asyncio.ensure_future(self._handle_request(msg, addr))
        elif "r" in msg:
            nonce = msg["r"]
            fut = self._pending.pop(nonce, None)
            if fut is not None and not fut.done():
                sender = DHTID.from_bytes(msg["sender"])
                self.routing_table.add_or_update_node(int(sender), list(addr))
                fut.set_result(msg.get("value"))

    # ---- client side ---------------------------------------------------
    async def call(self, endpoint: Tuple[str, int], rpc: str, *args: Any) -> Optional[Any]:
        """Send one RPC; return response value or None on timeout/failure."""
        nonce = os.urandom(8)
        payload = MSGPackSerializer.dumps(
            {"q": nonce, "rpc": rpc, "sender": self.node_id.to_bytes(), "args": list(args)}
        )
        fut: asyncio.Future = asyncio.get_event_loop().create_future()
        self._pending[nonce] = fut
        try:
            self.transport.sendto(payload, tuple(endpoint))
            result = await asyncio.wait_for(fut, self.rpc_timeout)
            self._blacklist.pop(tuple(endpoint), None)
            return result
        except (asyncio.TimeoutError, OSError):
            self._pending.pop(nonce, None)
            import time as _time

            self._blacklist[tuple(endpoint)] = _time.monotonic() + self.blacklist_ttl
            return None

    def is_blacklisted(self, endpoint) -> bool:
        import time as _time

        deadline = self._blacklist.get(tuple(endpoint))
        if deadline is None:
            return False
        if _time.monotonic() >= deadline:
            del self._blacklist[tuple(endpoint)]
            return False
        return True

    async def call_ping(self, endpoint) -> Optional[int]:
        value = await self.call(endpoint, "ping")
        return None if value is None else int.from_bytes(value, "big")

    async def call_store(self, endpoint, key_id: DHTID, value: Any, expiration: float) -> Optional[bool]:
        res = await self.call(endpoint, "store", key_id.to_bytes(), value, expiration)
        return None if res is None else bool(res)

    async def call_store_many(self, endpoint, items) -> Optional[list]:
        """items: list of [key_id_bytes, value, expiration]; one datagram."""
        res = await self.call(endpoint, "store_many", items)
        return None if res is None else list(res)

    async def call_find_node(self, endpoint, key_id: DHTID) -> Optional[List[Tuple[int, Any]]]:
        res = await self.call(endpoint, "find_node", key_id.to_bytes())
        if res is None:
            return None
        return [(int.from_bytes(nid, "big"), tuple(ep)) for nid, ep in res]

    async def call_find_value(self, endpoint, key_id: DHTID):
        """Returns (maybe_value, maybe_expiration, nearest_nodes) or None."""
        res = await self.call(endpoint, "find_value", key_id.to_bytes())
        if res is None:
            return None
        value, expiration, nearest = res
        nearest = [(int.from_bytes(nid, "big"), tuple(ep)) for nid, ep in nearest]
        return value, expiration, nearest

    # ---- server side ---------------------------------------------------
    async def _handle_request(self, msg: dict, addr: Tuple[str, int]) -> None:
        sender = DHTID.from_bytes(msg["sender"])
        # Every peer sends RPCs from the socket it listens on, so the UDP
        # source address IS the peer's listening endpoint: learn it.
        self.routing_table.add_or_update_node(int(sender), list(addr))
        rpc, args = msg["rpc"], msg.get("args", [])
        value: Any = None
        if rpc == "ping":
            value = self.node_id.to_bytes()
        elif rpc == "store":
            key_b, val, expiration = args
            value = self.storage.store(DHTID.from_bytes(key_b), val, float(expiration))
        elif rpc == "store_many":
            value = [
                self.storage.store(DHTID.from_bytes(kb), val, float(exp))
                for kb, val, exp in args[0]
            ]
        elif rpc == "find_node":
            key = DHTID.from_bytes(args[0])
            nearest = self.routing_table.get_nearest_neighbors(int(key), self.bucket_size, exclude=(int(sender),))
            value = [[nid.to_bytes(20, "big"), list(ep)] for nid, ep in nearest]
        elif rpc == "find_value":
            key = DHTID.from_bytes(args[0])
            stored, expiration = self.storage.get(key)
            if stored is None:
                stored, expiration = self.cache.get(key)
            nearest = self.routing_table.get_nearest_neighbors(int(key), self.bucket_size, exclude=(int(sender),))
            value = [stored, expiration, [[nid.to_bytes(20, "big"), list(ep)] for nid, ep in nearest]]
        reply = MSGPackSerializer.dumps({"r": msg["q"], "sender": self.node_id.to_bytes(), "value": value})
        if self.transport is not None:
            self.transport.sendto(reply, addr)

    def update_routing(self, node_id: int, endpoint) -> None:
        self.routing_table.add_or_update_node(node_id, list(endpoint))
