"""DHT process façade (SURVEY.md §2 "DHT process façade", §3.3): runs a
DHTNode on an asyncio loop inside a daemon process; synchronous API via pipe.

Expert discovery contract (the fault-tolerance/discovery mechanism — see
SURVEY.md §5 "Failure detection"): servers `declare_experts` under a liveness
lease (`expiration`); dead servers vanish from discovery when the lease
lapses; clients `get_experts` / `first_k_active` to resolve uids → endpoints.

Key scheme: full uid "ffn.3.7" → [host, port] under key "expert.ffn.3.7";
every proper dot-prefix is also declared ("prefix.ffn.3", "prefix.ffn") so
that MoE beam search can test prefix liveness in O(1) DHT gets.
"""
from __future__ import annotations

import asyncio
import multiprocessing as mp
import os
import threading
from typing import Any, Dict, List, Optional, Sequence, Tuple

from ..utils import get_dht_time, get_logger
from .node import DHTNode
from .routing import DHTID, RoutingTable, KBucket  # re-export for tests
from .protocol import DHTProtocol

logger = get_logger(__name__)

UID_DELIMITER = "."
EXPERT_PREFIX = "expert."
PREFIX_PREFIX = "prefix."


def uid_prefixes(uid: str) -> List[str]:
    """All proper dot-prefixes of an expert uid, e.g. 'ffn.3.7' → ['ffn', 'ffn.3']."""
    parts = uid.split(UID_DELIMITER)
    return [UID_DELIMITER.join(parts[: i + 1]) for i in range(len(parts) - 1)]


class DHT(mp.Process):
    """A Kademlia DHT peer in a daemon process with a synchronous façade.

    >>> dht = DHT(start=True)
    >>> dht.store("key", "value", get_dht_time() + 60)
    >>> dht.get("key")  # -> ("value", expiration)
    """

    def __init__(
        self,
        listen_on: Tuple[str, int] = ("127.0.0.1", 0),
        initial_peers: Sequence[Tuple[str, int]] = (),
        *,
        start: bool = False,
        daemon: bool = True,
        bucket_size: int = 20,
        alpha: int = 3,
        rpc_timeout: float = 3.0,
    ):
        super().__init__(daemon=daemon)
        self.listen_on = tuple(listen_on)
        self.initial_peers = [tuple(p) for p in initial_peers]
        self.bucket_size, self.alpha, self.rpc_timeout = bucket_size, alpha, rpc_timeout
        self._parent_conn, self._child_conn = mp.Pipe()
        self._port = mp.Value("i", 0)
        self.ready = mp.Event()
        self._lock_obj: Optional[threading.Lock] = None  # lazy: keeps DHT spawn-picklable
        if start:
            self.run_in_background()

    @property
    def _lock(self) -> threading.Lock:
        if self._lock_obj is None:
            self._lock_obj = threading.Lock()
        return self._lock_obj

    def __getstate__(self):
        state = self.__dict__.copy()
        state["_lock_obj"] = None
        return state

    # ---- lifecycle ----------------------------------------------------
    def run_in_background(self, await_ready: bool = True, timeout: float = 120.0) -> None:
        # generous: a spawn-context child re-imports torch (~2-10 s, worse under load)
        self.start()
        if await_ready and not self.ready.wait(timeout):
            raise TimeoutError("DHT process failed to start")

    def run(self) -> None:  # executed in the child process
        loop = asyncio.new_event_loop()
        asyncio.set_event_loop(loop)
        node = loop.run_until_complete(
            DHTNode.create(
                listen_on=self.listen_on,
                initial_peers=self.initial_peers,
                bucket_size=self.bucket_size,
                alpha=self.alpha,
                rpc_timeout=self.rpc_timeout,
            )
        )
        with self._port.get_lock():
            self._port.value = node.port
        self.ready.set()

        conn = self._child_conn

        def on_command() -> None:
            try:
                req_id, method, args, kwargs = conn.recv()
            except (EOFError, OSError):
                loop.stop()
                return
            if method == "_shutdown":
                node.shutdown()
                loop.stop()
                conn.send((req_id, "ok", None))
                return
            asyncio.ensure_future(dispatch(req_id, method, args, kwargs))

        async def dispatch(req_id, method, args, kwargs) -> None:
            try:
                result = await getattr(self, f"_rpc_{method}")(node, *args, **kwargs)
                conn.send((req_id, "ok", result))
            except Exception as e:  # noqa: BLE001
                conn.send((req_id, "error", e))

        loop.add_reader(conn.fileno(), on_command)
        try:
            loop.run_forever()
        finally:
            node.shutdown()

    def shutdown(self) -> None:
        if self.is_alive():
            try:
                self._call("_shutdown", timeout=5.0)
            except Exception:  # noqa: BLE001
                pass
            self.join(timeout=5.0)
            if self.is_alive():
                self.terminate()

    @property
    def port(self) -> int:
        return self._port.value

    @property
    def endpoint(self) -> Tuple[str, int]:
        host = self.listen_on[0] if self.listen_on[0] not in ("0.0.0.0", "") else "127.0.0.1"
        return (host, self.port)

    # ---- sync façade ---------------------------------------------------
    def _call(self, method: str, *args, timeout: Optional[float] = None, **kwargs) -> Any:
        with self._lock:
            req_id = os.urandom(8)
            self._parent_conn.send((req_id, method, args, kwargs))
            if not self._parent_conn.poll(timeout if timeout is not None else 60.0):
                raise TimeoutError(f"DHT call {method} timed out")
            got_id, status, payload = self._parent_conn.recv()
            assert got_id == req_id, "DHT pipe protocol desync"
        if status == "error":
            raise payload
        return payload

    def store(self, key: Any, value: Any, expiration_time: float) -> bool:
        return self._call("store", key, value, expiration_time)

    def get(self, key: Any, sufficient_expiration: Optional[float] = None) -> Tuple[Optional[Any], Optional[float]]:
        return tuple(self._call("get", key, sufficient_expiration))

    def declare_experts(self, uids: Sequence[str], addr: str, port: int, expiration: float = 300.0) -> bool:
        """Announce experts under a liveness lease of `expiration` seconds."""
        return self._call("declare_experts", list(uids), addr, port, expiration)

    def get_experts(self, uids: Sequence[str]) -> List[Optional[Tuple[str, int]]]:
        """Resolve full uids → endpoints ((host, port)); None for unknown/expired."""
        raw = self._call("get_experts", list(uids))
        return [tuple(ep) if ep is not None else None for ep in raw]

    def first_k_active(self, prefixes: Sequence[str], k: int) -> Dict[str, Tuple[str, int]]:
        """Of `prefixes` (in order), return the first k that have at least one
        live declared expert, mapped to one live endpoint each."""
        raw = self._call("first_k_active", list(prefixes), k)
        return {prefix: tuple(ep) for prefix, ep in raw}

    # ---- async implementations (run inside the DHT process) ------------
    @staticmethod
    async def _rpc_store(node: DHTNode, key, value, expiration_time) -> bool:
        return await node.store(key, value, expiration_time)

    @staticmethod
    async def _rpc_get(node: DHTNode, key, sufficient_expiration=None):
        return await node.get(key, sufficient_expiration)

    @staticmethod
    async def _rpc_declare_experts(node: DHTNode, uids, addr, port, expiration) -> bool:
        expiration_time = get_dht_time() + expiration
        endpoint = [addr, port]
        items = []
        seen_prefixes = set()
        for uid in uids:
            items.append((EXPERT_PREFIX + uid, endpoint, expiration_time))
            for prefix in uid_prefixes(uid):
                if prefix not in seen_prefixes:
                    seen_prefixes.add(prefix)
                    items.append((PREFIX_PREFIX + prefix, endpoint, expiration_time))
        # batched multi-key store: one traversal per key only on neighbor-cache
        # miss, one store RPC per destination peer (not per key x replica)
        return await node.store_many(items, cache_neighbors=True)

    @staticmethod
    async def _rpc_get_experts(node: DHTNode, uids):
        # bounded concurrency: an unbounded gather over many uids bursts
        # hundreds of UDP datagrams at once; dropped ones each cost a full
        # rpc_timeout (measured: 64 uids @ 48 peers 3.0s -> ~0.1s bounded)
        sem = asyncio.Semaphore(16)

        async def bounded_get(uid):
            async with sem:
                return await node.get(EXPERT_PREFIX + uid)

        results = await asyncio.gather(*(bounded_get(uid) for uid in uids))
        return [value for value, _expiration in results]

    @staticmethod
    async def _rpc_first_k_active(node: DHTNode, prefixes, k):
        """Probe prefixes in order with bounded concurrency, preserving order."""
        sem = asyncio.Semaphore(8)

        async def probe(prefix):
            async with sem:
                value, _ = await node.get(PREFIX_PREFIX + prefix)
                return prefix, value

        results = await asyncio.gather(*(probe(p) for p in prefixes))
        active = []
        for prefix, value in results:
            if value is not None:
                active.append([prefix, value])
                if len(active) >= k:
                    break
        return active


__all__ = ["DHT", "DHTNode", "DHTID", "DHTProtocol", "RoutingTable", "KBucket",
           "uid_prefixes", "UID_DELIMITER", "EXPERT_PREFIX", "PREFIX_PREFIX"]
