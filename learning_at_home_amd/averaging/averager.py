"""DecentralizedAverager — group parameter averaging with DHT matchmaking
(SURVEY.md §2 "DecentralizedAverager", §3.4 call stack; Moshpit SGD rounds).

Matchmaking (Moshpit grid): the prefix owns `num_group_slots` concurrent
group slots, each with its own liveness-leased leader key
"{prefix}.slot{i}.leader". A peer scans the slots in random order and joins
the first joinable leader ('join' TCP request); if every advertised leader
is full/stale it becomes the leader of a random free slot. Leaders accept
joiners until `averaging_expiration` elapses (or the group is full), then
broadcast the member list. With P > target_group_size peers, multiple slots
host groups CONCURRENTLY, and the random slot assignment re-shuffles group
composition every round, so repeated rounds mix values globally (Moshpit
SGD, arXiv:2103.03239) instead of serializing through one leader.

One averaging round = butterfly all-reduce by parts:
member i owns part i of the flat parameter vector; everyone sends part i to
member i ('reduce' RPC), the owner returns the weighted average in the RPC
reply (reduce-scatter + all-gather in one round trip per part).

Each live averager also re-announces itself under a hashed per-peer state
key "{prefix}.peer{h}" (liveness-leased), so `load_state_from_peers` can
find ANY live peer for a cold start — not just a transient round leader.

Fault tolerance: any member failing mid-round aborts this peer's round
gracefully (tensors left unchanged; step() returns None) — the next round
re-forms a group without the dead peer, matching the reference's
renormalize-or-abort semantics at the round granularity.

The RCCL side-stream variant for intra-node GPU groups is rccl.py.
"""
from __future__ import annotations

import os
import random
import socket
import threading
import time
from concurrent.futures import ThreadPoolExecutor
from contextlib import contextmanager
from typing import Dict, List, Optional, Sequence, Tuple

import torch

from ..dht import DHT
from ..utils import (
    Endpoint,
    PytorchSerializer,
    get_dht_time,
    get_logger,
    recv_message,
    send_message,
)

logger = get_logger(__name__)


class _GroupInfo:
    def __init__(self, group_id: str, members: List[Endpoint], weights: List[float]):
        self.group_id = group_id
        self.members = [tuple(m) for m in members]
        self.weights = list(weights)

    @property
    def size(self) -> int:
        return len(self.members)


class _RoundState:
    """Per-round accumulator for the part this peer owns."""

    def __init__(self, expected_ranks: int):
        self.expected = expected_ranks
        self.parts: Dict[int, Tuple[torch.Tensor, float]] = {}
        self.ready = threading.Event()
        self.result: Optional[torch.Tensor] = None
        self.lock = threading.Lock()

    def add_part(self, rank: int, part: torch.Tensor, weight: float) -> None:
        with self.lock:
            self.parts[rank] = (part, weight)
            self._maybe_finalize_locked()

    def set_expected(self, expected: int) -> None:
        with self.lock:
            self.expected = expected
            self._maybe_finalize_locked()

    def _maybe_finalize_locked(self) -> None:
        # expected == -1 means the group size is not yet known here
        if self.expected > 0 and len(self.parts) >= self.expected and self.result is None:
            total_w = sum(w for _, w in self.parts.values())
            any_part = next(iter(self.parts.values()))[0]
            acc = torch.zeros_like(any_part, dtype=torch.float32)
            for p, w in self.parts.values():
                acc += p.float() * (w / total_w)
            self.result = acc
            self.ready.set()


class DecentralizedAverager:
    def __init__(
        self,
        averaged_tensors: Sequence[torch.Tensor],
        dht: DHT,
        *,
        prefix: str,
        target_group_size: int = 8,
        averaging_expiration: float = 3.0,
        request_timeout: float = 5.0,
        listen_on: Endpoint = ("127.0.0.1", 0),
        compression: str = "none",
        num_group_slots: int = 4,
        state_announce_period: float = 2.0,
        start: bool = False,
    ):
        self.tensors = list(averaged_tensors)
        self.dht = dht
        self.prefix = prefix
        self.target_group_size = target_group_size
        self.averaging_expiration = averaging_expiration
        self.request_timeout = request_timeout
        self.listen_on = listen_on
        # "bf16": ship butterfly parts and averaged replies as bfloat16 over
        # the wire (half the bytes; owners still accumulate in fp32) — the
        # reference's tensor-compression knob. All group peers must agree.
        assert compression in ("none", "bf16"), compression
        self.compression = compression
        assert num_group_slots >= 1
        self.num_group_slots = num_group_slots
        self.state_announce_period = state_announce_period
        self._tensor_lock = threading.Lock()
        self._rounds: Dict[str, _RoundState] = {}
        self._rounds_lock = threading.Lock()
        self._leader_state: Optional[dict] = None
        self._leader_lock = threading.Lock()
        self._server_sock: Optional[socket.socket] = None
        self.port: Optional[int] = None
        self._stop = threading.Event()
        self.num_rounds = 0
        if start:
            self.run_in_background()

    # ---- lifecycle ----------------------------------------------------
    def run_in_background(self) -> None:
        self._server_sock = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
        self._server_sock.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
        self._server_sock.bind(self.listen_on)
        self._server_sock.listen(64)
        self.port = self._server_sock.getsockname()[1]
        threading.Thread(target=self._accept_loop, daemon=True, name="averager_server").start()
        threading.Thread(target=self._announce_loop, daemon=True, name="averager_announce").start()

    def _announce_loop(self) -> None:
        """Liveness-leased self-announcement so cold-starting peers can find
        ANY live averager's state (don't rely on the transient
        leader key). Lease = 2× period, matching the server announcer."""
        key = self._state_key
        while not self._stop.is_set():
            try:
                self.dht.store(key, list(self.endpoint),
                               get_dht_time() + 2 * self.state_announce_period)
            except Exception as e:  # noqa: BLE001
                logger.debug(f"averager state announce failed: {e!r}")
            self._stop.wait(self.state_announce_period)

    @property
    def endpoint(self) -> Endpoint:
        host = self.listen_on[0] if self.listen_on[0] not in ("0.0.0.0", "") else "127.0.0.1"
        return (host, self.port)

    def shutdown(self) -> None:
        self._stop.set()
        if self._server_sock is not None:
            # closing the listener does NOT wake a thread blocked in accept()
            # on Linux — poke it with a self-connection first so the accept
            # loop observes _stop and exits (verified: the accept thread
            # outlived shutdown forever without this)
            try:
                with socket.create_connection(("127.0.0.1", self.port), timeout=1.0):
                    pass
            except OSError:
                pass
            try:
                self._server_sock.close()
            except OSError:
                pass

    @contextmanager
    def get_tensors(self):
        """Caller-facing lock: averaged tensors must be read/written under it
        (the round writes results back under the same lock, so an optimizer
        step never interleaves with the averager's write-back)."""
        with self._tensor_lock:
            yield self.tensors

    # ---- server side ---------------------------------------------------
    def _accept_loop(self) -> None:
        while not self._stop.is_set():
            try:
                conn, _ = self._server_sock.accept()
            except OSError:
                return
            threading.Thread(target=self._serve_conn, args=(conn,), daemon=True).start()

    def _serve_conn(self, conn: socket.socket) -> None:
        try:
            conn.settimeout(self.request_timeout + self.averaging_expiration + 5.0)
            while not self._stop.is_set():
                try:
                    kind, payload = recv_message(conn)
                except (ConnectionError, socket.timeout, OSError):
                    return
                try:
                    reply_kind, reply = self._handle(kind, payload)
                except Exception as e:  # noqa: BLE001
                    reply_kind, reply = "error", repr(e).encode()
                try:
                    send_message(conn, reply_kind, reply)
                except OSError:
                    return
        finally:
            try:
                conn.close()
            except OSError:
                pass

    def _handle(self, kind: str, payload: bytes) -> Tuple[str, bytes]:
        if kind == "join":
            endpoint, weight = PytorchSerializer.loads(payload)
            return self._handle_join(tuple(endpoint), float(weight))
        elif kind == "reduce":
            group_id, rank, weight, part = PytorchSerializer.loads(payload)
            return self._handle_reduce(group_id, int(rank), float(weight), part)
        elif kind == "get_state":
            with self.get_tensors() as tensors:
                blob = PytorchSerializer.dumps([t.detach().cpu() for t in tensors])
            return "result", blob
        raise ValueError(f"unknown averager message {kind!r}")

    def _handle_join(self, endpoint: Endpoint, weight: float) -> Tuple[str, bytes]:
        with self._leader_lock:
            state = self._leader_state
            if state is None or state["closed"] or len(state["members"]) >= self.target_group_size:
                return "retry", b""
            state["members"].append(endpoint)
            state["weights"].append(weight)
            event: threading.Event = state["event"]
            if len(state["members"]) >= self.target_group_size:
                self._finalize_group_locked()
        # wait (outside leader lock) for the leader to finalize the group
        if not event.wait(self.averaging_expiration + self.request_timeout):
            return "retry", b""
        state_final = state["final"]
        if state_final is None or endpoint not in state_final.members:
            return "retry", b""
        return "group", PytorchSerializer.dumps(
            [state_final.group_id, [list(m) for m in state_final.members], state_final.weights]
        )

    def _finalize_group_locked(self) -> None:
        state = self._leader_state
        if state is None or state["closed"]:
            return
        state["closed"] = True
        if len(state["members"]) >= 2:
            group_id = state["group_id"]
            state["final"] = _GroupInfo(group_id, state["members"], state["weights"])
        else:
            state["final"] = None
        state["event"].set()

    def _handle_reduce(self, group_id: str, rank: int, weight: float, part: torch.Tensor) -> Tuple[str, bytes]:
        with self._rounds_lock:
            round_state = self._rounds.get(group_id)
            if round_state is None:
                # remote peer reached its reduce before we even formed the
                # round locally; create a placeholder sized lazily
                round_state = self._rounds[group_id] = _RoundState(expected_ranks=-1)
        round_state.add_part(rank, part, weight)
        if not round_state.ready.wait(self.request_timeout + self.averaging_expiration):
            # drop the placeholder so aborted rounds don't accumulate forever
            # (idempotent: a concurrent local _run_allreduce holds its own
            # reference and pops again in its finally block)
            with self._rounds_lock:
                if self._rounds.get(group_id) is round_state and round_state.result is None:
                    self._rounds.pop(group_id, None)
            raise TimeoutError("averaging round timed out waiting for parts")
        result = round_state.result
        if self.compression == "bf16":
            result = result.bfloat16()
        return "result", PytorchSerializer.dumps(result)

    # ---- client side ---------------------------------------------------
    def _rpc(self, endpoint: Endpoint, kind: str, obj, timeout: Optional[float] = None):
        timeout = timeout or (self.request_timeout + self.averaging_expiration + 5.0)
        with socket.create_connection(endpoint, timeout=timeout) as sock:
            sock.setsockopt(socket.IPPROTO_TCP, socket.TCP_NODELAY, 1)
            send_message(sock, kind, PytorchSerializer.dumps(obj))
            reply_kind, payload = recv_message(sock)
        if reply_kind == "error":
            raise RuntimeError(payload.decode(errors="replace"))
        if reply_kind == "retry":
            return None
        return PytorchSerializer.loads(payload)

    _STATE_KEY_SLOTS = 16

    def _slot_leader_key(self, slot: int) -> str:
        return f"{self.prefix}.slot{slot}.leader"

    @property
    def _state_key(self) -> str:
        import zlib

        h = zlib.crc32(repr(self.endpoint).encode()) % self._STATE_KEY_SLOTS
        return f"{self.prefix}.peer{h}"

    def step(self, weight: float = 1.0, timeout: Optional[float] = None) -> Optional[_GroupInfo]:
        """One averaging round: matchmake on the slot grid, butterfly
        all-reduce, write back. Returns the group info on success, None if no
        group formed / aborted. Slot order is randomized per attempt so group
        composition re-shuffles every round (Moshpit global mixing)."""
        deadline = time.monotonic() + (timeout if timeout is not None else
                                       3 * (self.averaging_expiration + self.request_timeout))
        while time.monotonic() < deadline and not self._stop.is_set():
            slots = list(range(self.num_group_slots))
            random.shuffle(slots)
            free_slots = []
            joined = None
            for slot in slots:
                leader_ep, _expiration = self.dht.get(self._slot_leader_key(slot))
                if leader_ep is None or tuple(leader_ep) == self.endpoint:
                    free_slots.append(slot)
                    continue
                reply = None
                try:
                    reply = self._rpc(tuple(leader_ep), "join", [list(self.endpoint), weight])
                except (OSError, RuntimeError):
                    free_slots.append(slot)  # stale/dead leader: slot is takeable
                    continue
                if reply is not None:
                    joined = _GroupInfo(reply[0], [tuple(m) for m in reply[1]], reply[2])
                    break
                # leader full/closed: try the next slot
            if joined is not None:
                return self._run_allreduce(joined, weight)
            lead_slot = random.choice(free_slots) if free_slots else random.choice(slots)
            if self._try_lead(lead_slot, weight):
                group = self._leader_state["final"]
                self._leader_state = None
                if group is not None:
                    return self._run_allreduce(group, weight)
            # jittered retry: two peers starting simultaneously otherwise
            # re-collide on the leader key in lock-step (same failure shape
            # as the elastic-group rendezvous livelock; see elastic.py)
            time.sleep(0.05 + random.random() * 0.2)
        return None

    def _try_lead(self, slot: int, weight: float) -> bool:
        """Become leader of `slot`: advertise in DHT, collect joiners until
        expiration."""
        with self._leader_lock:
            group_id = os.urandom(8).hex()
            self._leader_state = {
                "group_id": group_id,
                "members": [self.endpoint],
                "weights": [weight],
                "closed": False,
                "final": None,
                "event": threading.Event(),
            }
        stored = self.dht.store(self._slot_leader_key(slot), list(self.endpoint),
                                get_dht_time() + self.averaging_expiration)
        if not stored:
            with self._leader_lock:
                self._leader_state = None
            return False
        event = self._leader_state["event"]
        event.wait(self.averaging_expiration)  # joiners may fill the group early
        with self._leader_lock:
            self._finalize_group_locked()
        return True

    def _run_allreduce(self, group: _GroupInfo, my_weight: float) -> Optional[_GroupInfo]:
        try:
            rank = group.members.index(self.endpoint)
        except ValueError:
            return None
        M = group.size
        with self.get_tensors() as tensors:
            flat = torch.cat([t.detach().reshape(-1).float() for t in tensors])
        numel = flat.numel()
        part_len = (numel + M - 1) // M
        padded = torch.zeros(part_len * M, dtype=torch.float32)
        padded[:numel] = flat
        parts = padded.chunk(M)

        with self._rounds_lock:
            if group.group_id not in self._rounds:
                self._rounds[group.group_id] = _RoundState(expected_ranks=-1)
            round_state = self._rounds[group.group_id]
        round_state.set_expected(M)
        own_part = parts[rank].clone()
        if self.compression == "bf16":  # match what remote peers' parts lost
            own_part = own_part.bfloat16()
        round_state.add_part(rank, own_part, my_weight)

        try:
            results: List[Optional[torch.Tensor]] = [None] * M
            with ThreadPoolExecutor(max_workers=max(1, M - 1)) as pool:
                wire = (lambda t: t.bfloat16()) if self.compression == "bf16" else (lambda t: t.clone())
                futures = {
                    j: pool.submit(self._rpc, group.members[j], "reduce",
                                   [group.group_id, rank, my_weight, wire(parts[j])])
                    for j in range(M) if j != rank
                }
                for j, fut in futures.items():
                    reply = fut.result(timeout=self.request_timeout + self.averaging_expiration + 5.0)
                    if reply is None:
                        raise RuntimeError(f"peer {group.members[j]} rejected reduce")
                    results[j] = reply
            if not round_state.ready.wait(self.request_timeout + self.averaging_expiration):
                raise TimeoutError("own part never completed")
            results[rank] = round_state.result
            averaged = torch.cat([r.float() for r in results])[:numel]
            with self.get_tensors() as tensors:
                offset = 0
                for t in tensors:
                    n = t.numel()
                    t.copy_(averaged[offset : offset + n].reshape(t.shape).to(t.dtype))
                    offset += n
            self.num_rounds += 1
            return group
        except Exception as e:  # noqa: BLE001
            logger.warning(f"averaging round {group.group_id} aborted: {e!r}")
            return None
        finally:
            with self._rounds_lock:
                self._rounds.pop(group.group_id, None)

    # ---- state sharing (cold start) ------------------------------------
    def _live_peer_candidates(self) -> List[Endpoint]:
        """All endpoints advertised under the prefix: per-peer state keys
        (liveness-leased self-announcements) plus any current slot leaders."""
        keys = [f"{self.prefix}.peer{h}" for h in range(self._STATE_KEY_SLOTS)]
        keys += [self._slot_leader_key(s) for s in range(self.num_group_slots)]
        seen, candidates = set(), []
        for key in keys:
            ep, _ = self.dht.get(key)
            if ep is None:
                continue
            ep = tuple(ep)
            if ep != self.endpoint and ep not in seen:
                seen.add(ep)
                candidates.append(ep)
        return candidates

    def load_state_from_peers(self, timeout: Optional[float] = None) -> bool:
        """Fetch current tensors from ANY live peer advertised under the
        prefix (per-peer state announcements, falling back to slot leaders) —
        not just a transient round leader."""
        candidates = self._live_peer_candidates()
        random.shuffle(candidates)
        for ep in candidates:
            try:
                tensors = self._rpc(ep, "get_state", None, timeout)
            except (OSError, RuntimeError):
                continue
            if tensors is None:
                continue
            with self.get_tensors() as mine:
                for dst, src in zip(mine, tensors):
                    dst.copy_(src.to(dst.dtype))
            return True
        return False
