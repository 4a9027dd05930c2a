"""Elastic expert-parallel membership (SURVEY.md §5 "Failure detection /
elastic recovery", BASELINE.json config #5: drop/rejoin an expert rank
mid-epoch with DHT re-discovery).

Mechanism (the reference's lease semantics, lifted to RCCL):
- every EP rank owns a STABLE SLOT (0..max_slots-1) and announces
  "{prefix}.rank.{slot}" -> (host, port) under a liveness lease; a dead rank
  vanishes when its lease lapses;
- the lowest-slot alive rank is the COORDINATOR: it publishes
  "{prefix}.gen" -> [gen_no, [slots...], master_port] whenever membership
  changes;
- on a new generation every alive rank tears down its process group and
  re-initializes torch.distributed (RCCL/gloo) over the listed slots via a
  fresh TCPStore rendezvous — communicators are static in RCCL, so elasticity
  = generation-fenced rebuild (SURVEY.md §5 "RCCL wrinkle");
- experts stay OWNED by their slot: a dead slot's experts simply drop out of
  routing (the MoE layer masks their gating logits to -inf — the reference's
  "drop dead experts, renormalize over survivors"), and return when the slot
  rejoins (optionally restored from its checkpoint).
"""
from __future__ import annotations

import datetime
import os
import random
import socket
import threading
import time
from typing import List, Optional, Tuple

import torch.distributed as dist

from ..dht import DHT
from ..utils import get_dht_time, get_logger

logger = get_logger(__name__)


def _free_port() -> int:
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


class ElasticGroup:
    def __init__(
        self,
        dht: DHT,
        slot: int,
        max_slots: int = 64,
        *,
        prefix: str = "ep",
        lease: float = 6.0,
        master_addr: str = "127.0.0.1",
        backend: str = "gloo",
        timeout: float = 60.0,
    ):
        self.dht = dht
        self.slot = slot
        self.max_slots = max_slots
        self.prefix = prefix
        self.lease = lease
        self.master_addr = master_addr
        self.backend = backend
        self.timeout = timeout
        self.gen = -1
        self.members: List[int] = []
        self.group = None  # the default PG after re-init
        self._stop = threading.Event()
        self._announcer: Optional[threading.Thread] = None

    # ---- liveness ------------------------------------------------------
    def _rank_key(self, slot: int) -> str:
        return f"{self.prefix}.rank.{slot}"

    def announce(self) -> None:
        self.dht.store(self._rank_key(self.slot), [self.master_addr, os.getpid()],
                       get_dht_time() + self.lease)

    def start_announcer(self) -> None:
        self.announce()

        def loop():
            while not self._stop.wait(self.lease / 3 + random.random() * self.lease / 10):
                try:
                    self.announce()
                except Exception as e:  # noqa: BLE001
                    logger.warning(f"announce failed: {e!r}")

        self._announcer = threading.Thread(target=loop, daemon=True, name="ep_announcer")
        self._announcer.start()

    def alive_slots(self) -> List[int]:
        alive = []
        for s in range(self.max_slots):
            value, _ = self.dht.get(self._rank_key(s))
            if value is not None:
                alive.append(s)
        return alive

    def _stable_alive(self, settle: float = 0.5, max_wait: float = 10.0) -> List[int]:
        """Debounced membership view: right after a peer dies, survivors'
        announce traversals stall on the dead DHT node (rpc timeout) long
        enough for EVERY lease to lapse — a transient window where each rank
        sees only itself. Require two consecutive identical views before
        acting on membership."""
        deadline = time.monotonic() + max_wait
        prev = self.alive_slots()
        while time.monotonic() < deadline and not self._stop.is_set():
            time.sleep(settle)
            cur = self.alive_slots()
            if cur == prev and self.slot in cur:
                return cur
            prev = cur
        return sorted(set(prev + [self.slot]))

    # ---- generation protocol -------------------------------------------
    @property
    def _gen_key(self) -> str:
        return f"{self.prefix}.gen"

    def _published_gen(self) -> Optional[Tuple[int, List[int], int]]:
        value, _ = self.dht.get(self._gen_key)
        if value is None:
            return None
        gen_no, slots, port = value
        return int(gen_no), [int(s) for s in slots], int(port)

    def _publish_gen(self, gen_no: int, slots: List[int], port: int) -> None:
        self.dht.store(self._gen_key, [gen_no, slots, port], get_dht_time() + 3600)

    def membership_changed(self) -> bool:
        """Non-blocking-ish check the engine calls between steps."""
        alive = self.alive_slots()
        return sorted(alive) != sorted(self.members)

    # ---- bounded-time comm teardown (SURVEY §7 hard part (b)) ----------
    def abort_comm(self, timeout: float = 10.0) -> None:
        """Tear down the current process group WITHOUT deadlocking on
        in-flight collectives. A rank that died mid-all-to-all leaves
        survivors' RCCL kernels spinning; plain destroy_process_group would
        then block on the pending work. _abort_process_group kills the
        communicator (and its in-flight ops) immediately; the watchdogged
        destroy is the fallback for backends without an abort (gloo)."""
        if not dist.is_initialized():
            self.group = None
            return
        try:
            from torch.distributed import distributed_c10d as c10d

            if hasattr(c10d, "_abort_process_group"):
                c10d._abort_process_group()  # aborts NCCL/RCCL comms + pending work
        except Exception as e:  # noqa: BLE001
            logger.warning(f"slot {self.slot}: process-group abort failed ({e!r})")
        done = threading.Event()

        def _destroy():
            try:
                if dist.is_initialized():
                    dist.destroy_process_group()
            except Exception as e:  # noqa: BLE001
                logger.warning(f"slot {self.slot}: destroy after abort failed ({e!r})")
            done.set()

        t = threading.Thread(target=_destroy, daemon=True, name="pg_destroy")
        t.start()
        if not done.wait(timeout):
            logger.warning(
                f"slot {self.slot}: destroy_process_group stuck > {timeout}s "
                "after abort; abandoning it (daemon thread) and rebuilding fresh"
            )
        self.group = None

    def form_group(self, expected: Optional[int] = None, deadline: float = 90.0) -> None:
        """(Re-)form the process group over the currently alive slots.
        Every alive rank must call this (coordinator publishes, others poll)."""
        t_end = time.monotonic() + deadline
        if dist.is_initialized():
            self.abort_comm()
        while time.monotonic() < t_end and not self._stop.is_set():
            self.announce()  # our own lease may have lapsed during a stall
            alive = self._stable_alive()
            if self.slot not in alive:
                alive = sorted(set(alive + [self.slot]))
            if expected is not None and len(alive) < expected:
                time.sleep(0.25 + random.random() * 0.25)
                continue
            if alive[0] == self.slot:  # coordinator
                gen_no = self.gen + 1
                published = self._published_gen()
                if published is not None:
                    gen_no = max(gen_no, published[0] + 1)
                port = _free_port()
                self._publish_gen(gen_no, alive, port)
                target = (gen_no, alive, port)
            else:
                published = self._published_gen()
                if published is None or published[0] <= self.gen or self.slot not in published[1]:
                    time.sleep(0.25 + random.random() * 0.25)
                    continue
                target = published
            gen_no, slots, port = target
            try:
                rank = slots.index(self.slot)
                dist.init_process_group(
                    self.backend,
                    init_method=f"tcp://{self.master_addr}:{port}",
                    rank=rank, world_size=len(slots),
                    timeout=datetime.timedelta(seconds=self.timeout),
                )
                self.gen, self.members = gen_no, slots
                logger.info(f"slot {self.slot}: joined gen {gen_no} as rank {rank}/{len(slots)}")
                return
            except Exception as e:  # noqa: BLE001
                logger.warning(f"slot {self.slot}: gen {gen_no} join failed ({e!r}); retrying")
                self.abort_comm()
                # JITTERED backoff: identical retry periods phase-lock the
                # whole group into a rendezvous livelock (each member is
                # inside a doomed init exactly when the next gen appears —
                # observed under churn in experiments/elastic_train.py)
                time.sleep(0.5 + random.random() * 1.5)
        raise TimeoutError(f"slot {self.slot}: could not form a group")

    @property
    def rank(self) -> int:
        return self.members.index(self.slot)

    @property
    def world_size(self) -> int:
        return len(self.members)

    def leave(self) -> None:
        """Graceful departure: stop announcing; the lease lapses within
        `lease` seconds and survivors re-form without us."""
        self._stop.set()
        self.abort_comm()

    def shutdown(self) -> None:
        self.leave()
