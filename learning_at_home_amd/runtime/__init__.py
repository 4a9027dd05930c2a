"""Runtime — the single device-owner loop (SURVEY.md §2 "TesseractRuntime").

One Runtime per GPU (or CPU) drains this rank's TaskPools in priority order
(oldest task first), moves batches to the device, runs ExpertBackend
forward/backward, and scatters outputs back to the waiting futures.

MI355X notes: the Runtime owns the device's main compute HIP stream; the
DecentralizedAverager runs on a side stream (averaging/), and the RCCL
dispatch plane (parallel/) bypasses pools entirely for intra-node traffic.
"""
from __future__ import annotations

import threading
from typing import Dict, Optional

import torch

from ..utils import get_logger
from .expert_backend import ExpertBackend
from .task_pool import TaskPool

logger = get_logger(__name__)


class Runtime:
    def __init__(
        self,
        expert_backends: Dict[str, ExpertBackend],
        device: Optional[torch.device] = None,
        stats_report_interval: float = 0.0,
    ):
        self.expert_backends = expert_backends
        self.device = torch.device(device) if device is not None else None
        self.pools = []
        self._cond = threading.Condition()
        self._shutdown = False
        self._thread: Optional[threading.Thread] = None
        self.samples_processed = 0
        for backend in expert_backends.values():
            if self.device is not None:
                backend.expert.to(self.device)
            for pool in backend.get_pools():
                pool._notify = self._on_task
                self.pools.append(pool)

    def _on_task(self, pool: TaskPool) -> None:
        with self._cond:
            self._cond.notify()

    # ------------------------------------------------------------------
    def run(self) -> None:
        """Hot loop: pick the pool with the oldest pending task and process
        one batch. Exits when shutdown() is called."""
        logger.debug(f"Runtime started with {len(self.pools)} pools on {self.device}")
        while True:
            with self._cond:
                # a pool is eligible once ready(): min_batch_size reached or
                # its oldest task aged past the pool's batching timeout
                while not self._shutdown and not any(p.ready() for p in self.pools):
                    self._cond.wait(timeout=0.05)
                if self._shutdown:
                    break
            eligible = [p for p in self.pools if p.ready()]
            if not eligible:
                continue
            pool = min(eligible, key=lambda p: p.priority)
            if self.device is not None and self.device.type != "cpu":
                processed = self._process_on_device(pool)
            else:
                processed = pool.process_batch()
            self.samples_processed += processed

    def _process_on_device(self, pool: TaskPool) -> int:
        """Same as pool.process_batch but moves tensors host→device→host."""
        tasks, batch = pool.load_batch()
        if not tasks:
            return 0
        try:
            device_batch = [t.to(self.device, non_blocking=True) for t in batch]
            outputs = pool.process_func(*device_batch)
            outputs = tuple(o.to("cpu") for o in outputs)
        except Exception as e:  # noqa: BLE001
            for t in tasks:
                if not t.future.done():
                    t.future.set_exception(e)
            return 0
        offset = 0
        for t in tasks:
            part = tuple(out[offset : offset + t.batch_size] for out in outputs)
            offset += t.batch_size
            if not t.future.done():
                t.future.set_result(part)
        return offset

    def run_in_background(self) -> None:
        self._thread = threading.Thread(target=self.run, name="Runtime", daemon=True)
        self._thread.start()

    def shutdown(self) -> None:
        with self._cond:
            self._shutdown = True
            self._cond.notify_all()
        if self._thread is not None:
            self._thread.join(timeout=5.0)


__all__ = ["Runtime", "ExpertBackend", "TaskPool"]
