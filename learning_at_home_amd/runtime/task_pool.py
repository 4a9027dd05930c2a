"""TaskPool — per-(expert, direction) batching queue (SURVEY.md §2 "TaskPool").

MI355X-first departure from the reference: the server is ONE process per GPU
(connection handlers are asyncio tasks / threads in that process), so tasks
never cross a process boundary and the pool is a thread-safe in-process queue
— no shared-memory pickling. Batching semantics are preserved: tasks (each a
client batch) accumulate and are concatenated along dim 0 up to
`max_batch_size` samples; priority = oldest queued task; outputs are split
back per task. The intra-node GPU tensor plane bypasses pools entirely
(RCCL all-to-all → learning_at_home_amd/parallel).
"""
from __future__ import annotations

import threading
import time
from collections import deque
from typing import Callable, List, Optional, Sequence, Tuple

import torch

from ..utils import MPFuture, get_logger

logger = get_logger(__name__)


class Task:
    __slots__ = ("args", "future", "timestamp", "batch_size")

    def __init__(self, args: Sequence[torch.Tensor], future: MPFuture):
        self.args = args
        self.future = future
        self.timestamp = time.monotonic()
        self.batch_size = args[0].shape[0] if args else 1


class TaskPool:
    """Accumulates tasks for one (expert uid, direction) and forms batches."""

    def __init__(
        self,
        process_func: Callable[..., Tuple[torch.Tensor, ...]],
        name: str,
        max_batch_size: int = 16384,
        min_batch_size: int = 1,
        timeout: float = 0.0,
    ):
        self.process_func = process_func
        self.name = name
        self.max_batch_size = max_batch_size
        self.min_batch_size = min_batch_size
        self.timeout = timeout  # extra wait to accumulate min_batch_size
        self._queue: deque[Task] = deque()
        self._lock = threading.Lock()
        self._notify: Optional[Callable[["TaskPool"], None]] = None  # set by Runtime

    # ---- producer side (connection handlers) ---------------------------
    def submit_task(self, *args: torch.Tensor) -> MPFuture:
        future = MPFuture()
        task = Task(args, future)
        with self._lock:
            self._queue.append(task)
        if self._notify is not None:
            self._notify(self)
        return future

    # ---- consumer side (Runtime) ---------------------------------------
    @property
    def priority(self) -> float:
        """Timestamp of the oldest queued task (smaller = more urgent); inf if empty."""
        with self._lock:
            return self._queue[0].timestamp if self._queue else float("inf")

    def __len__(self) -> int:
        with self._lock:
            return len(self._queue)

    def ready(self) -> bool:
        """True when a batch should be formed NOW: queued samples reach
        min_batch_size, or the oldest task has waited `timeout` seconds
        (latency bound on the batching wait)."""
        with self._lock:
            if not self._queue:
                return False
            if self.min_batch_size <= 1:
                return True
            total = sum(t.batch_size for t in self._queue)
            if total >= self.min_batch_size:
                return True
            import time as _time

            return (_time.monotonic() - self._queue[0].timestamp) >= self.timeout

    def load_batch(self) -> Tuple[List[Task], List[torch.Tensor]]:
        """Pop tasks up to max_batch_size samples; concat their args along dim 0."""
        tasks: List[Task] = []
        total = 0
        with self._lock:
            while self._queue and total + self._queue[0].batch_size <= self.max_batch_size:
                task = self._queue.popleft()
                tasks.append(task)
                total += task.batch_size
            if not tasks and self._queue:  # single oversized task: take it anyway
                tasks.append(self._queue.popleft())
        live = [t for t in tasks if not t.future.cancelled()]
        if not live:
            return [], []
        if len(live) == 1:
            return live, list(live[0].args)
        batch = [torch.cat([t.args[i] for t in live], dim=0) for i in range(len(live[0].args))]
        return live, batch

    def process_batch(self) -> int:
        """Form one batch, run process_func, scatter outputs to futures.
        Returns the number of samples processed."""
        tasks, batch = self.load_batch()
        if not tasks:
            return 0
        try:
            outputs = self.process_func(*batch)
        except Exception as e:  # noqa: BLE001
            for t in tasks:
                try:
                    if not t.future.done():
                        t.future.set_exception(e)
                except RuntimeError:  # cancel() raced the done() check
                    pass
            return 0
        offset = 0
        for t in tasks:
            part = tuple(out[offset : offset + t.batch_size] for out in outputs)
            offset += t.batch_size
            # a cancel() racing between done() and set_result() must not kill
            # the Runtime thread: the client already gave up on this task
            try:
                if not t.future.done():
                    t.future.set_result(part)
            except RuntimeError:
                logger.debug(f"task in pool {self.name} finished after client cancel; dropping result")
        return offset
