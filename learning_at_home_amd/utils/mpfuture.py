"""MPFuture — a future whose result is set in one process (or thread) and
awaited in another (SURVEY.md §2 "MPFuture / SharedFuture").

Backed by a one-way multiprocessing Pipe: the *setter* half is picklable and
can travel to a worker process; the *awaiter* half stays with the creator.
Within one process it degrades to a cheap Event-based future (the common case
in this engine: one process per GPU, connection handlers are asyncio tasks /
threads, so most futures never cross a process boundary).
"""
from __future__ import annotations

import multiprocessing as mp
import threading
import time
from concurrent.futures import CancelledError, TimeoutError as FutureTimeoutError
from typing import Any, Optional, Tuple

_PENDING, _DONE, _CANCELLED = "PENDING", "DONE", "CANCELLED"


class MPFuture:
    """Cross-process future. Create with MPFuture() (in-process fast path) or
    MPFuture.make_pair() for an explicit (setter, awaiter) pair."""

    def __init__(self, _recv_conn=None, _send_conn=None, _use_pipe: bool = False):
        self._state = _PENDING
        self._result: Any = None
        self._exception: Optional[BaseException] = None
        self._event = threading.Event()
        self._use_pipe = _use_pipe
        self._recv_conn = _recv_conn
        self._send_conn = _send_conn
        if _use_pipe and _recv_conn is None and _send_conn is None:
            self._recv_conn, self._send_conn = mp.Pipe(duplex=False)

    # ---- constructors -------------------------------------------------
    @classmethod
    def make_pair(cls) -> Tuple["MPFuture", "MPFuture"]:
        """Return (setter, awaiter): setting the result on `setter` (possibly in
        another process) makes it available on `awaiter`."""
        recv_conn, send_conn = mp.Pipe(duplex=False)
        setter = cls(_send_conn=send_conn, _use_pipe=True)
        awaiter = cls(_recv_conn=recv_conn, _use_pipe=True)
        return setter, awaiter

    # ---- setter side ---------------------------------------------------
    def set_result(self, value: Any) -> None:
        if self._state != _PENDING:
            raise RuntimeError(f"future already {self._state}")
        if self._send_conn is not None:
            try:
                # serialize to bytes: torch's fd-based tensor sharing breaks if
                # the sender process exits before the receiver detaches the fd
                from .serializer import PytorchSerializer

                self._send_conn.send_bytes(PytorchSerializer.dumps(["result", value], allow_pickle=True))
            except (BrokenPipeError, OSError):
                pass  # awaiter is gone (e.g. client timed out): drop silently
        self._result = value
        self._finish(_DONE)

    def set_exception(self, exc: BaseException) -> None:
        if self._state != _PENDING:
            raise RuntimeError(f"future already {self._state}")
        if self._send_conn is not None:
            try:
                from .serializer import PytorchSerializer

                self._send_conn.send_bytes(PytorchSerializer.dumps(["exception", exc], allow_pickle=True))
            except (BrokenPipeError, OSError):
                pass
        self._exception = exc
        self._finish(_DONE)

    def _finish(self, state: str) -> None:
        self._state = state
        self._event.set()

    # ---- awaiter side --------------------------------------------------
    def _poll_pipe(self, timeout: Optional[float]) -> bool:
        assert self._recv_conn is not None
        deadline = None if timeout is None else time.monotonic() + timeout
        while True:
            remaining = None if deadline is None else max(0.0, deadline - time.monotonic())
            if self._recv_conn.poll(remaining):
                from .serializer import PytorchSerializer

                kind, payload = PytorchSerializer.loads(self._recv_conn.recv_bytes(), allow_pickle=True)
                if kind == "result":
                    self._result = payload
                else:
                    self._exception = payload
                self._finish(_DONE)
                return True
            if deadline is not None and time.monotonic() >= deadline:
                return False

    def result(self, timeout: Optional[float] = None) -> Any:
        if self._state == _CANCELLED:
            raise CancelledError()
        if self._state == _PENDING:
            if self._recv_conn is not None:
                if not self._poll_pipe(timeout):
                    raise FutureTimeoutError()
            else:
                if not self._event.wait(timeout):
                    raise FutureTimeoutError()
        if self._state == _CANCELLED:
            raise CancelledError()
        if self._exception is not None:
            raise self._exception
        return self._result

    def exception(self, timeout: Optional[float] = None) -> Optional[BaseException]:
        try:
            self.result(timeout)
        except (CancelledError, FutureTimeoutError):
            raise
        except BaseException as e:  # noqa: BLE001
            return e
        return self._exception

    def done(self) -> bool:
        if self._state == _PENDING and self._recv_conn is not None and self._recv_conn.poll(0):
            self._poll_pipe(0)
        return self._state != _PENDING

    def cancel(self) -> bool:
        if self._state != _PENDING:
            return False
        self._finish(_CANCELLED)
        return True

    def cancelled(self) -> bool:
        return self._state == _CANCELLED

    # ---- pickling: only the setter half travels ------------------------
    def __getstate__(self):
        if self._send_conn is None:
            raise RuntimeError("only the setter half of an MPFuture pair is picklable")
        return {"send_conn": self._send_conn}

    def __setstate__(self, state):
        self.__init__(_send_conn=state["send_conn"], _use_pipe=True)
