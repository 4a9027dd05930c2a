"""ConnectionHandler — TCP request loop for fwd/bwd/info expert RPC
(SURVEY.md §2 "ConnectionHandler"). Cross-host / CPU-parity transport only:
intra-node GPU traffic rides RCCL all-to-all (parallel/), never this path.

One asyncio TCP server on a dedicated thread serves all experts of this
server process. Messages (utils/connection.py framing):
  "fwd"  payload = serialized [uid, [tensors...]]          -> "result" [outputs...]
  "bwd"  payload = serialized [uid, [inputs+grad_outputs]] -> "result" [grad_inputs...]
  "info" payload = serialized uid                          -> "result" info dict
  errors -> "error" with a message string.
"""
from __future__ import annotations

import asyncio
import threading
from concurrent.futures import ThreadPoolExecutor
from typing import Dict, Optional, Tuple

from ..runtime import ExpertBackend
from ..utils import PytorchSerializer, aread_message, awrite_message, get_logger

logger = get_logger(__name__)


class ConnectionHandler:
    def __init__(self, expert_backends: Dict[str, ExpertBackend],
                 listen_on: Tuple[str, int] = ("127.0.0.1", 0), num_waiters: int = 64):
        self.expert_backends = expert_backends
        self.listen_on = listen_on
        self.port: Optional[int] = None
        self._loop: Optional[asyncio.AbstractEventLoop] = None
        self._thread: Optional[threading.Thread] = None
        self._server: Optional[asyncio.AbstractServer] = None
        self._waiters = ThreadPoolExecutor(max_workers=num_waiters, thread_name_prefix="rpc_waiter")
        self.ready = threading.Event()

    # ------------------------------------------------------------------
    def run_in_background(self, timeout: float = 10.0) -> None:
        self._thread = threading.Thread(target=self._run, name="ConnectionHandler", daemon=True)
        self._thread.start()
        if not self.ready.wait(timeout):
            raise TimeoutError("ConnectionHandler failed to start")

    def _run(self) -> None:
        self._loop = asyncio.new_event_loop()
        asyncio.set_event_loop(self._loop)
        self._loop.run_until_complete(self._start_server())
        self.ready.set()
        try:
            self._loop.run_forever()
        finally:
            self._loop.close()

    async def _start_server(self) -> None:
        self._server = await asyncio.start_server(self._handle_client, *self.listen_on)
        self.port = self._server.sockets[0].getsockname()[1]

    def shutdown(self) -> None:
        if self._loop is not None and self._loop.is_running():
            def _stop() -> None:
                if self._server is not None:
                    self._server.close()
                for task in asyncio.all_tasks(self._loop):
                    task.cancel()
                self._loop.call_soon(self._loop.stop)

            self._loop.call_soon_threadsafe(_stop)
        if self._thread is not None:
            self._thread.join(timeout=5.0)
        self._waiters.shutdown(wait=False)

    # ------------------------------------------------------------------
    async def _handle_client(self, reader: asyncio.StreamReader, writer: asyncio.StreamWriter) -> None:
        try:
            while True:
                try:
                    kind, payload = await aread_message(reader)
                except (asyncio.IncompleteReadError, ConnectionError):
                    break
                # one request at a time per connection: the frame has no
                # request ids, so replies must come back in request order even
                # for a client that pipelines. Concurrency comes from separate
                # connections (client/_SocketPool opens one per in-flight call),
                # each served by its own asyncio task.
                await self._handle_request(kind, payload, writer)
        finally:
            try:
                writer.close()
            except Exception:  # noqa: BLE001
                pass

    async def _handle_request(self, kind: str, payload: bytes, writer: asyncio.StreamWriter) -> None:
        loop = asyncio.get_event_loop()
        try:
            if kind == "info":
                uid = PytorchSerializer.loads(payload)
                backend = self._get_backend(uid)
                response = PytorchSerializer.dumps(backend.get_info())
            elif kind in ("fwd", "bwd"):
                uid, tensors = PytorchSerializer.loads(payload)
                backend = self._get_backend(uid)
                pool = backend.forward_pool if kind == "fwd" else backend.backward_pool
                future = pool.submit_task(*tensors)
                # MPFuture.result() blocks: park it on the waiter thread pool
                outputs = await loop.run_in_executor(self._waiters, future.result, 60.0)
                response = PytorchSerializer.dumps(list(outputs))
            else:
                raise ValueError(f"unknown message kind {kind!r}")
            await awrite_message(writer, "result", response)
        except Exception as e:  # noqa: BLE001
            logger.debug(f"request {kind} failed: {e!r}")
            try:
                await awrite_message(writer, "error", repr(e).encode())
            except Exception:  # noqa: BLE001
                pass

    def _get_backend(self, uid: str) -> ExpertBackend:
        backend = self.expert_backends.get(uid)
        if backend is None:
            raise KeyError(f"expert {uid!r} is not hosted here")
        return backend
