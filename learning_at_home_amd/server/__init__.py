"""Server — one per rank: hosts ExpertBackends + a DHT peer + a Runtime +
a ConnectionHandler (SURVEY.md §2 "TesseractServer", §3.2 startup stack).

MI355X-first layout: the reference's {N handler processes + 1 runtime process}
collapses into ONE process per GPU (asyncio handlers + a runtime thread that
owns the device), because the heavy tensor plane is RCCL-over-xGMI between
per-GPU ranks (parallel/), not socket RPC. The discovery / liveness contract
is preserved verbatim: a background announcer re-declares all hosted experts
to the DHT every `announce_period`, with lease expiration = 2× period, so a
dead server vanishes from discovery within ~2 periods (SURVEY.md §5
"Failure detection").
"""
from __future__ import annotations

import threading
from pathlib import Path
from typing import Dict, Optional, Sequence, Tuple

import torch

from ..dht import DHT
from ..models import name_to_block
from ..runtime import ExpertBackend, Runtime
from ..utils import BatchTensorDescriptor, get_logger
from .connection_handler import ConnectionHandler
from .checkpoints import CheckpointSaver, load_experts, store_experts

logger = get_logger(__name__)


class Server:
    def __init__(
        self,
        dht: Optional[DHT],
        expert_backends: Dict[str, ExpertBackend],
        *,
        listen_on: Tuple[str, int] = ("127.0.0.1", 0),
        announce_period: float = 30.0,
        device: Optional[torch.device] = None,
        checkpoint_dir: Optional[Path] = None,
        checkpoint_period: float = 300.0,
        start: bool = False,
    ):
        self.dht = dht
        self.expert_backends = expert_backends
        self.listen_on = listen_on
        self.announce_period = announce_period
        self.conn_handler = ConnectionHandler(expert_backends, listen_on)
        self.runtime = Runtime(expert_backends, device=device)
        self.checkpoint_saver = (
            CheckpointSaver(expert_backends, checkpoint_dir, checkpoint_period)
            if checkpoint_dir is not None
            else None
        )
        self._announcer: Optional[threading.Thread] = None
        self._stop = threading.Event()
        self.ready = threading.Event()
        if start:
            self.run_in_background()

    # ------------------------------------------------------------------
    @property
    def addr(self) -> str:
        return self.listen_on[0] if self.listen_on[0] not in ("0.0.0.0", "") else "127.0.0.1"

    @property
    def port(self) -> Optional[int]:
        return self.conn_handler.port

    def run_in_background(self, await_ready: bool = True, timeout: float = 15.0) -> None:
        self.conn_handler.run_in_background()
        self.runtime.run_in_background()
        if self.dht is not None:
            self._announce_once()
            self._announcer = threading.Thread(target=self._announce_loop, daemon=True, name="announcer")
            self._announcer.start()
        if self.checkpoint_saver is not None:
            self.checkpoint_saver.start()
        self.ready.set()
        if await_ready and not self.ready.wait(timeout):
            raise TimeoutError("Server failed to start")

    def _announce_once(self) -> None:
        try:
            self.dht.declare_experts(
                list(self.expert_backends.keys()), self.addr, self.port,
                expiration=2.0 * self.announce_period,
            )
        except Exception as e:  # noqa: BLE001
            logger.warning(f"expert announce failed: {e!r}")

    def _announce_loop(self) -> None:
        import random as _random

        # jittered re-announce: many servers sharing one period otherwise
        # burst the DHT simultaneously (lease = 2x period absorbs the drift)
        while not self._stop.wait(self.announce_period * (1.0 + 0.1 * _random.random())):
            self._announce_once()

    def shutdown(self) -> None:
        self._stop.set()
        if self.checkpoint_saver is not None:
            self.checkpoint_saver.stop()
        self.conn_handler.shutdown()
        self.runtime.shutdown()

    def __enter__(self) -> "Server":
        if not self.ready.is_set():
            self.run_in_background()
        return self

    def __exit__(self, *exc) -> None:
        self.shutdown()

    # ------------------------------------------------------------------
    @classmethod
    def create(
        cls,
        num_experts: int,
        expert_cls: str = "ffn",
        hidden_dim: int = 1024,
        *,
        expert_prefix: str = "expert",
        expert_uids: Optional[Sequence[str]] = None,
        expert_offset: int = 0,
        optim_cls=torch.optim.SGD,
        optim_kwargs: Optional[dict] = None,
        max_batch_size: int = 16384,
        device: Optional[torch.device] = None,
        dtype: Optional[torch.dtype] = None,
        listen_on: Tuple[str, int] = ("127.0.0.1", 0),
        initial_peers: Sequence[Tuple[str, int]] = (),
        dht: Optional[DHT] = None,
        dht_port: int = 0,
        no_dht: bool = False,
        announce_period: float = 30.0,
        checkpoint_dir: Optional[Path] = None,
        start: bool = False,
        **expert_kwargs,
    ) -> "Server":
        """Convenience constructor mirroring the reference's Server.create
        (SURVEY.md §3.2): builds `num_experts` experts of `expert_cls` named
        `{expert_prefix}.{i}`, one optimizer each (async SGD), joins/creates
        a DHT, and optionally restores from checkpoints."""
        if dht is None and not no_dht:
            dht = DHT(listen_on=("127.0.0.1", dht_port), initial_peers=initial_peers, start=True)
        optim_kwargs = optim_kwargs if optim_kwargs is not None else {"lr": 0.01}
        block_cls = name_to_block[expert_cls]
        if expert_uids is None:
            expert_uids = [f"{expert_prefix}.{expert_offset + i}" for i in range(num_experts)]
        backends = {}
        for uid in expert_uids:
            expert = block_cls(hidden_dim, **expert_kwargs)
            if device is not None or dtype is not None:
                # dtype=bfloat16 + GPU enables the fused MFMA serving forward
                # (ExpertBackend casts at the RPC boundary, so fp32 clients
                # interoperate transparently)
                expert.to(device=device, dtype=dtype)
            optimizer = optim_cls(expert.parameters(), **optim_kwargs)
            backends[uid] = ExpertBackend(
                uid, expert, optimizer,
                args_schema=(BatchTensorDescriptor.make(hidden_dim),),
                max_batch_size=max_batch_size,
            )
        if checkpoint_dir is not None:
            n = load_experts(backends, checkpoint_dir)
            if n:
                logger.info(f"restored {n}/{len(backends)} experts from {checkpoint_dir}")
        return cls(
            dht, backends, listen_on=listen_on, announce_period=announce_period,
            device=device, checkpoint_dir=checkpoint_dir, start=start,
        )


__all__ = ["Server", "ConnectionHandler", "CheckpointSaver", "store_experts", "load_experts"]
