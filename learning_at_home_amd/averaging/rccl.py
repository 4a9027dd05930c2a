"""RcclAverager — the MI355X tensor-plane averager (SURVEY.md §5 "Distributed
communication backend"): butterfly group all-reduce ≡ RCCL reduce-scatter +
all-gather over xGMI, launched on a SIDE HIP stream so averaging overlaps
the next step's compute (the reference overlaps averaging with training via
a background process; here the overlap is stream-level).

xGMI notes (MI355X: 7 p2p links × ≈153 GB/s per GPU): for ≤8-rank intra-node
groups reduce-scatter+all-gather moves 2·(M-1)/M of the payload per GPU; we
keep ONE flat fp32 (or bf16) fusion buffer so RCCL sees a single large
contiguous collective (per-link bound, no bucket fragmentation needed below
~1 GB payloads).
"""
from __future__ import annotations

from typing import Optional, Sequence

import torch
import torch.distributed as dist

from ..utils import get_logger

logger = get_logger(__name__)


class RcclAverager:
    def __init__(
        self,
        tensors: Sequence[torch.Tensor],
        *,
        process_group: Optional[dist.ProcessGroup] = None,
        dtype: torch.dtype = torch.float32,
    ):
        assert dist.is_initialized(), "torch.distributed must be initialized (RCCL backend)"
        self.tensors = list(tensors)
        self.group = process_group
        self.world_size = dist.get_world_size(process_group)
        self.device = self.tensors[0].device
        self.dtype = dtype
        numel = sum(t.numel() for t in self.tensors)
        # pad so the flat buffer splits evenly into world_size shards
        self.padded = ((numel + self.world_size - 1) // self.world_size) * self.world_size
        self.flat = torch.zeros(self.padded, dtype=dtype, device=self.device)
        self.shard = torch.zeros(self.padded // self.world_size, dtype=dtype, device=self.device)
        self.use_cuda = self.device.type == "cuda"
        self.stream = torch.cuda.Stream(device=self.device) if self.use_cuda else None
        self._done_event = torch.cuda.Event() if self.use_cuda else None
        self.num_rounds = 0

    @torch.no_grad()
    def _pack(self) -> None:
        offset = 0
        for t in self.tensors:
            n = t.numel()
            self.flat[offset : offset + n].copy_(t.reshape(-1).to(self.dtype))
            offset += n

    @torch.no_grad()
    def _unpack(self) -> None:
        offset = 0
        for t in self.tensors:
            n = t.numel()
            t.reshape(-1).copy_(self.flat[offset : offset + n].to(t.dtype))
            offset += n

    @torch.no_grad()
    def step(self, async_op: bool = True) -> "RcclAverager":
        """Launch one averaging round on the side stream. With async_op, call
        wait() before the next optimizer step touches the tensors."""
        if self.world_size == 1:
            return self
        if self.use_cuda:
            # side stream waits for all pending work on the current stream
            self.stream.wait_stream(torch.cuda.current_stream(self.device))
            with torch.cuda.stream(self.stream):
                self._run_round()
            self._done_event.record(self.stream)
            if not async_op:
                self.wait()
        else:
            self._run_round()
        self.num_rounds += 1
        return self

    def _run_round(self) -> None:
        self._pack()
        backend = dist.get_backend(self.group)
        if backend == "nccl":  # RCCL on ROCm: butterfly ≡ reduce-scatter + all-gather
            dist.reduce_scatter_tensor(self.shard, self.flat, op=dist.ReduceOp.AVG, group=self.group)
            dist.all_gather_into_tensor(self.flat, self.shard, group=self.group)
        else:  # gloo (CPU tests): reduce_scatter/AVG unsupported there
            dist.all_reduce(self.flat, op=dist.ReduceOp.SUM, group=self.group)
            self.flat.div_(self.world_size)
        self._unpack()

    def wait(self) -> None:
        """Make the CURRENT stream wait until the averaged values are written."""
        if self.use_cuda and self._done_event is not None:
            torch.cuda.current_stream(self.device).wait_event(self._done_event)

    @torch.no_grad()
    def rebind(self, process_group: Optional[dist.ProcessGroup] = None) -> None:
        """Adopt a REBUILT process group after an elastic membership change
        (parallel/elastic.py generation rebuild): world size, padding and
        shard sizing are re-derived for the new group. Callers invoke this
        between rounds, after ElasticGroup.form_group() — any in-flight
        round on the old communicator was already torn down by abort_comm,
        and wait() here fences the side stream before buffers are resized."""
        self.wait()
        assert dist.is_initialized(), "rebind requires an initialized process group"
        self.group = process_group
        self.world_size = dist.get_world_size(process_group)
        numel = sum(t.numel() for t in self.tensors)
        padded = ((numel + self.world_size - 1) // self.world_size) * self.world_size
        if padded != self.padded or self.flat.numel() != padded:
            self.padded = padded
            self.flat = torch.zeros(padded, dtype=self.dtype, device=self.device)
        shard_n = padded // self.world_size
        if self.shard.numel() != shard_n:
            self.shard = torch.zeros(shard_n, dtype=self.dtype, device=self.device)
