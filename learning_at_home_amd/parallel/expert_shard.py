"""Expert-parallel MoE layer — the MI355X tensor plane (SURVEY.md §2
"TaskPool/ConnectionHandler MI355X equivalent", §5 "Distributed communication
backend"): token dispatch and gradient return ride RCCL all-to-all over xGMI
between one-process-per-GPU ranks; each rank owns an ExpertStack shard whose
FFNs run through the hand-written CDNA4 kernels (ops/).

Fault-tolerance semantics preserved from the reference: slots over a fixed
per-expert CAPACITY are dropped and the combine renormalizes the gating
softmax over the surviving slots (= "drop non-responders, renormalize"),
asynchronous SGD per expert shard (no global optimizer barrier).

Memory layout (288 GB HBM3E per GPU): per-layer stacked weights
W1 [E_local, ffn, d] / W2 [E_local, d, ffn] as fp32 masters + bf16 compute
copies + bf16 transposed copies (so every activation GEMM is MFMA-natural
NT — see ops/hip/ffn_gemm.hip) + fp32 main_grad buffers consumed by the
fused SGD kernel.
"""
from __future__ import annotations

import math
from typing import Optional, Tuple

import torch
import torch.distributed as dist
import torch.nn as nn

from ..ops import HAS_HIP, _C
from ..ops import functional as Fops
from ..utils import get_logger

logger = get_logger(__name__)


class ExpertStack(nn.Module):
    """E_local experts' FFN weights, stacked for grouped kernels."""

    def __init__(self, num_local: int, d_model: int, ffn_dim: int):
        super().__init__()
        self.num_local, self.d_model, self.ffn_dim = num_local, d_model, ffn_dim
        # fp32 masters (the checkpointable parameters; nn.Linear-style init)
        bound1 = 1.0 / math.sqrt(d_model)
        bound2 = 1.0 / math.sqrt(ffn_dim)
        self.W1 = nn.Parameter(torch.empty(num_local, ffn_dim, d_model).uniform_(-bound1, bound1))
        self.W1_bias = nn.Parameter(torch.empty(num_local, ffn_dim).uniform_(-bound1, bound1))
        self.W2 = nn.Parameter(torch.empty(num_local, d_model, ffn_dim).uniform_(-bound2, bound2))
        self.W2_bias = nn.Parameter(torch.empty(num_local, d_model).uniform_(-bound2, bound2))
        self._gpu_state_ready = False
        # grad_ready is tracked per expert RANGE so the chunked-a2a pipeline
        # can run the backward of each chunk's slice independently: chunk 2's
        # first backward of a step must WRITE its fp32 grad slice (not
        # accumulate into last step's values just because chunk 1's backward
        # already ran). Ranges are half-open [start, end) over local experts.
        self._grad_ready_ranges: set = set()

    @property
    def grad_ready(self) -> bool:
        """True iff every local expert's grad slice was written this step."""
        return sum(e - s for s, e in self._grad_ready_ranges) >= self.num_local

    @grad_ready.setter
    def grad_ready(self, value: bool) -> None:
        self._grad_ready_ranges = {(0, self.num_local)} if value else set()

    def range_grad_ready(self, start: int, end: int) -> bool:
        """Has [start, end) been written this step? (ranges are disjoint and
        chunk boundaries are fixed within a step, so exact/whole matching
        suffices)"""
        if (0, self.num_local) in self._grad_ready_ranges:
            return True
        return (start, end) in self._grad_ready_ranges

    def mark_range_grad_ready(self, start: int, end: int) -> None:
        self._grad_ready_ranges.add((start, end))

    def slice_view(self, start: int, end: int) -> "_ExpertStackSlice":
        """A zero-copy view of experts [start, end) sharing every buffer —
        the unit the chunked all-to-all pipeline feeds to batched_ffn."""
        return _ExpertStackSlice(self, start, end)

    # ---- GPU compute-state management ---------------------------------
    def init_gpu_state(self, master_weights: bool = True) -> None:
        """Allocate bf16 compute/transposed copies + fp32 grad buffers.
        master_weights=False → stochastic-rounding mode: the fp32 masters of
        W1/W2 move to host (updates happen directly on the bf16 copies with
        unbiased rounding — see optim.hip sgd_step_sr_t); call
        sync_master_weights() before checkpointing in that mode."""
        assert self.W1.is_cuda, "move the stack to GPU first"
        dev = self.W1.device
        self.master_weights = master_weights
        self.W1c = self.W1.detach().bfloat16().contiguous()
        self.W2c = self.W2.detach().bfloat16().contiguous()
        self.W1t = _C.transpose_bf16(self.W1c)  # [E, d, ffn]
        self.W2t = _C.transpose_bf16(self.W2c)  # [E, ffn, d]
        self.b1_f32 = self.W1_bias.detach().float().contiguous()
        self.b2_f32 = self.W2_bias.detach().float().contiguous()
        if not master_weights:
            self.W1.data = self.W1.data.cpu()  # free 2x HBM for weights
            self.W2.data = self.W2.data.cpu()
        self.main_grad_W1 = torch.zeros(self.W1.shape, device=dev)
        self.main_grad_W2 = torch.zeros(self.W2.shape, device=dev)
        self.main_grad_b1 = torch.zeros_like(self.W1_bias, device=dev)
        self.main_grad_b2 = torch.zeros_like(self.W2_bias, device=dev)
        self._gpu_state_ready = True

    # ---- per-expert checkpoint interop (server/checkpoints.py format) ---
    def export_expert_state(self, idx: int) -> dict:
        """FFNExpert-compatible state_dict of local expert `idx` — the same
        on-disk layout the socket Server writes, so EP shards and RPC servers
        can exchange checkpoints (SURVEY.md §5 'byte-compatible')."""
        self.sync_master_weights()
        return {
            "fc1.weight": self.W1[idx].detach().cpu().clone(),
            "fc1.bias": self.W1_bias[idx].detach().cpu().clone(),
            "fc2.weight": self.W2[idx].detach().cpu().clone(),
            "fc2.bias": self.W2_bias[idx].detach().cpu().clone(),
        }

    @torch.no_grad()
    def import_expert_state(self, idx: int, sd: dict) -> None:
        self.W1.data[idx].copy_(sd["fc1.weight"].to(self.W1.device))
        self.W1_bias.data[idx].copy_(sd["fc1.bias"].to(self.W1_bias.device))
        self.W2.data[idx].copy_(sd["fc2.weight"].to(self.W2.device))
        self.W2_bias.data[idx].copy_(sd["fc2.bias"].to(self.W2_bias.device))
        if self._gpu_state_ready:
            from ..ops import _C

            self.W1c[idx].copy_(self.W1.data[idx].bfloat16().to(self.W1c.device))
            self.W2c[idx].copy_(self.W2.data[idx].bfloat16().to(self.W2c.device))
            self.W1t[idx].copy_(self.W1c[idx].t().contiguous())
            self.W2t[idx].copy_(self.W2c[idx].t().contiguous())
            self.b1_f32[idx].copy_(self.W1_bias.data[idx].to(self.b1_f32.device))
            self.b2_f32[idx].copy_(self.W2_bias.data[idx].to(self.b2_f32.device))

    def save_checkpoints(self, checkpoint_dir, uid_prefix: str, first_global_idx: int = 0,
                         update_count: int = 0, optimizer=None) -> None:
        """Per-uid checkpoint files in the server/checkpoints.py on-disk
        format. Pass the shard's optimizer (FusedExpertSGD / ExpertAdam) to
        persist momentum / Adam moments / the SR step counter — a restart
        is then a true RESUME, not just a weight reload."""
        import time as _time
        from pathlib import Path

        ts = _time.strftime("%Y%m%d_%H%M%S")
        for i in range(self.num_local):
            uid = f"{uid_prefix}.{first_global_idx + i}"
            d = Path(checkpoint_dir) / uid
            d.mkdir(parents=True, exist_ok=True)
            opt_state = optimizer.state_for_expert(i) if optimizer is not None else {}
            blob = {"model": self.export_expert_state(i), "optimizer": opt_state,
                    "update_count": update_count}
            tmp = d / f".tmp_{ts}.pt"
            torch.save(blob, tmp)
            import os as _os

            _os.replace(tmp, d / f"checkpoint_{ts}.pt")
            import shutil as _shutil

            _shutil.copyfile(d / f"checkpoint_{ts}.pt", d / "checkpoint_last.pt")

    def load_checkpoints(self, checkpoint_dir, uid_prefix: str, first_global_idx: int = 0,
                         optimizer=None) -> int:
        from pathlib import Path

        n = 0
        for i in range(self.num_local):
            f = Path(checkpoint_dir) / f"{uid_prefix}.{first_global_idx + i}" / "checkpoint_last.pt"
            if f.exists():
                blob = torch.load(f, map_location="cpu", weights_only=False)
                self.import_expert_state(i, blob["model"])
                if optimizer is not None and blob.get("optimizer"):
                    optimizer.load_expert_state(i, blob["optimizer"])
                n += 1
        return n

    def sync_master_weights(self) -> None:
        """Pull bf16 compute weights back into the fp32 master Parameters
        (needed before state_dict/checkpoint in stochastic-rounding mode)."""
        if self._gpu_state_ready and not self.master_weights:
            self.W1.data = self.W1c.detach().float().to(self.W1.data.device)
            self.W2.data = self.W2c.detach().float().to(self.W2.data.device)

    def zero_main_grads(self) -> None:
        # the TN grad kernels WRITE (not accumulate) on the first backward of
        # a step (functional._BatchedFFNHip), so no fp32 memset is needed —
        # clearing the flag re-arms write-mode
        self.grad_ready = False


class _ExpertStackSlice:
    """Duck-typed view of an ExpertStack restricted to local experts
    [start, end): every tensor attribute batched_ffn touches is a contiguous
    dim-0 slice sharing storage with the parent, and grad_ready reads/writes
    the parent's per-range tracking. Created per forward chunk; holds no
    state of its own."""

    _SLICED = ("W1", "W2", "W1_bias", "W2_bias", "W1c", "W2c", "W1t", "W2t",
               "b1_f32", "b2_f32", "main_grad_W1", "main_grad_W2",
               "main_grad_b1", "main_grad_b2")

    def __init__(self, parent: ExpertStack, start: int, end: int):
        assert 0 <= start < end <= parent.num_local
        self._parent = parent
        self._range = (start, end)
        self.num_local = end - start
        self.d_model = parent.d_model
        self.ffn_dim = parent.ffn_dim

    def __getattr__(self, name: str):
        if name in self._SLICED:
            base = getattr(self._parent, name)
            return base[self._range[0] : self._range[1]]
        return getattr(self._parent, name)

    @property
    def grad_ready(self) -> bool:
        return self._parent.range_grad_ready(*self._range)

    @grad_ready.setter
    def grad_ready(self, value: bool) -> None:
        if value:
            self._parent.mark_range_grad_ready(*self._range)
        else:
            self._parent._grad_ready_ranges.discard(self._range)


class FusedExpertSGD:
    """Asynchronous per-shard SGD: one fused kernel pass per tensor updates
    the fp32 master AND re-quantizes the bf16 compute copy; transposed copies
    refreshed by the tiled transpose kernel. (SURVEY.md §2 hot-op #5.)"""

    def __init__(self, stack: ExpertStack, lr: float = 0.01, momentum: float = 0.0,
                 weight_decay: float = 0.0, stochastic_rounding: bool = False,
                 fused_backward: Optional[bool] = None):
        self.stack = stack
        self.lr, self.momentum, self.weight_decay = lr, momentum, weight_decay
        self.stochastic_rounding = stochastic_rounding
        if stochastic_rounding:
            assert momentum == 0.0, "SR mode supports plain SGD only"
            assert not getattr(stack, "master_weights", True), (
                "init_gpu_state(master_weights=False) first for SR mode")
        # fused dW+SGD: the backward's TN epilogue applies the SR update to
        # W/W^T directly (no fp32 grad buffer; ~14 -> 6 B/elem of HBM per
        # weight per step). SR mode + 256-aligned dims + one backward/step.
        # MEASURED REGIME (profiles/fused_dw_r02.md): wins at small
        # per-expert batches (M<=~512: 2.82 vs 3.28 ms at M=192) where the
        # pair is write-bound, LOSES at the flagship's M=2304 (12.5 vs ~9.7
        # ms: the GEMM turns compute-bound and the in-kernel epilogue is
        # slower than the streaming SR kernel) -> default OFF; opt in with
        # LAH_FUSED_DW=1 for small-batch / high-expert-count regimes.
        if fused_backward is None:
            import os as _os

            fused_backward = (stochastic_rounding
                              and stack.d_model % 256 == 0 and stack.ffn_dim % 256 == 0
                              and _os.environ.get("LAH_FUSED_DW", "0") == "1")
        self.fused_backward = bool(fused_backward)
        if self.fused_backward:
            assert stochastic_rounding, "fused_backward requires stochastic_rounding"
            assert stack.d_model % 256 == 0 and stack.ffn_dim % 256 == 0, (
                "fused dW+SGD needs d_model/ffn_dim multiples of 256")
            stack._fused_sr_opt = self
            if stack._gpu_state_ready:
                # the fp32 W grad buffers are never written in fused mode:
                # release them (flagship: ~8.6 GB of HBM per layer)
                dev = stack.main_grad_W1.device
                stack.main_grad_W1 = torch.empty(0, device=dev)
                stack.main_grad_W2 = torch.empty(0, device=dev)
        self.momentum_bufs = None
        if momentum > 0:
            self.momentum_bufs = [
                torch.zeros_like(stack.W1), torch.zeros_like(stack.W2),
                torch.zeros_like(stack.W1_bias), torch.zeros_like(stack.W2_bias),
            ]
        self.step_count = 0

    def sr_seed(self) -> int:
        return (self.step_count * 2654435761) & 0x7FFFFFFF

    def set_side_stream(self, stream: "torch.cuda.Stream") -> None:
        """Overlap the optimizer pass with the NEXT step's forward: step()
        launches its kernels on `stream` and records a completion event on
        the stack; batched_ffn's forward/backward wait that event before
        touching the weights/grads (layer L's update hides under the next
        step's pre-L compute — ~23 ms of SR traffic at the flagship)."""
        self._side_stream = stream

    @torch.no_grad()
    def step(self) -> None:
        side = getattr(self, "_side_stream", None)
        if side is not None and self.stack.grad_ready:
            side.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(side):
                self._step_body()
            ev = torch.cuda.Event()
            ev.record(side)
            self.stack._opt_done_event = ev
            return
        self._step_body()

    @torch.no_grad()
    def _step_body(self) -> None:
        st = self.stack
        if not st.grad_ready:
            return
        if self.fused_backward:
            # W1/W2 were updated inside the backward's TN epilogue; only the
            # bias updates (tiny tensors, colsum grads) remain
            pass
        elif self.stochastic_rounding:
            seed = self.sr_seed()
            _C.sgd_step_sr_t(st.W1c, st.main_grad_W1, st.W1t, self.lr,
                             self.weight_decay, seed)
            _C.sgd_step_sr_t(st.W2c, st.main_grad_W2, st.W2t, self.lr,
                             self.weight_decay, seed ^ 0x55555555)
        elif self.momentum == 0.0:
            # fused path: one pass updates master + bf16 copy + bf16^T copy
            _C.sgd_step_t(st.W1.data, st.main_grad_W1, st.W1c, st.W1t,
                          self.lr, self.weight_decay)
            _C.sgd_step_t(st.W2.data, st.main_grad_W2, st.W2c, st.W2t,
                          self.lr, self.weight_decay)
        else:
            mom = self.momentum_bufs
            _C.sgd_step(st.W1.data.view(-1), st.main_grad_W1.view(-1), mom[0].view(-1),
                        st.W1c.view(-1), self.lr, self.momentum, self.weight_decay)
            _C.sgd_step(st.W2.data.view(-1), st.main_grad_W2.view(-1), mom[1].view(-1),
                        st.W2c.view(-1), self.lr, self.momentum, self.weight_decay)
            st.W1t.copy_(_C.transpose_bf16(st.W1c))
            st.W2t.copy_(_C.transpose_bf16(st.W2c))
        mom = self.momentum_bufs or [None] * 4
        scratch = torch.empty_like(st.W1_bias, dtype=torch.bfloat16)
        _C.sgd_step(st.W1_bias.data.view(-1), st.main_grad_b1.view(-1),
                    mom[2].view(-1) if mom[2] is not None else None,
                    scratch.view(-1), self.lr, self.momentum, self.weight_decay)
        scratch2 = torch.empty_like(st.W2_bias, dtype=torch.bfloat16)
        _C.sgd_step(st.W2_bias.data.view(-1), st.main_grad_b2.view(-1),
                    mom[3].view(-1) if mom[3] is not None else None,
                    scratch2.view(-1), self.lr, self.momentum, self.weight_decay)
        st.b1_f32.copy_(st.W1_bias.data)
        st.b2_f32.copy_(st.W2_bias.data)
        st.zero_main_grads()
        self.step_count += 1

    # ---- per-expert optimizer checkpoint slices ------------------------
    def state_for_expert(self, idx: int) -> dict:
        """Momentum slice + step counter of local expert `idx` (the SR seed
        is derived from step_count, so persisting it keeps the stochastic-
        rounding stream deterministic across a restart)."""
        sd = {"type": "fused_sgd", "step_count": self.step_count}
        if self.momentum_bufs is not None:
            names = ("W1", "W2", "b1", "b2")
            sd["momentum"] = {n: buf[idx].detach().cpu().clone()
                              for n, buf in zip(names, self.momentum_bufs)}
        return sd

    @torch.no_grad()
    def load_expert_state(self, idx: int, sd: dict) -> None:
        self.step_count = max(self.step_count, int(sd.get("step_count", 0)))
        if self.momentum_bufs is not None and sd.get("momentum"):
            names = ("W1", "W2", "b1", "b2")
            for n, buf in zip(names, self.momentum_bufs):
                if n in sd["momentum"]:
                    buf[idx].copy_(sd["momentum"][n].to(buf.device))


@torch.no_grad()
def adam_update_(p: torch.Tensor, g: torch.Tensor, m: torch.Tensor, v: torch.Tensor,
                 step: int, lr: float, beta1: float, beta2: float, eps: float,
                 weight_decay: float) -> None:
    """One fp32 Adam update, in place — torch.optim.Adam semantics
    (decoupled=False: weight decay added to the gradient). Factored out so
    the math is unit-tested against torch.optim.Adam on CPU."""
    if weight_decay != 0.0:
        g = g.add(p, alpha=weight_decay)
    m.mul_(beta1).add_(g, alpha=1.0 - beta1)
    v.mul_(beta2).addcmul_(g, g, value=1.0 - beta2)
    bc1 = 1.0 - beta1 ** step
    bc2 = 1.0 - beta2 ** step
    denom = (v / bc2).sqrt_().add_(eps)
    p.addcdiv_(m, denom, value=-lr / bc1)


class ExpertAdam:
    """Asynchronous per-shard Adam over an ExpertStack (the optimizer-choice
    counterpart of FusedExpertSGD; SURVEY.md §2 hot-op #5 names "SGD/Adam").
    Torch-level (hipBLASLt-free elementwise ops) on the fp32 masters, then
    re-quantizes the bf16 compute copies and their transposes — the default
    remains FusedExpertSGD, which fuses all of that into one kernel pass.
    GPU mode requires init_gpu_state(master_weights=True); on CPU (gloo
    tests, resume tests) it reads autograd .grad of the masters instead of
    the main_grad buffers and skips the compute-copy refresh."""

    def __init__(self, stack: ExpertStack, lr: float = 1e-3,
                 betas: Tuple[float, float] = (0.9, 0.999), eps: float = 1e-8,
                 weight_decay: float = 0.0):
        assert getattr(stack, "master_weights", True), (
            "ExpertAdam needs fp32 masters: init_gpu_state(master_weights=True)")
        self.stack = stack
        self.lr, self.betas, self.eps, self.weight_decay = lr, betas, eps, weight_decay
        self.step_count = 0
        self._state = {}  # name -> (m, v)

    def _mv(self, name: str, like: torch.Tensor):
        if name not in self._state:
            self._state[name] = (torch.zeros_like(like), torch.zeros_like(like))
        return self._state[name]

    def _named_params_and_grads(self):
        st = self.stack
        if st._gpu_state_ready:
            return (("W1", st.W1.data, st.main_grad_W1),
                    ("W2", st.W2.data, st.main_grad_W2),
                    ("b1", st.W1_bias.data, st.main_grad_b1),
                    ("b2", st.W2_bias.data, st.main_grad_b2))
        return (("W1", st.W1.data, st.W1.grad),
                ("W2", st.W2.data, st.W2.grad),
                ("b1", st.W1_bias.data, st.W1_bias.grad),
                ("b2", st.W2_bias.data, st.W2_bias.grad))

    @torch.no_grad()
    def step(self) -> None:
        st = self.stack
        if st._gpu_state_ready and not st.grad_ready:
            return
        named = self._named_params_and_grads()
        if any(g is None for _, _, g in named):
            return
        self.step_count += 1
        b1, b2 = self.betas
        for name, p, g in named:
            m, v = self._mv(name, p)
            adam_update_(p, g.float(), m, v, self.step_count, self.lr, b1, b2,
                         self.eps, self.weight_decay)
        if st._gpu_state_ready:
            # refresh bf16 compute copies + transposes + fp32 bias mirrors
            st.W1c.copy_(st.W1.data.bfloat16())
            st.W2c.copy_(st.W2.data.bfloat16())
            if st.W1c.is_cuda and HAS_HIP:
                st.W1t.copy_(_C.transpose_bf16(st.W1c))
                st.W2t.copy_(_C.transpose_bf16(st.W2c))
            else:
                st.W1t.copy_(st.W1c.transpose(1, 2).contiguous())
                st.W2t.copy_(st.W2c.transpose(1, 2).contiguous())
            st.b1_f32.copy_(st.W1_bias.data)
            st.b2_f32.copy_(st.W2_bias.data)
            st.zero_main_grads()
        else:
            for _, p_named in (("W1", st.W1), ("W2", st.W2),
                               ("b1", st.W1_bias), ("b2", st.W2_bias)):
                p_named.grad = None

    # ---- per-expert optimizer checkpoint slices ------------------------
    def state_for_expert(self, idx: int) -> dict:
        sd = {"type": "adam", "step_count": self.step_count, "state": {}}
        for name, (m, v) in self._state.items():
            sd["state"][name] = {"m": m[idx].detach().cpu().clone(),
                                 "v": v[idx].detach().cpu().clone()}
        return sd

    @torch.no_grad()
    def load_expert_state(self, idx: int, sd: dict) -> None:
        self.step_count = max(self.step_count, int(sd.get("step_count", 0)))
        likes = {"W1": self.stack.W1.data, "W2": self.stack.W2.data,
                 "b1": self.stack.W1_bias.data, "b2": self.stack.W2_bias.data}
        for name, mv in sd.get("state", {}).items():
            m, v = self._mv(name, likes[name])
            m[idx].copy_(mv["m"].to(m.device))
            v[idx].copy_(mv["v"].to(v.device))


class _AllToAll(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x: torch.Tensor, group):
        ctx.group = group
        out = torch.empty_like(x)
        dist.all_to_all_single(out, x.contiguous(), group=group)
        return out

    @staticmethod
    def backward(ctx, grad: torch.Tensor):
        out = torch.empty_like(grad)
        dist.all_to_all_single(out, grad.contiguous(), group=ctx.group)
        return out, None


class ExpertParallelMoE(nn.Module):
    """Drop-in MoE FFN layer: [.., D] -> [.., D], experts sharded over the
    process group (E % world_size == 0), RCCL all-to-all dispatch/return."""

    def __init__(
        self,
        d_model: int,
        ffn_dim: int,
        num_experts: int,
        k: int = 4,
        capacity_factor: float = 1.25,
        aux_loss_coef: float = 0.01,
        router_noise_std: float = 0.3,
        z_loss_coef: float = 1e-3,
        group: Optional[object] = None,
        tokens_per_batch_hint: int = 8192,
        a2a_chunks: int = 1,
    ):
        super().__init__()
        self.d_model, self.ffn_dim = d_model, ffn_dim
        self.num_experts, self.k = num_experts, k
        self.capacity_factor = capacity_factor
        self.aux_loss_coef = aux_loss_coef
        self.router_noise_std = router_noise_std
        self.z_loss_coef = z_loss_coef
        self.group = group
        # >1 splits dispatch/return all-to-all into expert-range chunks and
        # pipelines each chunk's collective (side HIP stream) against the
        # previous chunk's expert FFN — hides a2a latency at N>1 over xGMI
        self.a2a_chunks = a2a_chunks
        self._comm_stream = None
        self.world_size = dist.get_world_size(group) if dist.is_initialized() else 1
        self.rank = dist.get_rank(group) if dist.is_initialized() else 0
        assert num_experts % self.world_size == 0, "experts must shard evenly"
        self.num_local = num_experts // self.world_size
        self.gate = nn.Linear(d_model, num_experts, bias=False)
        self.experts = ExpertStack(self.num_local, d_model, ffn_dim)
        self.aux_loss: Optional[torch.Tensor] = None
        self.last_counts: Optional[torch.Tensor] = None
        # observability (SURVEY §5 tracing): attach a utils.monitor.StageTimers
        # to time gate / dispatch / a2a / expert_ffn / combine separately
        self.stage_timers = None
        self._capacity_cache: dict = {}
        # elastic membership (parallel/elastic.py): member_slots = ordered
        # alive rank slots; None = static full membership. Dead slots'
        # experts are masked out of routing (reference semantics: dead
        # experts vanish; they return when the slot rejoins).
        self.total_slots = self.world_size
        self.member_slots: Optional[list] = None
        self.register_buffer("expert_alive", torch.ones(num_experts, dtype=torch.bool),
                             persistent=False)

    def update_membership(self, group, member_slots: list, total_slots: int) -> None:
        """Adopt a new generation: process group over `member_slots` (ordered
        as the group's ranks), expert ownership fixed by slot id."""
        self.group = group
        self.total_slots = total_slots
        self.member_slots = list(member_slots)
        self.world_size = len(member_slots)
        self.rank = None  # not needed: a2a uses group rank order
        per_slot = self.num_experts // total_slots
        alive = torch.zeros(self.num_experts, dtype=torch.bool)
        for s in member_slots:
            alive[s * per_slot : (s + 1) * per_slot] = True
        self.expert_alive.copy_(alive.to(self.expert_alive.device))

    def capacity(self, T: int) -> int:
        """Per-(source rank, expert) capacity, padded to the GEMM tile (128
        on GPU where the MFMA kernel runs, 8 on CPU tests)."""
        key = T
        if key not in self._capacity_cache:
            avg = T * self.k / self.num_experts
            pad = 64 if (HAS_HIP and self.gate.weight.is_cuda) else 8
            cap = max(pad, int(math.ceil(avg * self.capacity_factor / pad)) * pad)
            self._capacity_cache[key] = cap
        return self._capacity_cache[key]

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        from contextlib import nullcontext

        tm = self.stage_timers if self.stage_timers is not None else (lambda s: nullcontext())
        lead_shape = x.shape[:-1]
        D = x.shape[-1]
        flat = x.reshape(-1, D)
        T = flat.shape[0]
        E, k, W = self.num_experts, self.k, self.world_size
        C = self.capacity(T)

        elastic = self.member_slots is not None
        members = self.member_slots if elastic else list(range(W))
        P = self.num_experts // self.total_slots  # experts per slot (fixed)

        with tm("moe_gate"):
            logits = self.gate(flat).float()
        if elastic and not bool(self.expert_alive.all()):
            # dead slots' experts leave the routing (reference: dead experts
            # vanish until their server re-announces)
            logits = logits.masked_fill(~self.expert_alive.to(logits.device), -1e9)
        if self.training and self.router_noise_std > 0:
            # exploration noise (Shazeer-style noisy gating): every expert
            # keeps receiving tokens early on, preventing rich-get-richer
            # collapse before the balance loss can act
            logits = logits + self.router_noise_std * torch.randn_like(logits)
        with tm("moe_route"):
            topk_idx, topk_w, probs = Fops.gating_topk(logits, k)
        with torch.no_grad(), tm("moe_assign"):
            pos, counts, inv = Fops.assign_slots(topk_idx, E, C)
        with torch.no_grad():
            if elastic:
                # belt-and-braces: any slot that still picked a dead expert
                # (e.g. every candidate masked) is treated as dropped
                dead_sel = ~self.expert_alive.to(pos.device)[topk_idx.long()]
                pos = pos.masked_fill(dead_sel, -1)
        self.last_counts = counts

        compute_dtype = torch.bfloat16 if flat.is_cuda else flat.dtype
        with tm("moe_dispatch"):
            xbuf = Fops.dispatch(flat.to(compute_dtype), topk_idx, pos, inv, k)  # [E, C, D]

        n_chunks = 1 if elastic else max(1, min(self.a2a_chunks, self.num_local))
        if n_chunks > 1:
            ybuf = self._chunked_expert_pipeline(xbuf, W, P, C, D, n_chunks, tm)
        elif W > 1 or elastic:
            if elastic and members != list(range(self.total_slots)):
                xsend = torch.cat([xbuf[s * P : (s + 1) * P] for s in members])
            else:
                xsend = xbuf
            if W > 1:
                with tm("moe_a2a_fwd"):
                    xsend = _AllToAll.apply(xsend, self.group)  # [W*P, C, D] recv
            xrecv = (
                xsend.view(W, P, C, D).permute(1, 0, 2, 3)
                .reshape(P, W * C, D).contiguous()
            )
            with tm("moe_expert_ffn"):
                ybuf_local = Fops.batched_ffn(xrecv, self.experts)
            ysend = (
                ybuf_local.view(P, W, C, D).permute(1, 0, 2, 3)
                .reshape(W * P, C, D).contiguous()
            )
            if W > 1:
                with tm("moe_a2a_ret"):
                    ysend = _AllToAll.apply(ysend, self.group)
            if elastic and members != list(range(self.total_slots)):
                ybuf = torch.zeros(E, C, D, dtype=ysend.dtype, device=ysend.device)
                for i, s in enumerate(members):
                    ybuf[s * P : (s + 1) * P] = ysend[i * P : (i + 1) * P]
            else:
                ybuf = ysend
        else:
            with tm("moe_expert_ffn"):
                ybuf = Fops.batched_ffn(xbuf, self.experts)

        with tm("moe_combine"):
            y = Fops.combine(ybuf, topk_idx, pos, topk_w, inv)

        aux = None
        if self.aux_loss_coef > 0:
            f = (counts.float() / max(1, T * k)).detach()
            mean_probs = probs.float().mean(dim=0)
            aux = self.aux_loss_coef * E * torch.dot(f.to(mean_probs.device), mean_probs)
        if self.z_loss_coef > 0:
            # router z-loss (ST-MoE): keeps logit magnitudes bounded so the
            # gating softmax never saturates into a zero-gradient regime
            z = torch.logsumexp(logits, dim=-1)
            z_loss = self.z_loss_coef * (z * z).mean()
            aux = z_loss if aux is None else aux + z_loss
        self.aux_loss = aux
        return y.reshape(*lead_shape, D).to(x.dtype)

    def _chunked_expert_pipeline(self, xbuf: torch.Tensor, W: int, P: int, C: int,
                                 D: int, n_chunks: int, tm) -> torch.Tensor:
        """Chunked dispatch→FFN→return pipeline: the
        local expert range [0, P) is split into `n_chunks` contiguous ranges;
        each range's dispatch all-to-all is issued on a side HIP stream so
        chunk i+1's collective overlaps chunk i's expert FFN, and each
        range's return all-to-all overlaps the NEXT chunk's FFN. Backward
        overlaps symmetrically: autograd replays every op on the stream its
        forward used (torch stream-semantics-of-backward), so the reverse
        pipeline needs no extra code. Weight-grad write-vs-accumulate is
        tracked per expert range (ExpertStack._grad_ready_ranges) — see the
        slice-view docstring for the stale-accumulate pitfall this avoids.
        Numerics are IDENTICAL to the unchunked path (same kernels over the
        same per-expert batches; verified by tests on CPU, gloo×2 and GPU).
        """
        use_streams = xbuf.is_cuda
        if use_streams and self._comm_stream is None:
            self._comm_stream = torch.cuda.Stream(device=xbuf.device)
        comm = self._comm_stream
        step = (P + n_chunks - 1) // n_chunks
        bounds = [(p0, min(p0 + step, P)) for p0 in range(0, P, step)]
        x4 = xbuf.view(W, P, C, D)

        # phase 1: per-chunk send buffers (current stream), then ALL dispatch
        # collectives queued on the side stream
        with tm("moe_a2a_fwd"):
            xsends = [x4[:, p0:p1].reshape(W * (p1 - p0), C, D) for p0, p1 in bounds]
            xsends = [xs.contiguous() for xs in xsends]
            xrecv: list = []
            if W > 1 and use_streams:
                ev_in = torch.cuda.Event()
                ev_in.record()
                for xs in xsends:
                    with torch.cuda.stream(comm):
                        comm.wait_event(ev_in)
                        xr = _AllToAll.apply(xs, self.group)
                        ev = torch.cuda.Event()
                        ev.record()
                    xrecv.append((xr, ev))
            elif W > 1:
                xrecv = [(_AllToAll.apply(xs, self.group), None) for xs in xsends]
            else:
                xrecv = [(xs, None) for xs in xsends]

        # phase 2: per chunk: wait its dispatch, run the FFN slice, queue its
        # return collective on the side stream
        yret: list = []
        for i, (p0, p1) in enumerate(bounds):
            pc = p1 - p0
            xr, ev = xrecv[i]
            if ev is not None:
                torch.cuda.current_stream().wait_event(ev)
                xr.record_stream(torch.cuda.current_stream())
            xin = xr.view(W, pc, C, D).permute(1, 0, 2, 3).reshape(pc, W * C, D).contiguous()
            with tm("moe_expert_ffn"):
                y_local = Fops.batched_ffn(xin, self.experts.slice_view(p0, p1))
            ys = y_local.view(pc, W, C, D).permute(1, 0, 2, 3).reshape(W * pc, C, D).contiguous()
            if W > 1 and use_streams:
                ev_y = torch.cuda.Event()
                ev_y.record()
                with torch.cuda.stream(comm):
                    comm.wait_event(ev_y)
                    yr = _AllToAll.apply(ys, self.group)
                    ev_r = torch.cuda.Event()
                    ev_r.record()
                yret.append((yr, ev_r))
            elif W > 1:
                yret.append((_AllToAll.apply(ys, self.group), None))
            else:
                yret.append((ys, None))

        # phase 3: wait all returns, reassemble [W*P, C, D] in expert order
        with tm("moe_a2a_ret"):
            cur = torch.cuda.current_stream() if use_streams else None
            pieces = []
            for i, (yr, ev) in enumerate(yret):
                if ev is not None:
                    cur.wait_event(ev)
                    yr.record_stream(cur)
                p0, p1 = bounds[i]
                pieces.append(yr.view(W, p1 - p0, C, D))
            y4 = torch.cat(pieces, dim=1)
        return y4.reshape(W * P, C, D)
