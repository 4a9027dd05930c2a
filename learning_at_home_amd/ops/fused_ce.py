"""Chunked fused linear + cross-entropy (trunk-side memory/time saver).

`F.cross_entropy(lm_head(x).float(), labels)` materializes [T, V] logits in
bf16 AND fp32 (~19 GB at the flagship bench) plus full softmax fwd/bwd
passes. This computes the same loss token-chunk by token-chunk: each chunk's
logits live only transiently; the backward recomputes the chunk's softmax
and feeds dX / dW GEMMs directly. Library GEMMs (hipBLASLt) — this is a
torch-level fusion, not a kernel.

Numerics: per-chunk logits from the bf16 GEMM, read by the CE kernels in
their native dtype with fp32 lane conversion + fp32 accumulation — bitwise
the same result as materializing an fp32 copy first (the values are already
bf16-rounded), without the cast kernel or the fp32 [chunk, V] traffic.
Loss is the exact valid-token-mean CE. Labels equal to `ignore_index`
(default -100, matching F.cross_entropy) contribute neither to the loss
nor to the gradients, and the mean is over VALID tokens only — identical
semantics to the unfused `F.cross_entropy(..., ignore_index=-100)` path.
"""
from __future__ import annotations

import torch

from . import _C, HAS_HIP


def _use_ce_kernels(x: torch.Tensor, weight: torch.Tensor) -> bool:
    return (HAS_HIP and x.is_cuda and weight.shape[0] % 4 == 0
            and x.dtype in (torch.bfloat16, torch.float32))


class _ChunkedLinearCE(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x: torch.Tensor, weight: torch.Tensor, labels: torch.Tensor,
                chunk: int, ignore_index: int):
        T = x.shape[0]
        total = torch.zeros((), dtype=torch.float32, device=x.device)
        n_valid = int((labels != ignore_index).sum().item())
        use_hip = _use_ce_kernels(x, weight)
        labels_i32 = labels.int().contiguous() if use_hip else None
        lse_all = torch.empty(T, dtype=torch.float32, device=x.device) if use_hip else None
        for s in range(0, T, chunk):
            xs = x[s : s + chunk]
            if use_hip:
                # one logits read: online lse + label gather (ce.hip) straight
                # off the GEMM's native dtype (bf16 weights → bf16 logits; the
                # kernel converts lanes to fp32 — identical numerics to an
                # explicit .float() materialization, minus the cast kernel and
                # the fp32 [chunk, V] traffic); the torch path materializes
                # log-softmax over [chunk, V] fp32 instead
                logits = xs @ weight.t()
                lse, picked = _C.ce_rowstats(logits, labels_i32[s : s + chunk],
                                             ignore_index)
                lse_all[s : s + chunk] = lse
                total += (lse - picked).sum()
            else:
                total += torch.nn.functional.cross_entropy(
                    (xs @ weight.t()).float(), labels[s : s + chunk],
                    reduction="sum", ignore_index=ignore_index,
                )
        ctx.save_for_backward(x, weight, labels,
                              lse_all if use_hip else torch.empty(0))
        ctx.chunk = chunk
        ctx.ignore_index = ignore_index
        ctx.n_valid = n_valid
        ctx.use_hip = use_hip
        return total / max(n_valid, 1)

    @staticmethod
    def backward(ctx, dloss: torch.Tensor):
        x, weight, labels, lse_all = ctx.saved_tensors
        T = x.shape[0]
        chunk = ctx.chunk
        ignore_index = ctx.ignore_index
        dx = torch.empty_like(x)
        dw = torch.zeros_like(weight, dtype=torch.float32)
        if ctx.use_hip:
            labels_i32 = labels.int().contiguous()
            # scalar sync once per backward: dloss is a 0-dim tensor (1.0 in
            # ordinary training); the fused bwd kernel takes gscale by value
            gs = float(dloss) / max(ctx.n_valid, 1)
            for s in range(0, T, chunk):
                xs = x[s : s + chunk]
                logits = xs @ weight.t()  # native dtype; kernel reads bf16/fp32
                dlogits = _C.ce_bwd(logits, labels_i32[s : s + chunk],
                                    lse_all[s : s + chunk].contiguous(), gs,
                                    ignore_index).to(x.dtype)
                dx[s : s + chunk] = dlogits @ weight
                dw += (dlogits.t() @ xs).float()
            return dx, dw.to(weight.dtype), None, None, None
        scale = dloss.float() / max(ctx.n_valid, 1)
        for s in range(0, T, chunk):
            xs = x[s : s + chunk]
            ls = labels[s : s + chunk]
            valid = ls != ignore_index
            logits = (xs @ weight.t()).float()
            probs = torch.softmax(logits, dim=-1)
            # clamp ignored labels to 0 for the scatter (their rows are zeroed
            # below) — an unclamped -100 would wrap to column V-100
            safe_labels = torch.where(valid, ls, torch.zeros_like(ls))
            probs[torch.arange(probs.shape[0], device=x.device), safe_labels] -= 1.0
            probs[~valid] = 0.0
            dlogits = (probs * scale).to(x.dtype)
            dx[s : s + chunk] = dlogits @ weight
            dw += (dlogits.t() @ xs).float()
        return dx, dw.to(weight.dtype), None, None, None


def chunked_linear_cross_entropy(x: torch.Tensor, weight: torch.Tensor,
                                 labels: torch.Tensor, chunk: int = 8192,
                                 ignore_index: int = -100) -> torch.Tensor:
    """loss = CE(x @ weight^T, labels), valid-token-mean, without
    materializing the full logits. x [T, D], weight [V, D] (tied lm_head),
    labels [T]; labels == ignore_index are skipped exactly as in
    F.cross_entropy."""
    return _ChunkedLinearCE.apply(x.reshape(-1, x.shape[-1]), weight,
                                  labels.reshape(-1), chunk, ignore_index)
