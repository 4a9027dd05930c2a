"""Chunked fused linear + cross-entropy (trunk-side memory/time saver).

`F.cross_entropy(lm_head(x).float(), labels)` materializes [T, V] logits in
bf16 AND fp32 (~19 GB at the flagship bench) plus full softmax fwd/bwd
passes. This computes the same loss token-chunk by token-chunk: each chunk's
logits live only transiently; the backward recomputes the chunk's softmax
and feeds dX / dW GEMMs directly. Library GEMMs (hipBLASLt) — this is a
torch-level fusion, not a kernel.

Numerics: per-chunk logits in fp32 via bf16 GEMM (same as the unfused path);
loss is the exact token-mean CE.
"""
from __future__ import annotations

import torch


class _ChunkedLinearCE(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x: torch.Tensor, weight: torch.Tensor, labels: torch.Tensor,
                chunk: int):
        T = x.shape[0]
        total = torch.zeros((), dtype=torch.float32, device=x.device)
        for s in range(0, T, chunk):
            xs = x[s : s + chunk]
            logits = (xs @ weight.t()).float()
            total += torch.nn.functional.cross_entropy(
                logits, labels[s : s + chunk], reduction="sum"
            )
        ctx.save_for_backward(x, weight, labels)
        ctx.chunk = chunk
        return total / T

    @staticmethod
    def backward(ctx, dloss: torch.Tensor):
        x, weight, labels = ctx.saved_tensors
        T = x.shape[0]
        chunk = ctx.chunk
        dx = torch.empty_like(x)
        dw = torch.zeros_like(weight, dtype=torch.float32)
        scale = (dloss.float() / T)
        for s in range(0, T, chunk):
            xs = x[s : s + chunk]
            logits = (xs @ weight.t()).float()
            probs = torch.softmax(logits, dim=-1)
            probs[torch.arange(probs.shape[0], device=x.device), labels[s : s + chunk]] -= 1.0
            dlogits = (probs * scale).to(x.dtype)
            dx[s : s + chunk] = dlogits @ weight
            dw += (dlogits.t() @ xs).float()
        return dx, dw.to(weight.dtype), None, None


def chunked_linear_cross_entropy(x: torch.Tensor, weight: torch.Tensor,
                                 labels: torch.Tensor, chunk: int = 8192) -> torch.Tensor:
    """loss = CE(x @ weight^T, labels), token-mean, without materializing the
    full logits. x [T, D], weight [V, D] (tied lm_head), labels [T]."""
    return _ChunkedLinearCE.apply(x.reshape(-1, x.shape[-1]), weight,
                                  labels.reshape(-1), chunk)
