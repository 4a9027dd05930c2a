"""Expert block zoo (experts are small trainable sub-networks,
e.g. FFN blocks; `name_to_block` mirrors the reference's expert_cls registry).

These are the PyTorch-semantics definitions — the numerics ground truth for
kernel tests. On MI355X the hot path does NOT call these modules per-expert:
expert weights are held in stacked [num_local_experts, ...] buffers and the
grouped CDNA4 HIP kernels (ops/) compute all local experts in one launch
(see parallel/expert_shard.py). These modules remain the source of truth for
CPU tests, single-expert RPC serving, and checkpoint layout.
"""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F


class FFNExpert(nn.Module):
    """Position-independent feed-forward expert: Linear(d, 4d) → GELU →
    Linear(4d, d). The canonical DMoE expert (SURVEY.md §2 hot-op #1)."""

    def __init__(self, hidden_dim: int, ffn_dim: int | None = None, activation: str = "gelu"):
        super().__init__()
        ffn_dim = ffn_dim or 4 * hidden_dim
        self.fc1 = nn.Linear(hidden_dim, ffn_dim)
        self.fc2 = nn.Linear(ffn_dim, hidden_dim)
        self.activation = activation

    def _act(self, x: torch.Tensor) -> torch.Tensor:
        if self.activation == "gelu":
            return F.gelu(x, approximate="tanh")
        elif self.activation == "relu":
            return F.relu(x)
        raise ValueError(f"unknown activation {self.activation}")

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return self.fc2(self._act(self.fc1(x)))


class TransformerExpert(nn.Module):
    """A full pre-norm transformer encoder layer as an expert (the reference's
    'transformer' block): LN → MHA → residual → LN → FFN → residual.
    Input [batch, seq, d] or [batch, d] (treated as seq=1)."""

    def __init__(self, hidden_dim: int, num_heads: int = 8, ffn_dim: int | None = None):
        super().__init__()
        self.ln1 = nn.LayerNorm(hidden_dim)
        self.attn = nn.MultiheadAttention(hidden_dim, num_heads, batch_first=True)
        self.ln2 = nn.LayerNorm(hidden_dim)
        self.ffn = FFNExpert(hidden_dim, ffn_dim)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        squeeze = x.dim() == 2
        if squeeze:
            x = x.unsqueeze(1)
        h = self.ln1(x)
        attn_out, _ = self.attn(h, h, h, need_weights=False)
        x = x + attn_out
        x = x + self.ffn(self.ln2(x))
        return x.squeeze(1) if squeeze else x


class NopExpert(nn.Module):
    """Identity expert with one dummy trainable scalar (reference 'nop' block:
    used by throughput benchmarks to isolate transport cost)."""

    def __init__(self, hidden_dim: int):
        super().__init__()
        self.scale = nn.Parameter(torch.zeros(1))

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return x + 0.0 * self.scale


name_to_block = {
    "ffn": FFNExpert,
    "transformer": TransformerExpert,
    "nop": NopExpert,
}
