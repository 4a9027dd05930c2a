"""Flagship model: Transformer-DMoE (BASELINE.json headline config —
"Transformer-DMoE 256-expert"). Each block = attention + a Mixture-of-Experts
FFN layer whose experts live on expert-parallel GPU ranks (parallel/) or
behind RemoteMixtureOfExperts (client/, reference-semantics RPC mode).

The MoE layer is injected via `moe_factory(layer_idx, config) -> nn.Module`
mapping hidden states [B, S, D] → [B, S, D], so the same trunk runs over
(a) the RCCL+HIP expert-parallel plane (bench.py hot path),
(b) socket-RPC RemoteMixtureOfExperts (CPU parity tests, config #1),
(c) a plain dense FFN (baseline comparisons).
"""
from __future__ import annotations

import os
from dataclasses import dataclass
from typing import Callable, Optional

import torch
import torch.nn as nn
import torch.nn.functional as F


def _use_flash_attention() -> bool:
    """LAH_FLASH_ATTN=1 routes block attention through the CDNA4 flash
    kernels (ops.functional.flash_attention; falls back to sdpa whenever the
    shape/dtype doesn't qualify). Default OFF until the kernel's performance
    pass lands (docs/ROUND2_NOTES.md §2) — numerics are already verified."""
    return os.environ.get("LAH_FLASH_ATTN", "0") == "1"


@dataclass
class TransformerDMoEConfig:
    vocab_size: int = 32000
    d_model: int = 1024
    n_heads: int = 16
    n_layers: int = 6
    seq_len: int = 1024
    num_experts: int = 256
    k_best: int = 4
    ffn_dim: int = 4096  # per-expert hidden
    grid_dims: tuple = ()  # uid grid, e.g. (16, 16); () -> flat (num_experts,)
    dropout: float = 0.0

    def __post_init__(self):
        if not self.grid_dims:
            self.grid_dims = (self.num_experts,)


class DenseFFN(nn.Module):
    def __init__(self, d_model: int, ffn_dim: int):
        super().__init__()
        self.fc1 = nn.Linear(d_model, ffn_dim)
        self.fc2 = nn.Linear(ffn_dim, d_model)

    def forward(self, x):
        return self.fc2(F.gelu(self.fc1(x), approximate="tanh"))


class DMoETransformerBlock(nn.Module):
    """Pre-norm block: x + attn(LN(x)); x + moe(LN(x))."""

    def __init__(self, config: TransformerDMoEConfig, moe: nn.Module):
        super().__init__()
        d = config.d_model
        self.ln1 = nn.LayerNorm(d)
        self.qkv = nn.Linear(d, 3 * d, bias=False)
        self.proj = nn.Linear(d, d, bias=False)
        self.n_heads = config.n_heads
        self.ln2 = nn.LayerNorm(d)
        self.moe = moe

    def forward(self, x: torch.Tensor, kv_cache: Optional[list] = None) -> torch.Tensor:
        """kv_cache (decode only): a 2-element list [k, v] (or empty, filled on
        prefill) of [B, H, T_past, Dh]; new keys/values are appended and the
        query attends over the whole cache (is_causal only without a past)."""
        B, S, D = x.shape
        h = self.ln1(x)
        qkv = self.qkv(h).reshape(B, S, 3, self.n_heads, D // self.n_heads)
        q, k, v = (qkv[:, :, i].transpose(1, 2) for i in range(3))
        if kv_cache is not None:
            if kv_cache:
                k = torch.cat([kv_cache[0], k], dim=2)
                v = torch.cat([kv_cache[1], v], dim=2)
            kv_cache[:] = [k, v]
            causal = k.shape[2] == S  # prefill; decode steps see only the past
        else:
            causal = True
        if causal and _use_flash_attention():
            from ..ops.functional import flash_attention

            attn = flash_attention(q, k, v)
        else:
            attn = F.scaled_dot_product_attention(q, k, v, is_causal=causal)
        x = x + self.proj(attn.transpose(1, 2).reshape(B, S, D))
        x = x + self.moe(self.ln2(x))
        return x


class TransformerDMoE(nn.Module):
    def __init__(
        self,
        config: TransformerDMoEConfig,
        moe_factory: Optional[Callable[[int, TransformerDMoEConfig], nn.Module]] = None,
    ):
        super().__init__()
        self.config = config
        if moe_factory is None:
            moe_factory = lambda i, cfg: DenseFFN(cfg.d_model, cfg.ffn_dim)  # noqa: E731
        self.embed = nn.Embedding(config.vocab_size, config.d_model)
        self.pos_embed = nn.Embedding(config.seq_len, config.d_model)
        nn.init.normal_(self.embed.weight, std=0.02)
        nn.init.normal_(self.pos_embed.weight, std=0.02)
        self.blocks = nn.ModuleList(
            [DMoETransformerBlock(config, moe_factory(i, config)) for i in range(config.n_layers)]
        )
        self.ln_f = nn.LayerNorm(config.d_model)
        self.lm_head = nn.Linear(config.d_model, config.vocab_size, bias=False)
        self.lm_head.weight = self.embed.weight  # tied

    def hidden_states(self, input_ids: torch.Tensor) -> torch.Tensor:
        B, S = input_ids.shape
        pos = torch.arange(S, device=input_ids.device)
        x = self.embed(input_ids) + self.pos_embed(pos)[None]
        for block in self.blocks:
            x = block(x)
        return self.ln_f(x)

    def forward(self, input_ids: torch.Tensor) -> torch.Tensor:
        return self.lm_head(self.hidden_states(input_ids))

    @torch.no_grad()
    def generate(self, input_ids: torch.Tensor, max_new_tokens: int,
                 temperature: float = 0.0) -> torch.Tensor:
        """Greedy (temperature=0) or sampled decode with per-block KV caches.
        Returns input_ids extended by up to max_new_tokens (bounded by the
        positional table, config.seq_len). Serving entry point — the MoE
        layers run their normal routed forward per decode step."""
        B, S = input_ids.shape
        caches = [[] for _ in self.blocks]

        def step(ids: torch.Tensor, pos0: int) -> torch.Tensor:
            pos = torch.arange(pos0, pos0 + ids.shape[1], device=ids.device)
            x = self.embed(ids) + self.pos_embed(pos)[None]
            for block, cache in zip(self.blocks, caches):
                x = block(x, kv_cache=cache)
            logits = self.lm_head(self.ln_f(x[:, -1]))
            if temperature > 0:
                probs = F.softmax(logits.float() / temperature, dim=-1)
                return torch.multinomial(probs, 1)
            return logits.argmax(dim=-1, keepdim=True)

        out = input_ids
        nxt = step(input_ids, 0)  # prefill
        for i in range(max_new_tokens):
            out = torch.cat([out, nxt], dim=1)
            if out.shape[1] >= self.config.seq_len or i == max_new_tokens - 1:
                break
            nxt = step(nxt, out.shape[1] - 1)
        return out

    def loss(self, input_ids: torch.Tensor, labels: torch.Tensor,
             fused_ce: bool = False) -> torch.Tensor:
        hidden = self.hidden_states(input_ids)
        if fused_ce:
            # chunked fused linear+CE: never materializes the full [T, V]
            # logits (-46 GB at the flagship bench, ~3% slower — memory knob)
            from ..ops.fused_ce import chunked_linear_cross_entropy

            loss = chunked_linear_cross_entropy(hidden, self.lm_head.weight, labels)
        else:
            logits = self.lm_head(hidden)
            loss = F.cross_entropy(logits.flatten(0, 1).float(), labels.flatten())
        # accumulate load-balance auxiliary losses collected by MoE layers
        for block in self.blocks:
            aux = getattr(block.moe, "aux_loss", None)
            if aux is not None:
                loss = loss + aux
        return loss
