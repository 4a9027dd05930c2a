"""hipGraph-captured single-token decode for TransformerDMoE serving.

The decode step is launch-bound (dozens of small kernels per token:
layernorms, qkv/proj GEMVs, gating/dispatch/FFN/combine, lm_head); replaying
one captured hipGraph per token removes the per-kernel launch gaps (the
north-star's "HIP streams and graphs instead of a tracing compiler" —
torch.cuda.CUDAGraph IS hipGraph on ROCm).

Graph-capture constraints shape the design:
- STATIC state: pre-allocated KV caches [B, H, S_max, Dh], a one-element
  position index tensor, an additive attention mask over the FULL cache
  (-inf beyond the current length), and static input/output token buffers.
  Between replays only cheap eager writes touch them (mask[pos]=0,
  pos.fill_, ids.copy_); the replayed graph reads/writes them in place.
- No host syncs / dynamic shapes inside the step: attention runs over the
  whole padded cache with the mask (wasted FLOPs are trivial at decode
  sizes), the MoE path's capacity buffers are static by construction, and
  greedy argmax keeps sampling device-side.

Use `GraphDecoder(model, ...)` for local (ExpertParallelMoE / dense) MoE
layers on GPU. `use_graph=False` runs the identical static-state step
eagerly — the A/B baseline and the CPU-debug path.
"""
from __future__ import annotations

import torch
import torch.nn.functional as F

from .transformer_dmoe import TransformerDMoE


class GraphDecoder:
    def __init__(self, model: TransformerDMoE, batch_size: int = 1,
                 max_len: int | None = None, use_graph: bool = True):
        self.model = model.eval()
        cfg = model.config
        p = next(model.parameters())
        self.device, self.dtype = p.device, p.dtype
        self.B = batch_size
        self.S = max_len or cfg.seq_len
        self.H = cfg.n_heads
        self.Dh = cfg.d_model // cfg.n_heads
        self.use_graph = use_graph and self.device.type == "cuda"
        dev, dt = self.device, self.dtype
        self.k_cache = [torch.zeros(self.B, self.H, self.S, self.Dh, device=dev, dtype=dt)
                        for _ in model.blocks]
        self.v_cache = [torch.zeros(self.B, self.H, self.S, self.Dh, device=dev, dtype=dt)
                        for _ in model.blocks]
        # additive mask over the padded cache; entries open up as length grows
        self.mask = torch.full((self.B, 1, 1, self.S), float("-inf"), device=dev,
                               dtype=torch.float32 if dt == torch.float32 else dt)
        self.pos = torch.zeros(1, dtype=torch.long, device=dev)
        self.ids = torch.zeros(self.B, 1, dtype=torch.long, device=dev)
        self._graph = None
        self._out = None

    # ---- the (capturable) single-token step ---------------------------
    def _step_body(self) -> torch.Tensor:
        m = self.model
        B = self.B
        x = m.embed(self.ids) + m.pos_embed(self.pos)[None]
        for i, block in enumerate(m.blocks):
            h = block.ln1(x)
            qkv = block.qkv(h).reshape(B, 1, 3, self.H, self.Dh)
            q = qkv[:, :, 0].transpose(1, 2)
            k = qkv[:, :, 1].transpose(1, 2)
            v = qkv[:, :, 2].transpose(1, 2)
            self.k_cache[i].index_copy_(2, self.pos, k)
            self.v_cache[i].index_copy_(2, self.pos, v)
            attn = F.scaled_dot_product_attention(
                q, self.k_cache[i], self.v_cache[i], attn_mask=self.mask.to(q.dtype))
            x = x + block.proj(attn.transpose(1, 2).reshape(B, 1, -1))
            x = x + block.moe(block.ln2(x))
        logits = m.lm_head(m.ln_f(x[:, -1]))
        return logits.argmax(dim=-1, keepdim=True)

    def _advance(self, next_ids: torch.Tensor, new_pos: int) -> None:
        """Eager between-replay state updates (cheap, fixed cost)."""
        self.ids.copy_(next_ids)
        self.mask[..., new_pos] = 0.0
        self.pos.fill_(new_pos)

    # ---- prefill (eager; writes the same static caches) ----------------
    @torch.no_grad()
    def prefill(self, prompt_ids: torch.Tensor) -> torch.Tensor:
        m = self.model
        B, S0 = prompt_ids.shape
        assert B == self.B and S0 < self.S
        pos = torch.arange(S0, device=self.device)
        x = m.embed(prompt_ids) + m.pos_embed(pos)[None]
        for i, block in enumerate(m.blocks):
            h = block.ln1(x)
            qkv = block.qkv(h).reshape(B, S0, 3, self.H, self.Dh)
            q = qkv[:, :, 0].transpose(1, 2)
            k = qkv[:, :, 1].transpose(1, 2)
            v = qkv[:, :, 2].transpose(1, 2)
            self.k_cache[i][:, :, :S0] = k
            self.v_cache[i][:, :, :S0] = v
            attn = F.scaled_dot_product_attention(q, k, v, is_causal=True)
            x = x + block.proj(attn.transpose(1, 2).reshape(B, S0, -1))
            x = x + block.moe(block.ln2(x))
        logits = m.lm_head(m.ln_f(x[:, -1]))
        nxt = logits.argmax(dim=-1, keepdim=True)
        self.mask[..., :S0] = 0.0
        self._advance(nxt, S0)
        return nxt

    def _ensure_graph(self) -> None:
        if self._graph is not None or not self.use_graph:
            return
        # warm up the exact step workload (allocator, kernel caches), then
        # capture. Warmup runs on a side stream per the torch contract; its
        # state side effects are idempotent (same pos/ids rewritten).
        s = torch.cuda.Stream(device=self.device)
        s.wait_stream(torch.cuda.current_stream(self.device))
        with torch.cuda.stream(s):
            for _ in range(2):
                self._step_body()
        torch.cuda.current_stream(self.device).wait_stream(s)
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g):
            self._out = self._step_body()
        self._graph = g

    # ---- public API -----------------------------------------------------
    @torch.no_grad()
    def generate(self, prompt_ids: torch.Tensor, max_new_tokens: int) -> torch.Tensor:
        """Greedy decode; returns prompt extended by max_new_tokens (bounded
        by the static cache length)."""
        toks = [self.prefill(prompt_ids)]
        n = min(max_new_tokens - 1, self.S - prompt_ids.shape[1] - 1)
        if self.use_graph:
            self._ensure_graph()
        base = prompt_ids.shape[1]
        for i in range(n):
            if self.use_graph:
                self._graph.replay()
                nxt = self._out.clone()
            else:
                nxt = self._step_body()
            toks.append(nxt)
            self._advance(nxt, base + i + 1)
        return torch.cat([prompt_ids] + toks, dim=1)
