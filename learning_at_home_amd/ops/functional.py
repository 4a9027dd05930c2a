"""Autograd-integrated ops over the MI355X kernels, with pure-torch CPU
fallbacks of IDENTICAL semantics (capacity drops, renormalize-over-survivors,
recomputed-free fused backward). The CPU path doubles as the numerics
reference for the GPU tests (tests/test_ops_gpu.py compares kernel outputs
against these functions in fp32).

GEMM policy: ALL six expert-FFN GEMMs are hand-written MFMA kernels — the
four activation-side NT GEMMs with fused bias/GELU epilogues (ffn_gemm.hip)
and the two weight-gradient TN GEMMs writing fp32 grads in place
(ffn_gemm_tn.hip); bias grads via the vectorized colsum kernel (optim.hip).
"""
from __future__ import annotations


import torch
import torch.nn.functional as F

from . import _C, HAS_HIP

# epilogue ids (ffn_gemm.hip)
EPI_NONE, EPI_BIAS, EPI_BIAS_GELU, EPI_GELU_GRAD = 0, 1, 2, 3


# ---------------------------------------------------------------------------
# gating: softmax + top-k + (optionally) differentiable full probs
# ---------------------------------------------------------------------------
class _GatingTopK(torch.autograd.Function):
    """GPU fused gating. Outputs (topk_idx int32, topk_w fp32, probs bf16).
    probs is differentiable for TOKEN-UNIFORM upstream grads only (the
    importance/load-balance losses) — d_probs is reduced to a per-expert
    vector for the fused backward kernel."""

    @staticmethod
    def forward(ctx, logits: torch.Tensor, k: int):
        probs, topk_idx, topk_w = _C.gating_topk_fwd(logits.contiguous(), k)
        ctx.save_for_backward(probs, topk_idx, topk_w)
        ctx.k = k
        ctx.mark_non_differentiable(topk_idx)
        return topk_idx, topk_w, probs

    @staticmethod
    def backward(ctx, d_idx, d_topk_w, d_probs):
        probs, topk_idx, topk_w = ctx.saved_tensors
        aux_g = None
        if d_probs is not None:
            # the kernel applies the per-token softmax jacobian with ONE row
            # vector g: for token-uniform d_probs the correct row is the MEAN
            # over tokens (sum would overcount by T — a 2000x gate-gradient
            # blowup observed at flagship scale)
            aux_g = d_probs.float().mean(dim=0).contiguous()
        if d_topk_w is None:
            d_topk_w = torch.zeros_like(topk_w)
        d_logits = _C.gating_topk_bwd(probs, topk_idx, topk_w,
                                      d_topk_w.float().contiguous(), aux_g)
        return d_logits, None


def gating_topk(logits: torch.Tensor, k: int):
    """Returns (topk_idx [T,k] int32, topk_w [T,k] fp32 = softmax over the
    selected logits, probs [T,E] = full softmax)."""
    if logits.is_cuda and HAS_HIP:
        return _GatingTopK.apply(logits.float(), k)
    probs = F.softmax(logits.float(), dim=-1)
    top_logits, topk_idx = logits.float().topk(k, dim=-1)
    topk_w = F.softmax(top_logits, dim=-1)
    return topk_idx.int(), topk_w, probs


# ---------------------------------------------------------------------------
# slot assignment (non-differentiable routing bookkeeping)
# ---------------------------------------------------------------------------
def assign_slots(topk_idx: torch.Tensor, E: int, C: int):
    """Returns (pos [T,k] int32 slot-in-expert or -1 dropped,
    counts [E] int32, inv [E,C] int32 flat (t*k+j) or -1)."""
    if topk_idx.is_cuda and HAS_HIP:
        pos, counts, inv = _C.assign_slots(topk_idx.contiguous(), E, C)
        return pos, counts, inv
    T, k = topk_idx.shape
    flat_e = topk_idx.reshape(-1).long()
    order = torch.argsort(flat_e, stable=True)
    sorted_e = flat_e[order]
    counts_full = torch.bincount(flat_e, minlength=E).int()
    starts = torch.zeros(E, dtype=torch.long)
    starts[1:] = torch.cumsum(counts_full, 0)[:-1].long()
    rank = torch.arange(flat_e.numel()) - starts[sorted_e]
    pos = torch.full_like(flat_e, -1, dtype=torch.int32)
    keep = rank < C
    pos[order[keep]] = rank[keep].int()
    inv = torch.full((E, C), -1, dtype=torch.int32)
    kept_slots = sorted_e[keep] * C + rank[keep]
    inv.view(-1)[kept_slots] = order[keep].int()
    counts = counts_full
    return pos.view(T, k), counts, inv


# ---------------------------------------------------------------------------
# dispatch: tokens -> capacity-bucketed expert batches
# ---------------------------------------------------------------------------
class _Dispatch(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x: torch.Tensor, topk_idx, pos, inv, k: int):
        ctx.save_for_backward(topk_idx, pos)
        ctx.T = x.shape[0]
        if x.is_cuda and HAS_HIP:
            return _C.dispatch_gather(x.contiguous(), inv.contiguous(), k)
        E, C = inv.shape
        D = x.shape[1]
        xbuf = torch.zeros(E, C, D, dtype=x.dtype)
        mask = inv.view(-1) >= 0
        tok = (inv.view(-1)[mask].long() // k)
        xbuf.view(-1, D)[mask] = x[tok]
        return xbuf

    @staticmethod
    def backward(ctx, dxbuf: torch.Tensor):
        topk_idx, pos = ctx.saved_tensors
        T = ctx.T
        if dxbuf.is_cuda and HAS_HIP:
            dx = _C.dispatch_bwd(dxbuf.contiguous(), topk_idx, pos, T)
            return dx, None, None, None, None
        E, C, D = dxbuf.shape
        dx = torch.zeros(T, D, dtype=dxbuf.dtype)
        flat_pos = pos.view(-1).long()
        flat_e = topk_idx.view(-1).long()
        keep = flat_pos >= 0
        rows = dxbuf.view(-1, D)[(flat_e[keep] * C + flat_pos[keep])]
        tok = torch.div(torch.nonzero(keep).squeeze(1), pos.shape[1], rounding_mode="floor")
        dx.index_add_(0, tok, rows)
        return dx, None, None, None, None


def dispatch(x, topk_idx, pos, inv, k):
    return _Dispatch.apply(x, topk_idx, pos, inv, k)


# ---------------------------------------------------------------------------
# combine: weighted sum over surviving slots (renormalized)
# ---------------------------------------------------------------------------
class _Combine(torch.autograd.Function):
    @staticmethod
    def forward(ctx, ybuf: torch.Tensor, topk_idx, pos, w, inv):
        if ybuf.is_cuda and HAS_HIP:
            y = _C.combine_fwd(ybuf.contiguous(), topk_idx, pos, w.contiguous())
        else:
            y = _combine_fwd_torch(ybuf, topk_idx, pos, w)
        ctx.save_for_backward(ybuf, y, topk_idx, pos, w, inv)
        return y

    @staticmethod
    def backward(ctx, dy: torch.Tensor):
        ybuf, y, topk_idx, pos, w, inv = ctx.saved_tensors
        if dy.is_cuda and HAS_HIP:
            dybuf, dw = _C.combine_bwd(dy.contiguous(), ybuf, y, topk_idx, pos,
                                       w.contiguous(), inv)
            return dybuf, None, None, dw, None
        dybuf, dw = _combine_bwd_torch(dy, ybuf, y, topk_idx, pos, w)
        return dybuf, None, None, dw, None


def _survivor_weights(pos, w):
    surv = pos >= 0
    s = (w * surv).sum(-1, keepdim=True)
    # clamped renormalization (matches dispatch.hip): bounds the gradient
    # amplification when capacity drops remove most of the softmax mass
    inv_s = torch.where(s > 0, 1.0 / s.clamp(min=0.2), torch.zeros_like(s))
    return (w * surv) * inv_s, surv, inv_s


def _combine_fwd_torch(ybuf, topk_idx, pos, w):
    E, C, D = ybuf.shape
    T, k = topk_idx.shape
    w_hat, surv, _ = _survivor_weights(pos, w)
    flat = (topk_idx.long() * C + pos.clamp(min=0).long()).view(-1)
    rows = ybuf.view(-1, D)[flat].view(T, k, D)
    return (w_hat.unsqueeze(-1).to(rows.dtype) * rows * surv.unsqueeze(-1)).sum(1)


def _combine_bwd_torch(dy, ybuf, y, topk_idx, pos, w):
    E, C, D = ybuf.shape
    T, k = topk_idx.shape
    w_hat, surv, inv_s = _survivor_weights(pos, w)
    dybuf = torch.zeros_like(ybuf)
    flat_valid = pos.view(-1) >= 0
    flat_slot = (topk_idx.long() * C + pos.clamp(min=0).long()).view(-1)[flat_valid]
    scaled = (w_hat.unsqueeze(-1).to(dy.dtype) * dy.unsqueeze(1)).view(-1, D)[flat_valid]
    dybuf.view(-1, D)[flat_slot] = scaled
    # dw_j = (<dy, y_j> - <dy, y>) / s for survivors
    rows = ybuf.view(-1, D)[(topk_idx.long() * C + pos.clamp(min=0).long()).view(-1)].view(T, k, D)
    dot_yj = torch.einsum("td,tkd->tk", dy.float(), rows.float())
    dot_y = torch.einsum("td,td->t", dy.float(), y.float()).unsqueeze(-1)
    s = (w * surv).sum(-1, keepdim=True)
    clamped = (s > 0) & (s < 0.2)  # in the clamp region d(w/0.2)/dw is diagonal
    dw = (torch.where(clamped, dot_yj, dot_yj - dot_y) * inv_s) * surv
    return dybuf, dw.float()


def combine(ybuf, topk_idx, pos, w, inv):
    return _Combine.apply(ybuf, topk_idx, pos, w, inv)


# ---------------------------------------------------------------------------
# batched expert FFN (GPU fused path; weight grads -> stack.main_grad_*)
# ---------------------------------------------------------------------------
def _wait_pending_opt(stack) -> None:
    # side-stream optimizer overlap (FusedExpertSGD.set_side_stream): the
    # previous step's weight update may still be in flight on the side
    # stream — fence before reading W (forward) or writing grads (backward).
    # Resolve through slice views to the owning stack so the event clears.
    owner = getattr(stack, "_parent", stack)
    ev = getattr(owner, "_opt_done_event", None)
    if ev is not None:
        torch.cuda.current_stream().wait_event(ev)
        owner._opt_done_event = None


class _BatchedFFNHip(torch.autograd.Function):
    @staticmethod
    def forward(ctx, xbuf: torch.Tensor, stack):
        _wait_pending_opt(stack)
        E, C, D = xbuf.shape
        # der = gelu'(pre-act), computed for free in the fwd epilogue (one tanh)
        der = torch.empty(E, C, stack.ffn_dim, dtype=torch.bfloat16, device=xbuf.device)
        h = _C.batched_gemm_nt(xbuf.contiguous(), stack.W1c, stack.b1_f32, der, EPI_BIAS_GELU)
        y = _C.batched_gemm_nt(h, stack.W2c, stack.b2_f32, None, EPI_BIAS)
        ctx.save_for_backward(xbuf, der, h)
        ctx.stack = stack
        return y

    @staticmethod
    def backward(ctx, dy: torch.Tensor):
        xbuf, der, h = ctx.saved_tensors
        stack = ctx.stack
        _wait_pending_opt(stack)
        dy = dy.contiguous()
        # dh_pre = (dy @ W2t^T) * der; dx = dh_pre @ W1t^T   (MFMA NT)
        # (computed BEFORE any fused weight update below: dx must see the
        # same weights the forward used)
        dh_pre = _C.batched_gemm_nt(dy, stack.W2t, None, der, EPI_GELU_GRAD)
        dxbuf = _C.batched_gemm_nt(dh_pre, stack.W1t, None, None, EPI_NONE)
        acc = stack.grad_ready
        fused = getattr(stack, "_fused_sr_opt", None)
        if fused is not None and not acc:
            # fused dW + stochastic-rounding SGD: the TN epilogue applies the
            # update to W/W^T in place — no fp32 grad buffer round trip
            # (ffn_gemm_tn.hip; one backward per step by contract)
            lr, wd = fused.lr, fused.weight_decay
            seed = fused.sr_seed()
            _C.fused_dw_sgd_sr(dh_pre, xbuf, stack.W1c, stack.W1t, lr, wd, seed)
            _C.fused_dw_sgd_sr(dy, h, stack.W2c, stack.W2t, lr, wd, seed ^ 0x55555555)
        elif fused is not None:
            raise RuntimeError(
                "fused dW+SGD supports exactly one backward per optimizer step; "
                "construct FusedExpertSGD(fused_backward=False) for gradient accumulation"
            )
        else:
            # weight grads: hand-written TN MFMA kernels write the fp32 grad
            # buffers in place (accumulate only after the first backward of a step)
            _C.batched_gemm_tn(dh_pre, xbuf, stack.main_grad_W1, acc)
            _C.batched_gemm_tn(dy, h, stack.main_grad_W2, acc)
        db1 = _C.colsum(dh_pre)
        db2 = _C.colsum(dy)
        if acc:
            stack.main_grad_b1.add_(db1)
            stack.main_grad_b2.add_(db2)
        else:
            stack.main_grad_b1.copy_(db1)
            stack.main_grad_b2.copy_(db2)
        stack.grad_ready = True
        return dxbuf, None


@torch.no_grad()
def flash_attention_fwd(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor) -> torch.Tensor:
    """Causal attention forward, [B, H, S, 64] bf16 (inference entry point).
    On GPU this is the hand-written CDNA4 flash kernel (ops/hip/attention.hip:
    online softmax, MFMA QK^T and PV, one wave per 16 query rows); elsewhere
    it falls back to torch sdpa with identical causal semantics."""
    if q.is_cuda and HAS_HIP and q.dtype == torch.bfloat16 and q.shape[-1] == 64 \
            and q.shape[-2] % 128 == 0:
        return _C.flash_attn_fwd(q.contiguous(), k.contiguous(), v.contiguous())
    return F.scaled_dot_product_attention(q, k, v, is_causal=True)


class _FlashAttention(torch.autograd.Function):
    """Differentiable causal flash attention (EXPERIMENTAL — numerics-verified,
    not yet wired into the training model; perf tiling is round-2 work).
    Backward is recompute-based from the saved log-sum-exp: two kernels
    (dQ over query blocks; dK/dV over key blocks), no atomics; the row term
    D = rowsum(dO*O) is computed here in fp32 torch (one cheap reduction)."""

    @staticmethod
    def forward(ctx, q, k, v):
        q, k, v = q.contiguous(), k.contiguous(), v.contiguous()
        o, lse = _C.flash_attn_fwd_lse(q, k, v)
        ctx.save_for_backward(q, k, v, o, lse)
        return o

    @staticmethod
    def backward(ctx, dout):
        q, k, v, o, lse = ctx.saved_tensors
        dout = dout.contiguous()
        d = (dout.float() * o.float()).sum(dim=-1)  # [B, H, S] fp32
        dq, dk, dv = _C.flash_attn_bwd(q, k, v, dout, lse, d)
        return dq, dk, dv


def flash_attention(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor) -> torch.Tensor:
    """Differentiable causal attention. CDNA4 flash kernels when the shapes
    qualify ([B, H, S, 64] bf16 on GPU, S % 64 == 0), torch sdpa otherwise."""
    if q.is_cuda and HAS_HIP and q.dtype == torch.bfloat16 and q.shape[-1] == 64 \
            and q.shape[-2] % 128 == 0:
        return _FlashAttention.apply(q, k, v)
    return F.scaled_dot_product_attention(q, k, v, is_causal=True)


@torch.no_grad()
def fused_ffn_infer(x: torch.Tensor, fc1_weight: torch.Tensor, fc1_bias: torch.Tensor,
                    fc2_weight: torch.Tensor, fc2_bias: torch.Tensor) -> torch.Tensor:
    """Inference-only fused FFN for ONE expert (the serving hot path:
    ExpertBackend.forward on a GPU server): y = gelu(x@W1^T + b1)@W2^T + b2
    as two MFMA NT launches with fused bias / bias+GELU epilogues — no
    transposed weight copies needed (nn.Linear stores W[out, in], which is
    exactly the NT layout). Requires bf16 CUDA tensors with
    d_model % 128 == 0 and ffn_dim % 128 == 0 (callers gate on
    `can_fuse_ffn_infer`); arbitrary token count M (64-row tail tiles)."""
    orig_shape = x.shape
    d = orig_shape[-1]
    ffn = fc1_weight.shape[0]
    x2 = x.reshape(1, -1, d).contiguous()
    M = x2.shape[1]
    # EPI_BIAS_GELU also emits gelu'(pre) for training backward; at inference
    # it is scratch (adding a no-derivative epilogue is not worth a variant)
    der = torch.empty(1, M, ffn, dtype=torch.bfloat16, device=x.device)
    h = _C.batched_gemm_nt(x2, fc1_weight.reshape(1, ffn, d).contiguous(),
                           fc1_bias.reshape(1, ffn).float().contiguous(), der, EPI_BIAS_GELU)
    y = _C.batched_gemm_nt(h, fc2_weight.reshape(1, d, ffn).contiguous(),
                           fc2_bias.reshape(1, d).float().contiguous(), None, EPI_BIAS)
    return y.reshape(orig_shape)


def can_fuse_ffn_infer(x: torch.Tensor, expert) -> bool:
    """True iff `fused_ffn_infer` applies: a gelu FFNExpert, everything bf16
    on GPU, dims aligned for the NT kernels (N%128, K%64 on both GEMMs)."""
    if not (HAS_HIP and x.is_cuda and x.dtype == torch.bfloat16 and x.numel() > 0):
        return False
    fc1 = getattr(expert, "fc1", None)
    fc2 = getattr(expert, "fc2", None)
    if fc1 is None or fc2 is None or getattr(expert, "activation", None) != "gelu":
        return False
    d, ffn = fc1.weight.shape[1], fc1.weight.shape[0]
    return (fc1.weight.dtype == torch.bfloat16 and fc2.weight.dtype == torch.bfloat16
            and fc1.bias is not None and fc2.bias is not None
            and d % 128 == 0 and ffn % 128 == 0 and x.shape[-1] == d)


def batched_ffn(xbuf: torch.Tensor, stack) -> torch.Tensor:
    """xbuf [E, C, D] -> y [E, C, D] through each expert's FFN."""
    if xbuf.is_cuda and HAS_HIP:
        return _BatchedFFNHip.apply(xbuf, stack)
    # CPU fallback: differentiable torch path over the fp32 master weights
    h_pre = torch.einsum("ecd,efd->ecf", xbuf.float(), stack.W1.float()) + stack.W1_bias.float()[:, None, :]
    h = F.gelu(h_pre, approximate="tanh")
    y = torch.einsum("ecf,edf->ecd", h, stack.W2.float()) + stack.W2_bias.float()[:, None, :]
    return y.to(xbuf.dtype)


# ---------------------------------------------------------------------------
# numerics references (fp32, used by GPU kernel tests)
# ---------------------------------------------------------------------------
def expert_ffn_reference(x, W1, b1, W2, b2):
    """fp32 reference of one expert FFN: gelu_tanh(x@W1^T+b1)@W2^T+b2."""
    h = F.gelu(x.float() @ W1.float().t() + b1.float(), approximate="tanh")
    return h @ W2.float().t() + b2.float()
