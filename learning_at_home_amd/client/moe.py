"""RemoteMixtureOfExperts — DMoE client layer (SURVEY.md §2
"RemoteMixtureOfExperts", §3.1 hot path).

Semantics preserved from the reference:
1. gating: one Linear produces per-grid-dimension logits; a full uid
   "prefix.i.j" scores as the sum of its per-dimension logits;
2. per-token top-k expert selection, filtered through the DHT for experts
   that are ACTIVE right now (liveness lease);
3. parallel dispatch with a quorum: proceed when >= k_min experts responded
   within the timeout, DROP the rest;
4. combine: softmax over the logits of the experts that actually responded
   (renormalization over survivors), weighted sum of their outputs;
5. backward mirrors 3-4: parallel "bwd" RPCs, dead experts contribute zero.

MI355X: this socket-RPC path is the cross-host / parity mode. The intra-node
hot path implements the same math with RCCL all-to-all + CDNA4 kernels in
parallel/expert_shard.py. The per-token selection logic here batches all DHT
probes per forward (one get_experts over the candidate union, cached for
`active_cache_ttl`) instead of per-sample DHT walks — same observable
semantics, control-plane cost O(unique candidates) not O(batch).
"""
from __future__ import annotations

import time
from concurrent.futures import FIRST_COMPLETED, ThreadPoolExecutor, wait as futures_wait
from typing import Callable, Dict, List, Optional, Sequence, Tuple

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..dht import DHT, UID_DELIMITER
from ..utils import Endpoint, get_logger
from .expert import rpc_call

logger = get_logger(__name__)


class RemoteMixtureOfExperts(nn.Module):
    def __init__(
        self,
        *,
        in_features: int,
        grid_size: Sequence[int],
        dht: DHT,
        uid_prefix: str,
        k_best: int,
        k_min: int = 1,
        forward_timeout: float = 10.0,
        backward_timeout: float = 10.0,
        backward_k_min: int = 1,
        timeout_after_k_min: float = 1.0,
        active_cache_ttl: float = 10.0,
        candidate_factor: int = 4,
        aux_loss_coef: float = 0.0,
    ):
        super().__init__()
        self.grid_size = tuple(grid_size)
        self.dht = dht
        self.uid_prefix = uid_prefix
        self.k_best, self.k_min, self.backward_k_min = k_best, k_min, backward_k_min
        self.forward_timeout, self.backward_timeout = forward_timeout, backward_timeout
        self.timeout_after_k_min = timeout_after_k_min
        self.active_cache_ttl = active_cache_ttl
        self.candidate_factor = candidate_factor
        self.aux_loss_coef = aux_loss_coef
        self.proj = nn.Linear(in_features, sum(self.grid_size))
        self._active_cache: Dict[str, Optional[Endpoint]] = {}
        self._active_cache_time = -float("inf")
        self.aux_loss: Optional[torch.Tensor] = None
        # observability: cumulative backward-RPC quorum
        # accounting — experts that returned grads vs dropped, and how many
        # backwards missed backward_k_min entirely (grads silently partial)
        self.backward_stats = {"experts_ok": 0, "experts_dropped": 0, "quorum_missed": 0}

    # ------------------------------------------------------------------
    def _candidate_uids(self, grid_logits: List[torch.Tensor]) -> Tuple[List[List[str]], torch.Tensor]:
        """Local beam over grid dims: per sample, the top
        (candidate_factor * k_best) full uids by summed per-dim logits.
        Returns (uid rows, scores [N, C])."""
        beam_width = min(self.candidate_factor * self.k_best,
                         int(torch.tensor(self.grid_size).prod()))
        N = grid_logits[0].shape[0]
        # beam state: scores [N, B], index tuples per sample
        scores = grid_logits[0]  # [N, grid_size[0]]
        topv, topi = scores.topk(min(beam_width, scores.shape[1]), dim=-1)
        beams: List[List[Tuple[int, ...]]] = [[(int(i),) for i in topi[n]] for n in range(N)]
        beam_scores = topv
        for dim in range(1, len(self.grid_size)):
            dim_logits = grid_logits[dim]  # [N, grid_size[dim]]
            dim_top_v, dim_top_i = dim_logits.topk(min(beam_width, dim_logits.shape[1]), dim=-1)
            combined = beam_scores.unsqueeze(-1) + dim_top_v.unsqueeze(1)  # [N, B, B']
            flat = combined.flatten(1)
            topv, topi = flat.topk(min(beam_width, flat.shape[1]), dim=-1)
            new_beams = []
            for n in range(N):
                row = []
                for flat_idx in topi[n].tolist():
                    b, bp = divmod(flat_idx, dim_top_i.shape[1])
                    row.append(beams[n][b] + (int(dim_top_i[n, bp]),))
                new_beams.append(row)
            beams, beam_scores = new_beams, topv
        uid_rows = [
            [self.uid_prefix + UID_DELIMITER + UID_DELIMITER.join(map(str, idxs)) for idxs in row]
            for row in beams
        ]
        return uid_rows, beam_scores

    def _resolve_active(self, uids: Sequence[str]) -> Dict[str, Optional[Endpoint]]:
        now = time.monotonic()
        if now - self._active_cache_time > self.active_cache_ttl:
            self._active_cache.clear()
            self._active_cache_time = now
        missing = [u for u in uids if u not in self._active_cache]
        if missing:
            endpoints = self.dht.get_experts(missing)
            for uid, ep in zip(missing, endpoints):
                self._active_cache[uid] = ep
        return {u: self._active_cache.get(u) for u in uids}

    # ------------------------------------------------------------------
    def forward(self, x: torch.Tensor) -> torch.Tensor:
        lead_shape, d = x.shape[:-1], x.shape[-1]
        flat = x.reshape(-1, d)
        N = flat.shape[0]
        all_logits = self.proj(flat)
        grid_logits = list(all_logits.split(list(self.grid_size), dim=-1))

        uid_rows, cand_scores = self._candidate_uids(grid_logits)
        unique_uids = sorted({u for row in uid_rows for u in row})
        active = self._resolve_active(unique_uids)

        # per sample: top-k_best ACTIVE candidates
        sel_uids: List[List[str]] = []
        sel_cols: List[List[int]] = []
        for n in range(N):
            chosen, cols = [], []
            for c, uid in enumerate(uid_rows[n]):
                if active.get(uid) is not None:
                    chosen.append(uid)
                    cols.append(c)
                    if len(chosen) >= self.k_best:
                        break
            if not chosen:
                raise RuntimeError(
                    f"RemoteMixtureOfExperts: no active experts among "
                    f"{len(uid_rows[n])} candidates (prefix={self.uid_prefix!r})"
                )
            sel_uids.append(chosen)
            sel_cols.append(cols)
        k_max = max(len(row) for row in sel_uids)
        sel_logits = torch.full((N, k_max), float("-inf"), dtype=cand_scores.dtype,
                                device=cand_scores.device)
        for n in range(N):
            sel_logits[n, : len(sel_cols[n])] = cand_scores[n, sel_cols[n]]

        endpoint_of = {u: active[u] for row in sel_uids for u in row}
        y = _RemoteMoECallBatched.apply(
            flat, sel_logits, sel_uids, endpoint_of, self.k_min,
            self.forward_timeout, self.backward_timeout, self.backward_k_min,
            self.timeout_after_k_min, self.backward_stats,
        )

        if self.aux_loss_coef > 0:
            self.aux_loss = self.aux_loss_coef * _importance_aux_loss(grid_logits)
        else:
            self.aux_loss = None
        return y.reshape(*lead_shape, -1)


def _importance_aux_loss(grid_logits: List[torch.Tensor]) -> torch.Tensor:
    """Shazeer-style importance penalty: squared coefficient of variation of
    per-expert-index softmax mass, summed over grid dimensions."""
    total = 0.0
    for logits in grid_logits:
        probs = F.softmax(logits, dim=-1)
        importance = probs.sum(0)
        cv_sq = importance.var(unbiased=False) / (importance.mean() ** 2 + 1e-10)
        total = total + cv_sq
    return total


def _wait_quorum_early_exit(
    futures: Dict[str, "object"],
    on_success: Callable[[str, object], None],
    quorum_met: Callable[[], bool],
    total_timeout: float,
    grace: float,
    direction: str,
) -> None:
    """The reference's `timeout_after_k_min` semantics: wait FIRST_COMPLETED
    in a loop; as soon as `quorum_met()` (k_min replies per token), start a
    `grace`-second countdown; when it (or `total_timeout`) expires, abandon
    the stragglers instead of waiting the full timeout. `on_success(uid,
    result)` runs for each completed-without-exception future, *before*
    quorum is re-evaluated. (SURVEY §2 RemoteMixtureOfExperts row (3);
    VERDICT r1 missing #2.)"""
    fut_uid = {f: uid for uid, f in futures.items()}
    pending = set(futures.values())
    deadline = time.monotonic() + total_timeout
    grace_deadline: Optional[float] = None
    while pending:
        eff = deadline if grace_deadline is None else min(deadline, grace_deadline)
        remaining = eff - time.monotonic()
        if remaining <= 0:
            break
        done, pending = futures_wait(pending, timeout=remaining, return_when=FIRST_COMPLETED)
        for fut in done:
            uid = fut_uid[fut]
            if fut.exception() is None:
                on_success(uid, fut.result())
            else:
                logger.warning(f"{direction} RPC to expert {uid} failed ({fut.exception()!r}); dropping it")
        if grace_deadline is None and quorum_met():
            grace_deadline = time.monotonic() + grace
    for fut in pending:
        fut.cancel()
        logger.warning(
            f"{direction} RPC to expert {fut_uid[fut]} dropped "
            f"({'straggler past k_min quorum + grace' if grace_deadline is not None else 'timed out'})"
        )


class _RemoteMoECallBatched(torch.autograd.Function):
    """Batched dispatch/combine with quorum fault tolerance and parallel
    backward RPCs (manual grads: softmax-over-survivors combine)."""

    @staticmethod
    def forward(ctx, x: torch.Tensor, sel_logits: torch.Tensor, sel_uids: List[List[str]],
                endpoint_of: Dict[str, Endpoint], k_min: int,
                forward_timeout: float, backward_timeout: float,
                backward_k_min: int = 1, timeout_after_k_min: float = 1.0,
                backward_stats: Optional[dict] = None):
        N, d = x.shape
        k_max = sel_logits.shape[1]
        x_cpu = x.detach().cpu()

        # group (token, slot) pairs per expert uid
        expert_tokens: Dict[str, List[Tuple[int, int]]] = {}
        for n, row in enumerate(sel_uids):
            for slot, uid in enumerate(row):
                expert_tokens.setdefault(uid, []).append((n, slot))

        outputs: Dict[str, torch.Tensor] = {}
        token_got = [0] * N
        token_need = [min(k_min, len(row)) for row in sel_uids]

        def _on_success(uid: str, result) -> None:
            outputs[uid] = result[0]
            for n, _slot in expert_tokens[uid]:
                token_got[n] += 1

        def _quorum_met() -> bool:
            return all(g >= t for g, t in zip(token_got, token_need))

        # NOT a with-block: __exit__ would wait for straggler RPC threads,
        # defeating the early exit. Abandoned threads finish on their own
        # socket timeout (success re-pools the socket; failure closes it).
        pool = ThreadPoolExecutor(max_workers=min(32, len(expert_tokens)))
        try:
            futures = {}
            for uid, pairs in expert_tokens.items():
                idx = [n for n, _ in pairs]
                futures[uid] = pool.submit(
                    rpc_call, endpoint_of[uid], "fwd", [uid, [x_cpu[idx]]], forward_timeout
                )
            _wait_quorum_early_exit(futures, _on_success, _quorum_met,
                                    forward_timeout, timeout_after_k_min, "forward")
        finally:
            pool.shutdown(wait=False)

        if not outputs:
            raise RuntimeError("RemoteMixtureOfExperts: every selected expert failed")
        d_out = next(iter(outputs.values())).shape[-1]
        o = torch.zeros(N, k_max, d_out, dtype=x.dtype)
        responded = torch.zeros(N, k_max, dtype=torch.bool)
        for uid, out in outputs.items():
            for local_i, (n, slot) in enumerate(expert_tokens[uid]):
                o[n, slot] = out[local_i].to(o.dtype)
                responded[n, slot] = True

        quorum = responded.sum(1)
        if (quorum < k_min).any():
            bad = int((quorum < k_min).sum())
            raise RuntimeError(
                f"RemoteMixtureOfExperts: {bad}/{N} tokens got fewer than k_min={k_min} expert replies"
            )

        masked_logits = sel_logits.detach().cpu().float().masked_fill(~responded, float("-inf"))
        w = F.softmax(masked_logits, dim=-1)  # softmax over survivors
        w = torch.nan_to_num(w, nan=0.0)
        y = torch.einsum("nk,nkd->nd", w.to(o.dtype), o)

        ctx.save_for_backward(x_cpu, o, w, responded)
        ctx.expert_tokens = expert_tokens
        ctx.responded_uids = set(outputs.keys())
        ctx.endpoint_of = endpoint_of
        ctx.backward_timeout = backward_timeout
        ctx.backward_k_min = backward_k_min
        ctx.timeout_after_k_min = timeout_after_k_min
        ctx.backward_stats = backward_stats
        ctx.device = x.device
        return y.to(x.device)

    @staticmethod
    def backward(ctx, grad_y: torch.Tensor):
        x_cpu, o, w, responded = ctx.saved_tensors
        g = grad_y.detach().cpu().to(o.dtype)
        N, k_max, d_out = o.shape

        # grads wrt gating logits through softmax-over-survivors
        dot = torch.einsum("nkd,nd->nk", o.float(), g.float())  # <o_nk, g_n>
        weighted = (w * dot).sum(-1, keepdim=True)
        dlogits = (w * (dot - weighted)).masked_fill(~responded, 0.0)

        # grads wrt x: parallel bwd RPCs, one per expert that responded in fwd;
        # same early-exit quorum as forward, but counted per EXPERT
        # (backward_k_min experts returned grads), since failed experts
        # contribute zero grads rather than failing the step
        dx = torch.zeros_like(x_cpu, dtype=torch.float32)
        live_experts = [uid for uid in ctx.expert_tokens if uid in ctx.responded_uids]
        ok_count = [0]
        idx_of: Dict[str, List[int]] = {}

        def _on_success(uid: str, result) -> None:
            dx[idx_of[uid]] += result[0].float()
            ok_count[0] += 1

        def _quorum_met() -> bool:
            return ok_count[0] >= min(ctx.backward_k_min, len(live_experts))

        pool = ThreadPoolExecutor(max_workers=min(32, max(1, len(live_experts))))
        try:
            futures = {}
            for uid in live_experts:
                pairs = ctx.expert_tokens[uid]
                idx = [n for n, _ in pairs]
                slots = [s for _, s in pairs]
                idx_of[uid] = idx
                grad_out = w[idx, slots].unsqueeze(1).to(g.dtype) * g[idx]
                futures[uid] = pool.submit(
                    rpc_call, ctx.endpoint_of[uid], "bwd",
                    [uid, [x_cpu[idx], grad_out]], ctx.backward_timeout,
                )
            _wait_quorum_early_exit(futures, _on_success, _quorum_met,
                                    ctx.backward_timeout, ctx.timeout_after_k_min, "backward")
        finally:
            pool.shutdown(wait=False)

        missed = ok_count[0] < min(ctx.backward_k_min, len(live_experts))
        if missed:
            # reference semantics: proceed anyway (failed experts contribute
            # zero grads), but surface that the backward quorum was missed
            logger.warning(
                f"backward quorum missed: {ok_count[0]}/{len(live_experts)} experts "
                f"returned grads (backward_k_min={ctx.backward_k_min})"
            )
        if ctx.backward_stats is not None:
            ctx.backward_stats["experts_ok"] += ok_count[0]
            ctx.backward_stats["experts_dropped"] += len(live_experts) - ok_count[0]
            ctx.backward_stats["quorum_missed"] += int(missed)

        return (dx.to(ctx.device, dtype=grad_y.dtype),
                dlogits.to(ctx.device), None, None, None, None, None, None, None, None)
