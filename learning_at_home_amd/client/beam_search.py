"""MoEBeamSearcher — DHT-driven beam search over a multi-dimensional expert
uid grid (SURVEY.md §2 "RemoteMixtureOfExperts" step 2 / hivemind's
MoEBeamSearcher): find the best ACTIVE experts without enumerating the full
grid, by expanding per-dimension score-ordered prefixes and filtering each
beam level through DHT `first_k_active`.

Used by RemoteMixtureOfExperts when the grid is too large to enumerate
candidates locally (prod of grid dims >> beam width); for small/flat grids
the local-beam + get_experts path in moe.py is cheaper.
"""
from __future__ import annotations

from typing import List, Sequence, Tuple

import torch

from ..dht import DHT, UID_DELIMITER
from ..utils import Endpoint, get_logger

logger = get_logger(__name__)


class MoEBeamSearcher:
    def __init__(self, dht: DHT, uid_prefix: str, grid_size: Sequence[int],
                 beam_size: int = 8):
        self.dht = dht
        self.uid_prefix = uid_prefix
        self.grid_size = tuple(grid_size)
        self.beam_size = beam_size

    def find_best_experts(self, grid_scores: Sequence[torch.Tensor], k_best: int
                          ) -> List[Tuple[str, Endpoint, float]]:
        """grid_scores: per-dimension 1-D score vectors (ONE sample).
        Returns up to k_best ACTIVE experts as (uid, endpoint, score),
        best-first. Each beam level keeps `beam_size` score-ordered prefixes
        and drops those with no live expert underneath (DHT prefix probe)."""
        assert len(grid_scores) == len(self.grid_size)
        beam: List[Tuple[float, str]] = [(0.0, self.uid_prefix)]
        for dim, scores in enumerate(grid_scores):
            order = torch.argsort(scores, descending=True)
            candidates: List[Tuple[float, str]] = []
            for beam_score, prefix in beam:
                for idx in order[: self.beam_size].tolist():
                    candidates.append(
                        (beam_score + float(scores[idx]), f"{prefix}{UID_DELIMITER}{idx}")
                    )
            candidates.sort(key=lambda t: -t[0])
            is_last = dim == len(self.grid_size) - 1
            if is_last:
                # resolve leaves to live endpoints
                uids = [uid for _, uid in candidates[: max(self.beam_size, k_best) * 2]]
                endpoints = self.dht.get_experts(uids)
                out = []
                for (score, uid), ep in zip(candidates, endpoints):
                    if ep is not None:
                        out.append((uid, ep, score))
                        if len(out) >= k_best:
                            break
                return out
            # prune to live prefixes, keep order
            prefixes = [uid for _, uid in candidates[: self.beam_size * 2]]
            active = self.dht.first_k_active(prefixes, k=self.beam_size)
            beam = [(score, uid) for score, uid in candidates if uid in active][: self.beam_size]
            if not beam:
                logger.warning(f"beam search: no active prefixes at dim {dim}")
                return []
        return []
