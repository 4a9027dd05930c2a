"""MasterSGD — SGD with fp32 master copies for bf16 model parameters.

Plain torch SGD on bf16 parameters silently loses any update smaller than
the parameter's bf16 ulp (~0.4% relative), which freezes large-d trunks at
practical learning rates. This keeps an fp32 master per parameter, steps in
fp32, and re-quantizes the bf16 parameter — the trunk-side analog of the
experts' fused master-SGD (parallel/FusedExpertSGD)."""
from __future__ import annotations

from typing import Iterable

import torch


class MasterSGD:
    def __init__(self, params: Iterable[torch.nn.Parameter], lr: float = 0.01,
                 momentum: float = 0.0, weight_decay: float = 0.0):
        self.params = [p for p in params if p.requires_grad]
        self.lr, self.momentum, self.weight_decay = lr, momentum, weight_decay
        self.masters = [p.detach().float().clone() for p in self.params]
        self.momentum_bufs = (
            [torch.zeros_like(m) for m in self.masters] if momentum > 0 else None
        )

    @torch.no_grad()
    def step(self, closure=None) -> None:
        for i, p in enumerate(self.params):
            if p.grad is None:
                continue
            g = p.grad.float()
            m = self.masters[i]
            if self.weight_decay:
                g = g.add(m, alpha=self.weight_decay)
            if self.momentum_bufs is not None:
                buf = self.momentum_bufs[i]
                buf.mul_(self.momentum).add_(g)
                g = buf
            m.add_(g, alpha=-self.lr)
            # single fused cast-copy: m.to(dtype) + copy_ was an extra full
            # DtoD pass per param per step (visible as ~20 ms/step of
            # copyBuffer when overlapped with the SR optimizer stream)
            p.copy_(m)

    @torch.no_grad()
    def sync_masters_from_params(self) -> None:
        """Re-seed masters from the (bf16) parameters — call after an
        external in-place update of the params (e.g. an averaging round)."""
        for p, m in zip(self.params, self.masters):
            m.copy_(p)

    def zero_grad(self, set_to_none: bool = True) -> None:
        for p in self.params:
            if set_to_none:
                p.grad = None
            elif p.grad is not None:
                p.grad.zero_()

    def state_dict(self):
        return {"masters": self.masters, "momentum": self.momentum_bufs}

    def load_state_dict(self, sd):
        for m, src in zip(self.masters, sd["masters"]):
            m.copy_(src)
        if sd.get("momentum") and self.momentum_bufs:
            for b, src in zip(self.momentum_bufs, sd["momentum"]):
                b.copy_(src)
