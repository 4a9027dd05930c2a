#!/usr/bin/env python3
"""Minimal attention-kernel driver for rocprofv3 PMC runs (fwd + bwd only,
flagship shape). Keep iteration counts tiny: every dispatch is profiled."""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))

import torch

from learning_at_home_amd.ops import _C

DEV = "cuda:0"


def main():
    torch.manual_seed(0)
    B, H, S, D = 96, 16, 1024, 64
    q, k, v = (torch.randn(B, H, S, D, device=DEV, dtype=torch.bfloat16) for _ in range(3))
    o, lse = _C.flash_attn_fwd_lse(q, k, v)
    do = torch.randn_like(o)
    d = (do.float() * o.float()).sum(dim=-1)
    for _ in range(3):
        _C.flash_attn_fwd(q, k, v)
    for _ in range(3):
        _C.flash_attn_bwd(q, k, v, do, lse, d)
    torch.cuda.synchronize()


if __name__ == "__main__":
    main()
