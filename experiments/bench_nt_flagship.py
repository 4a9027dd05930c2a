#!/usr/bin/env python3
"""NT GEMM at the REAL flagship in-step shapes (M = capacity = 2304), for
rocprofv3 PMC diagnosis of the K=1024 vs K=4096 efficiency gap."""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))

import time

import torch

from learning_at_home_amd.ops import _C

DEV = "cuda:0"


def run(E, M, N, K, epi, tag, iters=4):
    A = torch.randn(E, M, K, device=DEV).bfloat16()
    B = torch.randn(E, N, K, device=DEV).bfloat16()
    bias = torch.randn(E, N, device=DEV) if epi in (1, 2) else None
    pre = torch.empty(E, M, N, dtype=torch.bfloat16, device=DEV) if epi in (2, 3) else None
    if epi == 3:
        pre.normal_()
    _C.batched_gemm_nt(A, B, bias, pre, epi)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        _C.batched_gemm_nt(A, B, bias, pre, epi)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / iters
    print(f"{tag}: {dt*1e3:.3f} ms {2.0*E*M*N*K/dt/1e12:.1f} TF")


def main():
    torch.manual_seed(0)
    E, M = 64, 2304  # E=64 keeps memory sane; per-block work identical
    run(E, M, 4096, 1024, 2, "fwd1 K1024-N4096 epi2")
    run(E, M, 1024, 4096, 1, "fwd2 K4096-N1024 epi1")
    run(E, M, 4096, 1024, 3, "dh   K1024-N4096 epi3")
    run(E, M, 1024, 4096, 0, "dx   K4096-N1024 epi0")


if __name__ == "__main__":
    main()
