#!/usr/bin/env python3
"""Microbenchmarks of the CDNA4 kernel suite at flagship-bench shapes.
Run on an MI355X: `gpurun -- python experiments/bench_kernels.py`."""
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))

import torch

from learning_at_home_amd.ops import _C

DEV = "cuda:0"


def timeit(fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def bench_gemm_nt(E, M, N, K, epi, tag):
    A = torch.randn(E, M, K, device=DEV).bfloat16()
    B = torch.randn(E, N, K, device=DEV).bfloat16()
    bias = torch.randn(E, N, device=DEV) if epi in (1, 2) else None
    pre = torch.empty(E, M, N, dtype=torch.bfloat16, device=DEV) if epi in (2, 3) else None
    if epi == 3:
        pre.normal_()
    dt = timeit(lambda: _C.batched_gemm_nt(A, B, bias, pre, epi))
    tf = 2.0 * E * M * N * K / dt / 1e12
    print(f"gemm_nt[{tag}] E{E} {M}x{N}x{K} epi{epi}: {dt*1e3:7.3f} ms  {tf:7.1f} TF")


def bench_gemm_tn(E, M, I, J, tag):
    A = torch.randn(E, M, I, device=DEV).bfloat16()
    B = torch.randn(E, M, J, device=DEV).bfloat16()
    C = torch.zeros(E, I, J, device=DEV)
    dt = timeit(lambda: _C.batched_gemm_tn(A, B, C, False))
    tf = 2.0 * E * M * I * J / dt / 1e12
    print(f"gemm_tn[{tag}] E{E} m{M} {I}x{J}: {dt*1e3:7.3f} ms  {tf:7.1f} TF")


def bench_sgd_t(E, R, C, tag):
    master = torch.randn(E, R, C, device=DEV)
    grad = torch.randn(E, R, C, device=DEV)
    out = torch.empty(E, R, C, dtype=torch.bfloat16, device=DEV)
    outT = torch.empty(E, C, R, dtype=torch.bfloat16, device=DEV)
    dt = timeit(lambda: _C.sgd_step_t(master, grad, out, outT, 0.01, 0.0))
    n = E * R * C
    gbps = n * 16.0 / dt / 1e9  # r m4+g4, w m4+bf2+bfT2
    print(f"sgd_t[{tag}] {n/1e6:.0f}M params: {dt*1e3:7.3f} ms  {gbps:7.0f} GB/s eff")


def bench_attention(B, H, S, tag):
    """Flash attention vs torch sdpa at bench shapes, fwd and fwd+bwd.
    Causal flops: QK^T + PV = 2 * 2 * BH * (S^2/2) * D per fwd pass."""
    import torch.nn.functional as F

    from learning_at_home_amd.ops.functional import flash_attention

    D = 64
    flops_fwd = 2.0 * 2 * B * H * (S * S / 2) * D
    q, k, v = (torch.randn(B, H, S, D, device=DEV, dtype=torch.bfloat16) for _ in range(3))

    dt = timeit(lambda: _C.flash_attn_fwd(q, k, v), iters=30)
    print(f"attn_flash_fwd[{tag}]: {dt*1e3:7.3f} ms  {flops_fwd/dt/1e12:7.1f} TF")
    dt = timeit(lambda: F.scaled_dot_product_attention(q, k, v, is_causal=True), iters=30)
    print(f"attn_sdpa_fwd [{tag}]: {dt*1e3:7.3f} ms  {flops_fwd/dt/1e12:7.1f} TF")

    def fb(fn):
        qkv = [t.detach().clone().requires_grad_(True) for t in (q, k, v)]
        out = fn(*qkv)
        out.sum().backward()

    flops_fb = flops_fwd * 3.5  # bwd ~ 2.5x fwd (dq, dkv recompute)
    dt = timeit(lambda: fb(flash_attention), iters=15)
    print(f"attn_flash_f+b[{tag}]: {dt*1e3:7.3f} ms  {flops_fb/dt/1e12:7.1f} TF")
    dt = timeit(lambda: fb(lambda a, b, c: F.scaled_dot_product_attention(a, b, c, is_causal=True)), iters=15)
    print(f"attn_sdpa_f+b [{tag}]: {dt*1e3:7.3f} ms  {flops_fb/dt/1e12:7.1f} TF")


def main():
    torch.manual_seed(0)
    print("=== attention (flagship: B=96 H=16 S=1024 D=64) ===")
    bench_attention(96, 16, 1024, "flagship")
    print("=== flagship shapes (E=256, C=192, d=1024, ffn=4096) ===")
    bench_gemm_nt(256, 192, 4096, 1024, 2, "fwd1")
    bench_gemm_nt(256, 192, 1024, 4096, 1, "fwd2")
    bench_gemm_nt(256, 192, 4096, 1024, 3, "dh")
    bench_gemm_nt(256, 192, 1024, 4096, 0, "dx")
    bench_gemm_tn(256, 192, 4096, 1024, "dW1")
    bench_gemm_tn(256, 192, 1024, 4096, "dW2")
    bench_sgd_t(256, 4096, 1024, "W1")
    print("=== big-M reference (dense-like) ===")
    bench_gemm_nt(1, 4096, 4096, 4096, 0, "4k^3")
    bench_gemm_tn(1, 4096, 4096, 4096, "4k^3")
    print("=== EP-8 shapes (E_local=32, M=W*C=1536) ===")
    bench_gemm_nt(32, 1536, 4096, 1024, 2, "fwd1@8gpu")
    bench_gemm_tn(32, 1536, 4096, 1024, "dW1@8gpu")


if __name__ == "__main__":
    main()
