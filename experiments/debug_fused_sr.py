#!/usr/bin/env python3
"""Bitwise diagnostic for fused_dw_sgd_sr: emulate the kernel's hash-based
SR on the host and find WHERE the kernel's result deviates."""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))

import torch

from learning_at_home_amd.ops import _C

DEV = "cuda:0"


def hash_u32(x):
    x = x.to(torch.int64) & 0xFFFFFFFF
    x = x ^ (x >> 16)
    x = (x * 0x7FEB352D) & 0xFFFFFFFF
    x = x ^ (x >> 15)
    x = (x * 0x846CA68B) & 0xFFFFFFFF
    x = x ^ (x >> 16)
    return x


def round_sr_host(pv, rnd16):
    bits = pv.view(torch.int32).to(torch.int64) & 0xFFFFFFFF
    frac = bits & 0xFFFF
    trunc = bits & 0xFFFF0000
    up = frac > rnd16
    trunc = torch.where(up, trunc + 0x10000, trunc)
    hi = (trunc >> 16).to(torch.int32).to(torch.uint16 if hasattr(torch, "uint16") else torch.int16)
    return hi


def main():
    torch.manual_seed(17)
    E, M, I, J = 4, 192, 256, 512
    lr, wd, seed = 0.05, 0.01, 123
    A = (0.1 * torch.randn(E, M, I, device=DEV)).bfloat16()
    B = (0.1 * torch.randn(E, M, J, device=DEV)).bfloat16()
    W = torch.randn(E, I, J, device=DEV).bfloat16().contiguous()
    WT = W.transpose(1, 2).contiguous()
    Wf = W.clone()
    WTf = WT.clone()
    _C.fused_dw_sgd_sr(A, B, Wf, WTf, lr, wd, seed)

    g = torch.zeros(E, I, J, device=DEV)
    _C.batched_gemm_tn(A, B, g, False)
    pv = W.float() - lr * (g + wd * W.float())

    # emulate the kernel's rnd per element
    e_idx = torch.arange(E, device=DEV).view(E, 1, 1)
    row = torch.arange(I, device=DEV).view(1, I, 1)
    col = torch.arange(J, device=DEV).view(1, 1, J)
    chunk_off = (e_idx * I * J + row * J + (col & ~7)) & 0xFFFFFFFF
    h = hash_u32(chunk_off ^ seed)
    rnd16 = (hash_u32(h + (col & 7)) >> 8) & 0xFFFF
    exp_hi = round_sr_host(pv, rnd16).to(torch.int16)

    got_hi = Wf.view(torch.int16)
    mism = (exp_hi != got_hi)
    n = int(mism.sum())
    print(f"bitwise mismatches: {n} / {mism.numel()} ({n/mism.numel():.2%})")
    if n:
        idx = mism.nonzero()[:20]
        for e, r, c in idx.tolist():
            print(f"  e{e} row{r} col{c}: got {got_hi[e,r,c].item()&0xFFFF:04x} "
                  f"want {exp_hi[e,r,c].item()&0xFFFF:04x} pv {pv[e,r,c].item():.6f} "
                  f"w {W[e,r,c].float().item():.6f} g {g[e,r,c].item():.6f}")
        # structure of mismatch positions
        rs = mism.nonzero()[:, 1]
        cs = mism.nonzero()[:, 2]
        print("row%64 histogram:", torch.bincount(rs % 64, minlength=64).tolist())
        print("col%8 histogram:", torch.bincount(cs % 8, minlength=8).tolist())
    # WT consistency
    print("WT == W^T:", bool(torch.equal(WTf, Wf.transpose(1, 2).contiguous())))


if __name__ == "__main__":
    main()
