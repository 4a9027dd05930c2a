#!/usr/bin/env python3
"""Bisect GPU-vs-CPU training dynamics: tiny Transformer-DMoE memorization on
cuda:0 under (a) fp32-master fused SGD, (b) stochastic-rounding SGD, and (c)
torch-optimizer on the CPU reference path moved to GPU trunk... Run via gpurun."""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))

import torch

from learning_at_home_amd.models import TransformerDMoE, TransformerDMoEConfig
from learning_at_home_amd.parallel import ExpertParallelMoE, FusedExpertSGD


def run(mode: str, steps: int = 120, lr: float = 0.01, scale: str = "tiny"):
    torch.manual_seed(0)
    if scale == "flagship":
        cfg = TransformerDMoEConfig(vocab_size=32000, d_model=1024, n_heads=16,
                                    n_layers=6, seq_len=1024, num_experts=256,
                                    k_best=4, ffn_dim=4096)
    else:
        cfg = TransformerDMoEConfig(vocab_size=512, d_model=128, n_heads=4, n_layers=2,
                                    seq_len=64, num_experts=16, k_best=4, ffn_dim=256)

    aux_c = 0.0 if "noaux" in mode else 0.01
    z_c = 0.0 if "noz" in mode else 1e-3
    noise = 0.0 if "nonoise" in mode else 0.3

    def moe_factory(i, c):
        return ExpertParallelMoE(d_model=c.d_model, ffn_dim=c.ffn_dim,
                                 num_experts=c.num_experts, k=c.k_best,
                                 capacity_factor=1.5, aux_loss_coef=aux_c,
                                 z_loss_coef=z_c, router_noise_std=noise)

    model = TransformerDMoE(cfg, moe_factory).to("cuda:0", torch.bfloat16)
    sr = "sr" in mode
    expert_opts = []
    for block in model.blocks:
        block.moe.experts.float()
        block.moe.experts.init_gpu_state(master_weights=not sr)
        expert_opts.append(FusedExpertSGD(block.moe.experts, lr=lr,
                                          stochastic_rounding=sr))
    from learning_at_home_amd.optim import MasterSGD

    trunk = [p for n, p in model.named_parameters() if ".moe.experts." not in n]
    topt = MasterSGD(trunk, lr=lr) if mode != "bf16trunk" else torch.optim.SGD(trunk, lr=lr)
    B = 4 if scale == "tiny" else 2
    ids = torch.randint(0, cfg.vocab_size, (B, cfg.seq_len + 1), device="cuda:0")
    print(f"=== mode {mode} scale {scale} lr {lr} ===", flush=True)
    import torch.nn.functional as F
    for step in range(steps):
        logits = model(ids[:, :-1])
        ce = F.cross_entropy(logits.flatten(0, 1).float(), ids[:, 1:].flatten())
        aux = sum(b.moe.aux_loss for b in model.blocks if b.moe.aux_loss is not None)
        loss = ce + aux
        topt.zero_grad()
        loss.backward()
        for o in expert_opts:
            o.step()
        topt.step()
        if step % 10 == 0 or step == steps - 1:
            counts = model.blocks[0].moe.last_counts
            gate = model.blocks[0].moe.gate.weight
            gnorm = float(gate.grad.float().norm()) if gate.grad is not None else -1
            with torch.no_grad():
                lmax = float(gate.float().abs().max())
            print(f"step {step:3d} ce {float(ce):.4f} aux {float(aux):.4f} "
                  f"cmax {int(counts.max())} cmin {int(counts.min())} "
                  f"|gate.grad| {gnorm:.3e} |gate|max {lmax:.3e}", flush=True)


if __name__ == "__main__":
    scale = "tiny"
    steps = 120
    modes = []
    for a in sys.argv[1:]:
        if a in ("tiny", "flagship"):
            scale = a
        elif a.isdigit():
            steps = int(a)
        else:
            modes.append(a)
    for mode in modes or ["master", "sr"]:
        run(mode, steps=steps, scale=scale)
