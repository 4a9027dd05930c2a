#!/usr/bin/env python3
"""Learnable-structure convergence (the reference's MNIST/WikiText-2 stand-in
under the no-network constraint): train the Transformer-DMoE on sequences from
a fixed random first-order Markov chain whose conditional entropy is KNOWN —
CE must drop from ~ln(V) (unigram floor) toward the chain's entropy H.

Usage: python experiments/synthetic_lm_train.py --steps 150
"""
import argparse
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))

import torch

from learning_at_home_amd.models import TransformerDMoE, TransformerDMoEConfig
from learning_at_home_amd.parallel import ExpertParallelMoE


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--steps", type=int, default=150)
    p.add_argument("--vocab", type=int, default=64)
    p.add_argument("--seq", type=int, default=64)
    p.add_argument("--batch", type=int, default=16)
    p.add_argument("--lr", type=float, default=0.05)
    p.add_argument("--peaked", type=float, default=3.0,
                   help="transition-logit scale; larger = lower-entropy chain")
    args = p.parse_args()

    torch.manual_seed(0)
    V = args.vocab
    trans = torch.softmax(args.peaked * torch.randn(V, V), dim=-1)  # P(next|cur)
    # exact conditional entropy under the stationary distribution
    ent_rows = -(trans * trans.clamp_min(1e-12).log()).sum(-1)
    pi = torch.full((V,), 1.0 / V)
    for _ in range(200):
        pi = pi @ trans
    H = float((pi * ent_rows).sum())

    def sample_batch():
        seqs = torch.empty(args.batch, args.seq + 1, dtype=torch.long)
        seqs[:, 0] = torch.randint(0, V, (args.batch,))
        for t in range(args.seq):
            seqs[:, t + 1] = torch.multinomial(trans[seqs[:, t]], 1).squeeze(-1)
        return seqs[:, :-1], seqs[:, 1:]

    cfg = TransformerDMoEConfig(vocab_size=V, d_model=64, n_heads=4, n_layers=2,
                                seq_len=args.seq, num_experts=8, k_best=2, ffn_dim=128)
    model = TransformerDMoE(cfg, lambda i, c: ExpertParallelMoE(
        d_model=c.d_model, ffn_dim=c.ffn_dim, num_experts=c.num_experts,
        k=c.k_best, router_noise_std=0.1))
    opt = torch.optim.Adam(model.parameters(), lr=args.lr * 0.1)

    import math

    print(f"chain conditional entropy H = {H:.4f} nats "
          f"(unigram floor ln V = {math.log(V):.4f})")
    for step in range(args.steps):
        inputs, labels = sample_batch()
        loss = model.loss(inputs, labels)
        opt.zero_grad()
        loss.backward()
        opt.step()
        if step % 25 == 0 or step == args.steps - 1:
            print(f"step {step}: CE {float(loss.detach()):.4f}")


if __name__ == "__main__":
    main()
