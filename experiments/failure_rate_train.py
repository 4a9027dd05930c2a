#!/usr/bin/env python3
"""Convergence under expert failure rate — the reference paper's core
robustness claim, reproduced on the socket-RPC DMoE path: a fraction p of
ALL expert RPCs fail at random; the MoE drops non-responders, renormalizes
over survivors (zero grads for failed backwards), and training must still
converge with graceful degradation.

Usage: python experiments/failure_rate_train.py --steps 60 --rates 0 0.1 0.3
"""
import argparse
import os
import random
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))

import torch

from learning_at_home_amd import DHT, RemoteMixtureOfExperts, Server


def run(rate: float, steps: int, d: int = 64, experts: int = 4) -> list:
    import learning_at_home_amd.client.expert as expert_mod
    import learning_at_home_amd.client.moe as moe_mod

    orig = expert_mod.rpc_call

    def flaky_rpc(endpoint, kind, obj, timeout=30.0):
        if rate > 0 and kind in ("fwd", "bwd") and random.random() < rate:
            raise ConnectionError(f"emulated failure (p={rate})")
        return orig(endpoint, kind, obj, timeout)

    moe_mod.rpc_call = flaky_rpc
    try:
        dht = DHT(start=True)
        srv = Server.create(experts, "ffn", d, expert_prefix="fr",
                            initial_peers=[dht.endpoint], announce_period=5.0,
                            optim_kwargs={"lr": 0.05}, start=True)
        client_dht = DHT(initial_peers=[dht.endpoint], start=True)
        torch.manual_seed(0)
        random.seed(1234)
        moe = RemoteMixtureOfExperts(in_features=d, grid_size=(experts,),
                                     dht=client_dht, uid_prefix="fr", k_best=2,
                                     forward_timeout=5.0, backward_timeout=5.0)
        head = torch.nn.Linear(d, d)
        opt = torch.optim.SGD(list(moe.proj.parameters()) + list(head.parameters()), lr=0.05)
        x = torch.randn(16, d)
        target = torch.randn(16, d)
        losses = []
        skipped = 0
        for _ in range(steps):
            try:
                loss = (head(moe(x)) - target).pow(2).mean()
                opt.zero_grad()
                loss.backward()
                opt.step()
                losses.append(float(loss.detach()))
            except RuntimeError:
                # a token lost ALL its experts this step (possible at high
                # failure rates): skip the batch and keep training — the
                # realistic client policy for a total-quorum miss
                skipped += 1
        losses.append(skipped)  # stash for the caller
        srv.shutdown()
        if srv.dht is not None:
            srv.dht.shutdown()
        client_dht.shutdown()
        dht.shutdown()
        return losses  # [..., skipped_steps]
    finally:
        moe_mod.rpc_call = orig


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--steps", type=int, default=60)
    p.add_argument("--rates", type=float, nargs="+", default=[0.0, 0.1, 0.3])
    args = p.parse_args()
    for rate in args.rates:
        t0 = time.perf_counter()
        out = run(rate, args.steps)
        losses, skipped = out[:-1], out[-1]
        print(f"failure rate {rate:.0%}: loss {losses[0]:.4f} -> {losses[-1]:.4f} "
              f"over {len(losses)} completed steps ({skipped} skipped; "
              f"{time.perf_counter() - t0:.1f}s)")


if __name__ == "__main__":
    main()
