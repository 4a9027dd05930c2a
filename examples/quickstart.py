#!/usr/bin/env python3
"""Quickstart — the decentralized Learning@home workflow on one machine, CPU.

Walks the full user journey of the reference:
  1. start a DHT seed + two expert servers that announce under liveness leases
  2. build a RemoteMixtureOfExperts client that discovers experts via the DHT
  3. train a few steps (per-expert async SGD happens server-side on backward)
  4. kill one server mid-training -> the MoE drops it and keeps training
  5. checkpoint the survivors, restart a server from the checkpoint
  6. decode with the trained trunk through the same remote experts

For the single-node 8-GPU tensor plane (RCCL all-to-all + HIP kernels) see
bench.py; this example is the socket/DHT control-plane path that also works
cross-host.

Run: python examples/quickstart.py
"""
import os
import sys
import tempfile
import time

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))

import torch

from learning_at_home_amd import DHT, RemoteMixtureOfExperts, Server
from learning_at_home_amd.server import store_experts

D = 64


def main():
    print("1) DHT seed + 2 servers x 2 experts, announcing under 10s leases")
    seed = DHT(start=True)
    servers = [
        Server.create(2, "ffn", D, expert_prefix="demo", expert_offset=offset,
                      initial_peers=[seed.endpoint], announce_period=2.0, start=True)
        for offset in (0, 2)
    ]

    print("2) client discovers experts through its own DHT peer")
    client_dht = DHT(initial_peers=[seed.endpoint], start=True)
    moe = RemoteMixtureOfExperts(in_features=D, grid_size=(4,), dht=client_dht,
                                 uid_prefix="demo", k_best=2, active_cache_ttl=2.0)
    head = torch.nn.Linear(D, D)
    opt = torch.optim.SGD(list(moe.proj.parameters()) + list(head.parameters()), lr=0.05)

    torch.manual_seed(0)
    x, target = torch.randn(16, D), torch.randn(16, D)

    def train(steps):
        last = None
        for _ in range(steps):
            loss = (head(moe(x)) - target).pow(2).mean()
            opt.zero_grad(); loss.backward(); opt.step()
            last = float(loss.detach())
        return last

    print(f"3) 10 training steps: loss -> {train(10):.4f}")

    print("4) killing server 1 mid-training (experts demo.2, demo.3)...")
    servers[1].shutdown()
    time.sleep(5.0)  # lease (2 x announce period) lapses; cache TTL expires
    print(f"   ...training continues on survivors: loss -> {train(10):.4f}")

    print("5) checkpointing the surviving server + restarting from disk")
    with tempfile.TemporaryDirectory() as ckpt_dir:
        store_experts(servers[0].expert_backends, ckpt_dir)
        servers[0].shutdown()
        reborn = Server.create(2, "ffn", D, expert_prefix="demo",
                               initial_peers=[seed.endpoint], announce_period=2.0,
                               checkpoint_dir=ckpt_dir, start=True)
        time.sleep(2.0)
        print(f"   ...restored server announced; loss -> {train(5):.4f}")
        reborn.shutdown()
        if reborn.dht is not None:
            reborn.dht.shutdown()

    print("6) done — shutting down")
    for s in servers[:1]:
        if s.dht is not None:
            s.dht.shutdown()
    if servers[1].dht is not None:
        servers[1].dht.shutdown()
    client_dht.shutdown()
    seed.shutdown()


if __name__ == "__main__":
    main()
