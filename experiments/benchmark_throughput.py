#!/usr/bin/env python3
"""Reference-style DMoE throughput harness: spawn server
processes hosting experts + trainer threads driving RemoteMixtureOfExperts
over real sockets/DHT on localhost; report samples/s forward(+backward).

This measures the CONTROL-PLANE (RPC) path — the reference's own benchmark
shape. The GPU tensor-plane numbers come from bench.py.

Usage: python experiments/benchmark_throughput.py --servers 2 --experts-per-server 2 \
           --trainers 2 --batch 16 --steps 20 --backward
"""
import argparse
import os
import sys
import threading
import time

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))

import torch

from learning_at_home_amd import DHT, RemoteMixtureOfExperts, Server


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--servers", type=int, default=2)
    p.add_argument("--experts-per-server", type=int, default=2)
    p.add_argument("--trainers", type=int, default=2)
    p.add_argument("--hidden", type=int, default=128)
    p.add_argument("--batch", type=int, default=16)
    p.add_argument("--steps", type=int, default=20)
    p.add_argument("--k-best", type=int, default=2)
    p.add_argument("--backward", action="store_true")
    p.add_argument("--device", default=None, help="host experts on this device (e.g. cuda:0)")
    p.add_argument("--dtype", default=None, choices=["float32", "bfloat16"],
                   help="host experts in this dtype (bfloat16 + GPU = fused MFMA serving)")
    p.add_argument("--latency-ms", type=float, default=0.0,
                   help="software WAN emulation: add this one-way delay to every "
                        "expert RPC (the reference paper's tc-netem throughput "
                        "study, reproduced without root)")
    args = p.parse_args()

    if args.latency_ms > 0:
        import learning_at_home_amd.client.expert as _expert_mod
        import learning_at_home_amd.client.moe as _moe_mod

        _orig = _expert_mod.rpc_call

        def delayed_rpc(endpoint, kind, obj, timeout=30.0):
            time.sleep(args.latency_ms / 1e3)  # one-way request delay
            out = _orig(endpoint, kind, obj, timeout)
            time.sleep(args.latency_ms / 1e3)  # one-way response delay
            return out

        _expert_mod.rpc_call = delayed_rpc
        _moe_mod.rpc_call = delayed_rpc

    num_experts = args.servers * args.experts_per_server
    seed_dht = DHT(start=True)
    servers = []
    for s in range(args.servers):
        servers.append(Server.create(
            args.experts_per_server, "ffn", args.hidden, expert_prefix="bench",
            expert_offset=s * args.experts_per_server,
            initial_peers=[seed_dht.endpoint], announce_period=5.0, start=True,
            device=torch.device(args.device) if args.device else None,
            dtype=getattr(torch, args.dtype) if args.dtype else None,
        ))

    results = [0] * args.trainers

    def trainer(idx):
        dht = DHT(initial_peers=[seed_dht.endpoint], start=True)
        moe = RemoteMixtureOfExperts(
            in_features=args.hidden, grid_size=(num_experts,), dht=dht,
            uid_prefix="bench", k_best=args.k_best,
        )
        torch.manual_seed(idx)
        n = 0
        for _ in range(args.steps):
            x = torch.randn(args.batch, args.hidden, requires_grad=args.backward)
            y = moe(x)
            if args.backward:
                y.sum().backward()
                moe.zero_grad(set_to_none=True)
            n += args.batch
        results[idx] = n
        dht.shutdown()

    t0 = time.perf_counter()
    threads = [threading.Thread(target=trainer, args=(i,)) for i in range(args.trainers)]
    for t in threads:
        t.start()
    for t in threads:
        t.join()
    elapsed = time.perf_counter() - t0

    total = sum(results)
    mode = "fwd+bwd" if args.backward else "fwd"
    print(f"{total} samples ({mode}) in {elapsed:.2f}s = {total/elapsed:.1f} samples/s "
          f"({args.servers} servers x {args.experts_per_server} experts, "
          f"{args.trainers} trainers, batch {args.batch}, k={args.k_best})")

    for s in servers:
        s.shutdown()
        if s.dht is not None:
            s.dht.shutdown()
    seed_dht.shutdown()


if __name__ == "__main__":
    main()
