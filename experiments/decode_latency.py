#!/usr/bin/env python3
"""Decode (serving) latency harness: TransformerDMoE.generate with per-block
KV caches, MoE layers either local (ExpertParallelMoE) or decentralized
(RemoteMixtureOfExperts over socket RPC + DHT — config #1 substrate).

Usage:
  python experiments/decode_latency.py --mode local --tokens 32
  python experiments/decode_latency.py --mode remote --tokens 16
"""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))

import torch

from learning_at_home_amd.models import TransformerDMoE, TransformerDMoEConfig


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--mode", choices=["local", "remote", "graph"], default="local")
    p.add_argument("--d-model", type=int, default=256)
    p.add_argument("--layers", type=int, default=2)
    p.add_argument("--experts", type=int, default=4)
    p.add_argument("--tokens", type=int, default=32)
    p.add_argument("--prompt", type=int, default=8)
    args = p.parse_args()

    cfg = TransformerDMoEConfig(
        vocab_size=1024, d_model=args.d_model, n_heads=4, n_layers=args.layers,
        seq_len=args.prompt + args.tokens + 1, num_experts=args.experts,
        k_best=2, ffn_dim=4 * args.d_model)

    servers, dhts = [], []
    if args.mode == "remote":
        from learning_at_home_amd import DHT, RemoteMixtureOfExperts, Server

        seed = DHT(start=True)
        dhts.append(seed)
        srv = Server.create(args.experts, "ffn", args.d_model, expert_prefix="dec",
                            initial_peers=[seed.endpoint], announce_period=5.0, start=True)
        servers.append(srv)
        client_dht = DHT(initial_peers=[seed.endpoint], start=True)
        dhts.append(client_dht)

        def moe_factory(i, c):
            return RemoteMixtureOfExperts(in_features=c.d_model, grid_size=(args.experts,),
                                          dht=client_dht, uid_prefix="dec", k_best=2)
    else:
        from learning_at_home_amd.parallel import ExpertParallelMoE

        def moe_factory(i, c):
            return ExpertParallelMoE(d_model=c.d_model, ffn_dim=c.ffn_dim,
                                     num_experts=c.num_experts, k=c.k_best)

    torch.manual_seed(0)
    model = TransformerDMoE(cfg, moe_factory).eval()
    prompt = torch.randint(0, cfg.vocab_size, (1, args.prompt))

    if args.mode == "graph":
        # hipGraph-captured decode vs the same static step replayed eagerly
        # (GPU; per-token latency A/B — graphs remove the launch gaps)
        from learning_at_home_amd.models import GraphDecoder

        assert torch.cuda.is_available(), "--mode graph needs a GPU"
        dev = torch.device("cuda:0")
        model = model.to(dev)
        for block in model.blocks:
            if hasattr(block.moe, "experts"):
                block.moe.experts.float()
                block.moe.experts.init_gpu_state(master_weights=True)
        prompt = prompt.to(dev)
        for use_graph in (False, True):
            dec = GraphDecoder(model, batch_size=1, use_graph=use_graph)
            dec.generate(prompt, max_new_tokens=4)  # warmup incl. capture
            dec2 = GraphDecoder(model, batch_size=1, use_graph=use_graph)
            torch.cuda.synchronize()
            t0 = time.perf_counter()
            out = dec2.generate(prompt, max_new_tokens=args.tokens)
            torch.cuda.synchronize()
            dt = time.perf_counter() - t0
            n = out.shape[1] - args.prompt
            tag = "hipGraph replay" if use_graph else "eager static"
            print(f"mode=graph [{tag}]: {n} tokens in {dt:.3f}s = "
                  f"{1e3 * dt / n:.2f} ms/token (d={args.d_model}, L={args.layers}, "
                  f"E={args.experts}, k=2, GPU)")
        return

    model.generate(prompt, max_new_tokens=2)  # warmup
    t0 = time.perf_counter()
    out = model.generate(prompt, max_new_tokens=args.tokens)
    dt = time.perf_counter() - t0
    n = out.shape[1] - args.prompt
    print(f"mode={args.mode}: {n} tokens in {dt:.3f}s = {1e3 * dt / n:.1f} ms/token "
          f"(d={args.d_model}, L={args.layers}, E={args.experts}, k=2, CPU)")

    for s in servers:
        s.shutdown()
        if s.dht is not None:
            s.dht.shutdown()
    for d in dhts:
        d.shutdown()


if __name__ == "__main__":
    main()
