#!/usr/bin/env python3
"""DHT control-plane benchmark (parity with hivemind's benchmark_dht):
spawn N Kademlia peers on localhost, measure store and get throughput and
expert declare/resolve latency."""
import argparse
import os
import random
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))

from learning_at_home_amd import DHT
from learning_at_home_amd.utils import get_dht_time


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--peers", type=int, default=16)
    p.add_argument("--keys", type=int, default=200)
    p.add_argument("--expiration", type=float, default=300.0)
    args = p.parse_args()

    t0 = time.perf_counter()
    seed = DHT(start=True)
    peers = [DHT(initial_peers=[seed.endpoint], start=True) for _ in range(args.peers - 1)]
    all_peers = [seed] + peers
    print(f"{args.peers} peers up in {time.perf_counter()-t0:.2f}s")

    random.seed(0)
    t0 = time.perf_counter()
    for i in range(args.keys):
        peer = random.choice(all_peers)
        assert peer.store(f"key{i}", i, get_dht_time() + args.expiration)
    dt = time.perf_counter() - t0
    print(f"store: {args.keys} keys in {dt:.2f}s = {args.keys/dt:.1f} stores/s")

    t0 = time.perf_counter()
    hits = 0
    for i in range(args.keys):
        peer = random.choice(all_peers)
        value, _ = peer.get(f"key{i}")
        hits += value == i
    dt = time.perf_counter() - t0
    print(f"get:   {args.keys} keys in {dt:.2f}s = {args.keys/dt:.1f} gets/s, "
          f"{100*hits/args.keys:.1f}% hit rate")

    uids = [f"bench.{i}" for i in range(64)]
    t0 = time.perf_counter()
    all_peers[1].declare_experts(uids, "127.0.0.1", 9999, expiration=args.expiration)
    t_declare = time.perf_counter() - t0
    t0 = time.perf_counter()
    found = all_peers[-1].get_experts(uids)
    t_resolve = time.perf_counter() - t0
    ok = sum(e is not None for e in found)
    print(f"declare 64 experts: {t_declare*1e3:.1f} ms; resolve: {t_resolve*1e3:.1f} ms "
          f"({ok}/64 found)")

    for peer in all_peers:
        peer.shutdown()


if __name__ == "__main__":
    main()
