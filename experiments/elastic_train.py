#!/usr/bin/env python3
"""Config #5 end-to-end demo: fault-tolerant expert-parallel training with a
rank dropping and rejoining mid-epoch, DHT re-discovery, generation-fenced
group rebuild, dead-expert masking.

Spawns `--slots` worker processes (gloo on CPU; the identical code path runs
RCCL on GPUs). Slot (slots-1) exits at step `--drop-step` and a replacement
process rejoins at `--rejoin-step`. Workers train a tiny MoE on synthetic
data throughout; the script prints per-generation progress.

    python experiments/elastic_train.py --slots 3 --steps 60
"""
import argparse
import multiprocessing as mp
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))

import torch

from learning_at_home_amd import DHT


def worker(slot, seed_ep, args, is_rejoiner=False):
    from learning_at_home_amd.parallel import ExpertParallelMoE
    from learning_at_home_amd.parallel.elastic import ElasticGroup

    dht = DHT(initial_peers=[tuple(seed_ep)], start=True)
    group = ElasticGroup(dht, slot, max_slots=args.slots, lease=2.0, backend="gloo",
                         timeout=30.0)
    group.start_announcer()
    group.form_group(expected=None if is_rejoiner else args.slots, deadline=120.0)

    torch.manual_seed(42)  # identical model init on every slot (a rejoiner's
    # experts restart fresh, like a reference server without its checkpoint)
    moe = ExpertParallelMoE(d_model=64, ffn_dim=128, num_experts=args.slots * 2, k=2,
                            capacity_factor=4.0, aux_loss_coef=0.01, router_noise_std=0.1)
    moe.update_membership(torch.distributed.group.WORLD, group.members, args.slots)
    opt = torch.optim.SGD(moe.parameters(), lr=0.01)

    def reform(reason):
        print(f"[slot {slot}] re-forming ({reason})", flush=True)
        group.form_group(deadline=120.0)
        moe.update_membership(torch.distributed.group.WORLD, group.members, args.slots)
        print(f"[slot {slot}] gen {group.gen}: members {group.members}", flush=True)

    step = 0
    while step < args.steps:
        if slot == args.slots - 1 and not is_rejoiner and step == args.drop_step:
            print(f"[slot {slot}] leaving at step {step}", flush=True)
            group.leave()
            dht.shutdown()
            return
        # membership check between steps (cheap DHT poll)
        if step % 5 == 0 and group.membership_changed():
            reform("lease change detected")
        time.sleep(args.step_time)
        torch.manual_seed(1000 + step * 17 + slot)
        x = torch.randn(16, 64)
        try:
            y = moe(x)
            loss = y.pow(2).mean() + (moe.aux_loss if moe.aux_loss is not None else 0.0)
            opt.zero_grad()
            loss.backward()
            opt.step()
        except RuntimeError as e:  # peer died mid-collective: rebuild and go on
            reform(f"step failed: {str(e)[:90]}")
            continue
        if step % 10 == 0:
            print(f"[slot {slot}] step {step} gen {group.gen} world {group.world_size} "
                  f"loss {float(loss):.4f}", flush=True)
        step += 1
    print(f"[slot {slot}] done ({step} steps, final gen {group.gen})", flush=True)
    group.leave()
    dht.shutdown()


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--slots", type=int, default=3)
    p.add_argument("--steps", type=int, default=60)
    p.add_argument("--drop-step", type=int, default=15)
    p.add_argument("--rejoin-delay", type=float, default=8.0)
    p.add_argument("--step-time", type=float, default=0.3)
    args = p.parse_args()

    ctx = mp.get_context("spawn")
    seed = DHT(start=True)
    procs = [ctx.Process(target=worker, args=(s, list(seed.endpoint), args))
             for s in range(args.slots)]
    for proc in procs:
        proc.start()
    # rejoin the dropped slot well after the drop + survivor re-formation
    time.sleep(args.drop_step * args.step_time + args.rejoin_delay)
    print("[main] starting rejoiner", flush=True)
    rejoin = ctx.Process(target=worker, args=(args.slots - 1, list(seed.endpoint), args, True))
    rejoin.start()
    for proc in procs + [rejoin]:
        proc.join(timeout=300)
        if proc.is_alive():
            proc.terminate()
    seed.shutdown()
    print("[main] elastic run complete")


if __name__ == "__main__":
    main()
