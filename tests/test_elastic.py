"""Elastic EP membership tests (BASELINE.json config #5: drop/rejoin an
expert rank mid-epoch with DHT re-discovery; SURVEY.md §5 generation-fenced
communicator rebuild). Runs over gloo on localhost — the same code path
drives RCCL on GPUs."""
import multiprocessing as mp
import os
import time

import pytest
import torch
import torch.distributed as dist

from learning_at_home_amd import DHT

LEASE = 3.0


def _worker(slot, seed_endpoint, results, rejoiner=False):
    from learning_at_home_amd.parallel.elastic import ElasticGroup

    dht = DHT(initial_peers=[tuple(seed_endpoint)], start=True)
    group = ElasticGroup(dht, slot, max_slots=4, lease=LEASE, backend="gloo", timeout=60.0)
    group.start_announcer()
    try:
        if not rejoiner:
            # phase 1: all three form a group
            group.form_group(expected=3, deadline=120.0)
            t = torch.tensor([float(slot)])
            dist.all_reduce(t)
            assert t.item() == 3.0, f"phase1 allreduce got {t.item()}"
            results.put(("phase1", slot, group.gen, group.world_size))

            if slot == 2:
                group.leave()  # die mid-epoch
                dht.shutdown()
                return

            # phase 2: survivors wait out the lease, re-form as a pair
            deadline = time.monotonic() + 60
            while time.monotonic() < deadline:
                if 2 not in group.alive_slots():
                    break
                time.sleep(0.3)
            group.form_group(expected=2, deadline=120.0)
            t = torch.tensor([float(slot)])
            dist.all_reduce(t)
            assert t.item() == 1.0, f"phase2 allreduce got {t.item()}"
            results.put(("phase2", slot, group.gen, group.world_size))

            # phase 3: slot 2 rejoins (new process, same slot)
            deadline = time.monotonic() + 90
            while time.monotonic() < deadline:
                if 2 in group.alive_slots():
                    break
                time.sleep(0.3)
            group.form_group(expected=3, deadline=150.0)
            t = torch.tensor([float(slot)])
            dist.all_reduce(t)
            assert t.item() == 3.0, f"phase3 allreduce got {t.item()}"
            results.put(("phase3", slot, group.gen, group.world_size))
        else:
            group.form_group(expected=3, deadline=150.0)
            t = torch.tensor([float(slot)])
            dist.all_reduce(t)
            assert t.item() == 3.0
            results.put(("phase3", slot, group.gen, group.world_size))
    finally:
        try:
            group.leave()
        except Exception:  # noqa: BLE001
            pass
        dht.shutdown()


@pytest.mark.timeout(300)
def test_drop_and_rejoin_rank():
    ctx = mp.get_context("spawn")
    seed = DHT(start=True)
    results = ctx.Queue()
    procs = [
        ctx.Process(target=_worker, args=(slot, list(seed.endpoint), results))
        for slot in range(3)
    ]
    for p in procs:
        p.start()
    try:
        # collect phase1 from all three
        seen = {}
        deadline = time.monotonic() + 120
        while len([k for k in seen if k[0] == "phase1"]) < 3 and time.monotonic() < deadline:
            try:
                tag, slot, gen, world = results.get(timeout=5)
                seen[(tag, slot)] = (gen, world)
            except Exception:  # noqa: BLE001
                pass
        assert len([k for k in seen if k[0] == "phase1"]) == 3, f"phase1 incomplete: {seen}"
        assert all(w == 3 for k, (g, w) in seen.items() if k[0] == "phase1")

        # collect phase2 from survivors (0 and 1)
        while len([k for k in seen if k[0] == "phase2"]) < 2 and time.monotonic() < deadline:
            try:
                tag, slot, gen, world = results.get(timeout=5)
                seen[(tag, slot)] = (gen, world)
            except Exception:  # noqa: BLE001
                pass
        phase2 = {k: v for k, v in seen.items() if k[0] == "phase2"}
        assert len(phase2) == 2, f"phase2 incomplete: {seen}"
        assert all(w == 2 for g, w in phase2.values()), "survivors did not re-form as a pair"

        # now start the rejoiner for slot 2
        rejoin = ctx.Process(target=_worker, args=(2, list(seed.endpoint), results, True))
        rejoin.start()
        procs.append(rejoin)

        deadline = time.monotonic() + 150
        while len([k for k in seen if k[0] == "phase3"]) < 3 and time.monotonic() < deadline:
            try:
                tag, slot, gen, world = results.get(timeout=5)
                seen[(tag, slot)] = (gen, world)
            except Exception:  # noqa: BLE001
                pass
        phase3 = {k: v for k, v in seen.items() if k[0] == "phase3"}
        assert len(phase3) == 3, f"phase3 incomplete: {seen}"
        assert all(w == 3 for g, w in phase3.values()), "rejoin did not restore world=3"
    finally:
        for p in procs:
            p.join(timeout=30)
            if p.is_alive():
                p.terminate()
        seed.shutdown()


def _abort_worker(rank, world, port):
    """Rank 1 dies before joining rank 0's collective; rank 0's all_reduce
    times out, abort_comm() must return within bounded time and allow a
    fresh world-1 re-init (SURVEY §7 hard part (b): no deadlock on in-flight
    collectives during teardown)."""
    import datetime

    from learning_at_home_amd.parallel.elastic import ElasticGroup

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world,
                            timeout=datetime.timedelta(seconds=3))
    if rank == 1:
        return  # dies without participating
    t = torch.ones(4)
    raised = False
    try:
        dist.all_reduce(t)  # hangs until the 3 s gloo timeout, then raises
    except Exception:
        raised = True
    assert raised, "collective with a dead peer should have timed out"
    # bounded-time teardown despite the wedged collective
    group = ElasticGroup.__new__(ElasticGroup)
    group.slot = 0
    group.group = None
    import threading as _th

    group._stop = _th.Event()
    t0 = time.monotonic()
    group.abort_comm(timeout=10.0)
    assert time.monotonic() - t0 < 15.0, "abort_comm did not bound teardown time"
    assert not dist.is_initialized(), "process group still up after abort"
    # survivor re-forms alone and can use collectives again
    dist.init_process_group("gloo", rank=0, world_size=1,
                            init_method=f"tcp://127.0.0.1:{port + 1}")
    t2 = torch.full((4,), 2.0)
    dist.all_reduce(t2)
    assert torch.equal(t2, torch.full((4,), 2.0))
    dist.destroy_process_group()


def test_abort_comm_recovers_from_hung_collective():
    import socket

    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=_abort_worker, args=(r, 2, port)) for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(90)
    assert procs[0].exitcode == 0, "survivor rank failed abort/recovery"
    assert procs[1].exitcode == 0


def _rebind_worker(rank, world, port):
    """RcclAverager survives an elastic generation rebuild: average at
    world=2, lose a rank, abort, re-init at world=1, rebind, keep stepping."""
    from learning_at_home_amd.averaging import RcclAverager
    from learning_at_home_amd.parallel.elastic import ElasticGroup

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    params = [torch.full((6,), float(rank))]
    av = RcclAverager(params, dtype=torch.float32)
    av.step(async_op=False)
    assert torch.allclose(params[0], torch.full((6,), 0.5)), params[0]
    dist.barrier()  # both ranks finished round 1 before rank 1 departs
    if rank == 1:
        dist.destroy_process_group()
        return  # rank 1 leaves the job
    # survivor: generation rebuild to world=1, averager rebinds and still works
    group = ElasticGroup.__new__(ElasticGroup)
    group.slot = 0
    group.group = None
    import threading as _th

    group._stop = _th.Event()
    group.abort_comm(timeout=10.0)
    dist.init_process_group("gloo", rank=0, world_size=1,
                            init_method=f"tcp://127.0.0.1:{port + 1}")
    av.rebind()
    params[0].fill_(7.0)
    av.step(async_op=False)  # world 1: identity, must not hang or resize-crash
    assert torch.allclose(params[0], torch.full((6,), 7.0))
    dist.destroy_process_group()


def test_rccl_averager_rebind_after_membership_change():
    import socket

    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=_rebind_worker, args=(r, 2, port)) for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(90)
    assert all(p.exitcode == 0 for p in procs), [p.exitcode for p in procs]
