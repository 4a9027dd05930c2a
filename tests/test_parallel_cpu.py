"""Expert-parallel plane tests on CPU (the distributed path is
covered by multi-process gloo tests here; RCCL runs the same code on GPU)."""
import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp
import torch.nn.functional as F

from learning_at_home_amd.ops import functional as Fops
from learning_at_home_amd.parallel import ExpertParallelMoE


def _free_port() -> int:
    import socket

    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def _manual_moe_reference(x_flat, moe, capacity=None):
    """Dense reference: per-token top-k, softmax over selected, expert FFNs."""
    logits = moe.gate(x_flat).float()
    top_logits, topk_idx = logits.topk(moe.k, dim=-1)
    topk_w = F.softmax(top_logits, dim=-1)
    y = torch.zeros_like(x_flat)
    st = moe.experts
    for t in range(x_flat.shape[0]):
        for j in range(moe.k):
            e = int(topk_idx[t, j])
            out = Fops.expert_ffn_reference(
                x_flat[t : t + 1], st.W1[e], st.W1_bias[e], st.W2[e], st.W2_bias[e]
            )
            y[t] += topk_w[t, j] * out[0]
    return y


class TestAssignSlots:
    def test_basic(self):
        topk_idx = torch.tensor([[0, 1], [0, 2], [0, 1]], dtype=torch.int32)
        pos, counts, inv = Fops.assign_slots(topk_idx, E=4, C=2)
        assert counts.tolist() == [3, 2, 1, 0]
        # expert 0 got 3 assignments but capacity 2: exactly one dropped
        e0_pos = [pos[t, j] for t in range(3) for j in range(2) if topk_idx[t, j] == 0]
        assert sorted(int(p) for p in e0_pos) == [-1, 0, 1]
        # inv maps back to flat t*k+j
        for e in range(4):
            for c in range(2):
                src = int(inv[e, c])
                if src >= 0:
                    t, j = divmod(src, 2)
                    assert int(topk_idx[t, j]) == e
                    assert int(pos[t, j]) == c

    def test_no_drops_when_capacity_enough(self):
        torch.manual_seed(0)
        topk_idx = torch.randint(0, 8, (64, 2), dtype=torch.int32)
        pos, counts, inv = Fops.assign_slots(topk_idx, E=8, C=128)
        assert (pos >= 0).all()


class TestMoESingleProcess:
    def test_matches_dense_reference(self):
        torch.manual_seed(1)
        moe = ExpertParallelMoE(d_model=32, ffn_dim=64, num_experts=4, k=2,
                                capacity_factor=8.0, aux_loss_coef=0.0)
        moe.eval()  # exact-match test: no router exploration noise
        x = torch.randn(16, 32)
        y = moe(x)
        y_ref = _manual_moe_reference(x, moe)
        assert torch.allclose(y, y_ref, atol=1e-4), (y - y_ref).abs().max()

    def test_capacity_drop_renormalizes(self):
        torch.manual_seed(2)
        moe = ExpertParallelMoE(d_model=32, ffn_dim=64, num_experts=2, k=2,
                                capacity_factor=0.01, aux_loss_coef=0.0)
        x = torch.randn(64, 32)
        y = moe(x)
        assert torch.isfinite(y).all()

    def test_gradients(self):
        torch.manual_seed(3)
        moe = ExpertParallelMoE(d_model=32, ffn_dim=64, num_experts=4, k=2,
                                capacity_factor=8.0, aux_loss_coef=0.01)
        x = torch.randn(8, 32, requires_grad=True)
        y = moe(x)
        (y.sum() + moe.aux_loss).backward()
        assert x.grad is not None and torch.isfinite(x.grad).all()
        assert moe.gate.weight.grad is not None
        assert moe.experts.W1.grad is not None and moe.experts.W2_bias.grad is not None

    def test_gradcheck_combine(self):
        """combine() backward (incl. renormalization) vs autograd numerics."""
        torch.manual_seed(4)
        E, C, D, T, k = 3, 4, 8, 5, 2
        topk_idx = torch.randint(0, E, (T, k), dtype=torch.int32)
        pos, counts, inv = Fops.assign_slots(topk_idx, E, C)
        ybuf = torch.randn(E, C, D, dtype=torch.float64, requires_grad=True)
        w = torch.rand(T, k, dtype=torch.float64, requires_grad=True)

        def fn(ybuf_, w_):
            return Fops._combine_fwd_torch(ybuf_, topk_idx, pos, w_)

        assert torch.autograd.gradcheck(fn, (ybuf, w), atol=1e-6)


def _ep_worker(rank, world, tmpdir):
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")  # port set by parent
    dist.init_process_group("gloo", rank=rank, world_size=world)
    # single-rank subgroups so `full` runs all experts locally on each rank
    solo_groups = [dist.new_group([r]) for r in range(world)]
    torch.manual_seed(42)  # identical init everywhere
    full = ExpertParallelMoE(d_model=32, ffn_dim=64, num_experts=4, k=2,
                             capacity_factor=16.0, aux_loss_coef=0.0,
                             group=solo_groups[rank])
    torch.manual_seed(42)
    sharded = ExpertParallelMoE(d_model=32, ffn_dim=64, num_experts=4, k=2,
                                capacity_factor=16.0, aux_loss_coef=0.0)
    full.eval(); sharded.eval()  # exact-match: no router noise
    # shard the SAME weights: rank r keeps experts [r*2, r*2+2)
    n_local = 4 // world
    with torch.no_grad():
        for name in ("W1", "W1_bias", "W2", "W2_bias"):
            getattr(sharded.experts, name).copy_(
                getattr(full.experts, name)[rank * n_local : (rank + 1) * n_local]
            )
        sharded.gate.weight.copy_(full.gate.weight)
    torch.manual_seed(100 + rank)  # different tokens per rank (data parallel)
    x = torch.randn(12, 32)
    y_sharded = sharded(x)
    y_full = full(x)
    assert torch.allclose(y_sharded, y_full, atol=1e-4), (
        rank, (y_sharded - y_full).abs().max().item()
    )
    # gradient flow through the all-to-all
    y2 = sharded(x.requires_grad_(True))
    y2.sum().backward()
    assert sharded.experts.W1.grad is not None
    dist.destroy_process_group()


class TestExpertParallelGloo:
    def test_two_rank_matches_local(self, tmp_path):
        """2-rank expert-parallel forward == single-process all-local forward
        (token dispatch over all_to_all, experts sharded)."""
        os.environ["MASTER_PORT"] = str(_free_port())
        mp.spawn(_ep_worker, args=(2, str(tmp_path)), nprocs=2, join=True)


def _chunked_worker(rank, world, num_experts=8, n_chunks=2):
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")  # port set by parent
    dist.init_process_group("gloo", rank=rank, world_size=world)
    torch.manual_seed(7)
    plain = ExpertParallelMoE(d_model=16, ffn_dim=32, num_experts=num_experts, k=2,
                              capacity_factor=4.0, aux_loss_coef=0.01,
                              a2a_chunks=1)
    torch.manual_seed(7)
    chunked = ExpertParallelMoE(d_model=16, ffn_dim=32, num_experts=num_experts, k=2,
                                capacity_factor=4.0, aux_loss_coef=0.01,
                                a2a_chunks=n_chunks)
    plain.eval(); chunked.eval()
    torch.manual_seed(200 + rank)
    x = torch.randn(10, 16, requires_grad=True)
    x2 = x.detach().clone().requires_grad_(True)
    y1 = plain(x)
    y2 = chunked(x2)
    assert torch.allclose(y1, y2, atol=1e-6), (y1 - y2).abs().max().item()
    y1.pow(2).sum().backward()
    y2.pow(2).sum().backward()
    assert torch.allclose(x.grad, x2.grad, atol=1e-6)
    for n in ("W1", "W2", "W1_bias", "W2_bias"):
        g1 = getattr(plain.experts, n).grad
        g2 = getattr(chunked.experts, n).grad
        assert g1 is not None and g2 is not None
        assert torch.allclose(g1, g2, atol=1e-6), (n, (g1 - g2).abs().max().item())
    dist.destroy_process_group()


class TestChunkedA2AOverlap:
    """Chunked dispatch/return pipeline: numerics must be
    bit-identical to the unchunked path — same kernels, same per-expert
    batches — on 1 process and on 2 gloo ranks."""

    def test_single_process_chunked_matches_unchunked(self):
        for n_chunks in (2, 3, 4):
            torch.manual_seed(3)
            plain = ExpertParallelMoE(d_model=16, ffn_dim=32, num_experts=4, k=2,
                                      capacity_factor=8.0, router_noise_std=0.0,
                                      a2a_chunks=1)
            torch.manual_seed(3)
            chunked = ExpertParallelMoE(d_model=16, ffn_dim=32, num_experts=4, k=2,
                                        capacity_factor=8.0, router_noise_std=0.0,
                                        a2a_chunks=n_chunks)
            x = torch.randn(9, 16, requires_grad=True)
            x2 = x.detach().clone().requires_grad_(True)
            y1, y2 = plain(x), chunked(x2)
            assert torch.allclose(y1, y2, atol=1e-6)
            y1.sum().backward(); y2.sum().backward()
            assert torch.allclose(x.grad, x2.grad, atol=1e-6)
            assert torch.allclose(plain.experts.W1.grad, chunked.experts.W1.grad, atol=1e-6)

    def test_two_rank_gloo_chunked_matches_unchunked(self):
        os.environ["MASTER_PORT"] = str(_free_port())
        mp.spawn(_chunked_worker, args=(2,), nprocs=2, join=True)

    def test_four_rank_gloo_chunked_uneven_ranges(self):
        """4 gloo ranks, P=3 local experts, 2 chunks → uneven ranges (2+1):
        exercises the W>2 dispatch/return reshapes and non-divisible chunk
        bounds the driver's 8-GPU run will hit (e.g. E=256, N=8, chunks=4)."""
        os.environ["MASTER_PORT"] = str(_free_port())
        mp.spawn(_chunked_worker, args=(4, 12, 2), nprocs=4, join=True)

    def test_slice_grad_ready_ranges(self):
        """The stale-accumulate pitfall: after zero_main_grads, each chunk's
        FIRST backward of a step must see grad_ready=False for ITS range even
        if another chunk already ran (per-range tracking, not one bool)."""
        from learning_at_home_amd.parallel import ExpertStack

        st = ExpertStack(4, 8, 16)
        s01, s23 = st.slice_view(0, 2), st.slice_view(2, 4)
        assert not s01.grad_ready and not s23.grad_ready and not st.grad_ready
        s01.grad_ready = True
        assert s01.grad_ready and not s23.grad_ready, "chunk 2 must still WRITE"
        assert not st.grad_ready, "stack not ready until every range is"
        s23.grad_ready = True
        assert st.grad_ready
        st.zero_main_grads()
        assert not s01.grad_ready and not s23.grad_ready and not st.grad_ready
        # whole-stack mark (unchunked path) covers every slice
        st.grad_ready = True
        assert s01.grad_ready and s23.grad_ready and st.grad_ready


def _elastic_moe_worker(rank, world):
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")  # port set by parent
    dist.init_process_group("gloo", rank=rank, world_size=world)
    solo_groups = [dist.new_group([r]) for r in range(world)]
    torch.manual_seed(7)
    moe = ExpertParallelMoE(d_model=32, ffn_dim=64, num_experts=4, k=2,
                            capacity_factor=8.0, aux_loss_coef=0.0)
    moe.eval()
    x = torch.randn(10, 32)
    y_full = moe(x)
    assert torch.isfinite(y_full).all()

    # partition: each rank continues alone; the other slot's experts are dead
    moe.update_membership(solo_groups[rank], [rank], total_slots=world)
    y_solo = moe(x)
    assert torch.isfinite(y_solo).all()
    # dead experts must receive no tokens
    per_slot = 4 // world
    dead = [e for e in range(4) if e // per_slot != rank]
    for e in dead:
        assert int(moe.last_counts[e]) == 0, f"dead expert {e} got tokens"
    # gradients still flow through the surviving experts
    y2 = moe(x.requires_grad_(True))
    y2.sum().backward()
    assert moe.experts.W1.grad is not None

    # full membership restored
    moe.update_membership(None, [0, 1], total_slots=world)
    moe.group = None
    moe.world_size = world
    moe.group = dist.group.WORLD
    y_back = moe(x.detach())
    assert torch.isfinite(y_back).all()
    assert torch.allclose(y_back, y_full, atol=1e-4), "restored membership differs"
    dist.destroy_process_group()


class TestElasticMoE:
    def test_membership_shrink_and_restore(self):
        os.environ["MASTER_PORT"] = str(_free_port())
        mp.spawn(_elastic_moe_worker, args=(2,), nprocs=2, join=True)


def test_moe_stage_timers_populate():
    """SURVEY §5 tracing: attaching StageTimers to an ExpertParallelMoE times
    gate/route/dispatch/expert_ffn/combine individually."""
    from learning_at_home_amd.parallel import ExpertParallelMoE
    from learning_at_home_amd.utils.monitor import StageTimers

    torch.manual_seed(0)
    moe = ExpertParallelMoE(d_model=32, ffn_dim=64, num_experts=4, k=2)
    timers = StageTimers(use_gpu_events=False)
    moe.stage_timers = timers
    moe(torch.randn(3, 8, 32))
    s = timers.summary()
    for stage in ("moe_gate", "moe_route", "moe_assign", "moe_dispatch",
                  "moe_expert_ffn", "moe_combine"):
        assert s[stage]["count"] == 1, stage


def test_gating_topk_cpu_fallback_unit():
    """Direct unit check of the CPU gating fallback: full softmax probs,
    torch.topk selection, softmax-over-selected weights, and aux gradient
    flow (the 2048x-scale bug regression guard lives in the GPU suite; this
    pins the CPU reference the GPU kernels are tested against)."""
    torch.manual_seed(0)
    logits = torch.randn(64, 16, requires_grad=True)
    topk_idx, topk_w, probs = Fops.gating_topk(logits, k=4)
    assert torch.allclose(probs.sum(-1), torch.ones(64), atol=1e-5)
    ref_vals, ref_idx = logits.topk(4, dim=-1)
    assert torch.equal(topk_idx, ref_idx)
    assert torch.allclose(topk_w, F.softmax(ref_vals, dim=-1), atol=1e-5)
    assert torch.allclose(topk_w.sum(-1), torch.ones(64), atol=1e-5)
    (topk_w.sum() + probs.mean()).backward()
    assert logits.grad is not None and torch.isfinite(logits.grad).all()


def test_adam_update_matches_torch_adam():
    """adam_update_ (ExpertAdam's math) must exactly reproduce
    torch.optim.Adam over several steps, with and without weight decay."""
    from learning_at_home_amd.parallel.expert_shard import adam_update_

    torch.manual_seed(0)
    for wd in (0.0, 0.01):
        p_ref = torch.randn(64, requires_grad=True)
        p_mine = p_ref.detach().clone()
        m = torch.zeros_like(p_mine)
        v = torch.zeros_like(p_mine)
        opt = torch.optim.Adam([p_ref], lr=1e-2, betas=(0.9, 0.999), eps=1e-8,
                               weight_decay=wd)
        for step in range(1, 6):
            g = torch.randn(64)
            p_ref.grad = g.clone()
            opt.step()
            adam_update_(p_mine, g.clone(), m, v, step, 1e-2, 0.9, 0.999, 1e-8, wd)
            assert torch.allclose(p_mine, p_ref.detach(), atol=1e-7), (wd, step)
