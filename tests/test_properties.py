"""Property-based tests (hypothesis) for the invariants everything else
leans on: serialization roundtrips, nested-structure pack/unpack, Kademlia
id/metric/bucket laws, TimedStorage expiration ordering."""
import hypothesis.strategies as st
import pytest
import torch
from hypothesis import given, settings

from learning_at_home_amd.dht.routing import DHTID, RoutingTable
from learning_at_home_amd.utils import (
    MSGPackSerializer,
    PytorchSerializer,
    TimedStorage,
    nested_flatten,
    nested_pack,
)

DTYPES = [torch.float32, torch.float16, torch.bfloat16, torch.int64, torch.int32, torch.uint8]


@st.composite
def tensors(draw):
    dtype = draw(st.sampled_from(DTYPES))
    shape = draw(st.lists(st.integers(0, 5), min_size=0, max_size=3))
    if dtype.is_floating_point:
        t = torch.randn(shape, dtype=torch.float32).to(dtype)
        t.requires_grad_(draw(st.booleans()))
        return t
    return torch.randint(0, 100, shape, dtype=dtype)


class TestSerializerProperties:
    @settings(max_examples=50, deadline=None)
    @given(tensors())
    def test_tensor_roundtrip(self, t):
        out = PytorchSerializer.loads(PytorchSerializer.dumps(t))
        assert out.dtype == t.dtype and out.shape == t.shape
        assert out.requires_grad == t.requires_grad
        assert torch.equal(out, t)

    @settings(max_examples=50, deadline=None)
    @given(st.recursive(
        st.none() | st.booleans() | st.integers(-2**40, 2**40) | st.floats(allow_nan=False) | st.text(max_size=8),
        lambda inner: st.lists(inner, max_size=3) | st.dictionaries(st.text(max_size=4), inner, max_size=3),
        max_leaves=10,
    ))
    def test_msgpack_roundtrip(self, obj):
        out = MSGPackSerializer.loads(MSGPackSerializer.dumps(obj))
        # msgpack tuples<->lists are normalized to lists; our strategy emits lists only
        assert out == obj

    @settings(max_examples=30, deadline=None)
    @given(st.lists(tensors(), min_size=1, max_size=4))
    def test_mixed_payload_roundtrip(self, ts):
        payload = {"tensors": ts, "meta": {"k": 1}}
        out = PytorchSerializer.loads(PytorchSerializer.dumps(payload))
        assert out["meta"] == {"k": 1}
        assert all(torch.equal(a, b) for a, b in zip(out["tensors"], ts))


class TestNestedProperties:
    @settings(max_examples=50, deadline=None)
    @given(st.recursive(
        st.integers(0, 100),
        lambda inner: st.lists(inner, max_size=3).map(tuple) | st.dictionaries(
            st.sampled_from("abcd"), inner, max_size=3),
        max_leaves=12,
    ))
    def test_flatten_pack_identity(self, structure):
        flat = list(nested_flatten(structure))
        assert nested_pack(flat, structure) == structure


class TestRoutingProperties:
    @settings(max_examples=100, deadline=None)
    @given(st.binary(min_size=1, max_size=32), st.binary(min_size=1, max_size=32))
    def test_id_determinism_and_metric(self, a, b):
        ida, idb = DHTID.generate(a), DHTID.generate(b)
        assert ida == DHTID.generate(a)  # deterministic
        da, db = int(ida) ^ int(idb), int(idb) ^ int(ida)
        assert da == db  # symmetric
        assert (da == 0) == (a == b) or a != b  # identity (collisions aside)
        assert 0 <= int(ida) < 2 ** 160

    @settings(max_examples=20, deadline=None)
    @given(st.lists(st.integers(0, 2 ** 160 - 1), min_size=1, max_size=64, unique=True),
           st.integers(0, 2 ** 160 - 1), st.integers(1, 20))
    def test_nearest_neighbors_sorted_by_xor(self, ids, key, k):
        table = RoutingTable(DHTID.generate(b"self"), bucket_size=8)
        for n in ids:
            table.add_or_update_node(n, ("127.0.0.1", 1))
        got = table.get_nearest_neighbors(key, k)
        dists = [n ^ key for n, _ in got]
        assert dists == sorted(dists)  # closest-first
        assert len(got) <= k
        stored = {n for n, _ in got}
        assert stored.issubset(set(ids))


class TestTimedStorageProperties:
    @settings(max_examples=30, deadline=None)
    @given(st.lists(st.tuples(st.integers(0, 5), st.floats(1e9, 2e9), st.integers(0, 100)),
                    min_size=1, max_size=20))
    def test_latest_expiration_wins(self, writes):
        from learning_at_home_amd.utils import get_dht_time

        store = TimedStorage()
        now = get_dht_time()
        best = {}
        for key, exp_off, val in writes:
            exp = now + 100 + exp_off  # always in the future
            store.store(key, val, exp)
            if key not in best or exp >= best[key][1]:
                if key in best and exp == best[key][1]:
                    continue  # equal expiration: implementation may keep either
                best[key] = (val, exp)
        for key, (val, exp) in best.items():
            got_val, got_exp = store.get(key)
            assert got_exp == exp


class TestRoutingOpsProperties:
    """Invariants of the CPU routing-op fallbacks (identical semantics to the
    HIP kernels — the GPU suite checks kernel-vs-fallback equality, these
    check the fallback against the mathematical contract)."""

    @given(st.integers(1, 24), st.integers(2, 12), st.sampled_from([1, 2, 4]),
           st.integers(0, 2**31 - 1))
    @settings(max_examples=40, deadline=None)
    def test_gating_topk_simplex_and_uniqueness(self, T, E, k, seed):
        from learning_at_home_amd.ops.functional import gating_topk
        if k > E:
            return
        torch.manual_seed(seed)
        logits = torch.randn(T, E)
        idx, w, probs = gating_topk(logits, k)
        assert idx.shape == (T, k) and w.shape == (T, k)
        for t in range(T):
            assert len(set(idx[t].tolist())) == k  # k distinct experts
        assert torch.all(w >= 0)
        assert torch.allclose(w.sum(-1), torch.ones(T), atol=1e-5)
        assert torch.allclose(probs.float().sum(-1), torch.ones(T), atol=1e-4)
        # selected experts are exactly the top-k by logit value
        ref = logits.topk(k, dim=-1).indices.sort(-1).values
        assert torch.equal(idx.long().sort(-1).values, ref)

    @given(st.integers(1, 30), st.sampled_from([1, 2, 4]), st.integers(2, 8),
           st.integers(1, 6), st.integers(0, 2**31 - 1))
    @settings(max_examples=40, deadline=None)
    def test_assign_slots_invariants(self, T, k, E, C, seed):
        from learning_at_home_amd.ops.functional import assign_slots
        torch.manual_seed(seed)
        topk_idx = torch.randint(0, E, (T, k), dtype=torch.int32)
        pos, counts, inv = assign_slots(topk_idx, E, C)
        assert torch.equal(counts.long(),
                           torch.bincount(topk_idx.reshape(-1).long(), minlength=E))
        # every kept (t, j) occupies a unique in-bounds slot of ITS expert,
        # and inv is its exact inverse
        seen = set()
        for t in range(T):
            for j in range(k):
                p = int(pos[t, j])
                e = int(topk_idx[t, j])
                if p < 0:
                    assert int(counts[e]) > C  # only over-capacity drops
                    continue
                assert 0 <= p < C
                assert (e, p) not in seen
                seen.add((e, p))
                assert int(inv[e, p]) == t * k + j
        # every non-(-1) inv entry round-trips
        for e in range(E):
            kept = int(min(int(counts[e]), C))
            assert (inv[e, :kept] >= 0).all()
            assert (inv[e, kept:] == -1).all()

    @given(st.integers(1, 12), st.integers(2, 6), st.sampled_from([1, 2]),
           st.integers(0, 2**31 - 1))
    @settings(max_examples=25, deadline=None)
    def test_dispatch_combine_roundtrip_no_drops(self, T, E, k, seed):
        """With capacity >= T*k nothing drops, so combine(dispatch(x)) must be
        exactly the gate-weighted sum of (k copies of) each token."""
        from learning_at_home_amd.ops.functional import (assign_slots, combine,
                                                         dispatch, gating_topk)
        if k > E:
            return
        torch.manual_seed(seed)
        D, C = 8, T * k
        x = torch.randn(T, D)
        logits = torch.randn(T, E)
        idx, w, _ = gating_topk(logits, k)
        pos, counts, inv = assign_slots(idx, E, C)
        assert (pos >= 0).all()
        xbuf = dispatch(x, idx, pos, inv, k)
        y = combine(xbuf, idx, pos, w, inv)
        ref = (w.sum(-1, keepdim=True)) * x  # sum_j w_j * x (same token back)
        assert torch.allclose(y, ref, atol=1e-5), (y - ref).abs().max().item()


class TestChunkedCEProperties:
    @given(st.integers(1, 40), st.sampled_from([4, 8, 12]), st.integers(1, 45),
           st.floats(0.0, 1.0), st.integers(0, 2**31 - 1))
    @settings(max_examples=40, deadline=None)
    def test_matches_unfused_any_chunk_and_ignores(self, T, V, chunk, ign_frac, seed):
        """Loss and grads equal F.cross_entropy for ANY chunk size and ANY
        ignore pattern (including all tokens ignored)."""
        import torch.nn.functional as F
        from learning_at_home_amd.ops.fused_ce import chunked_linear_cross_entropy
        torch.manual_seed(seed)
        D = 6
        x = torch.randn(T, D, requires_grad=True)
        w = torch.randn(V, D, requires_grad=True)
        labels = torch.randint(0, V, (T,))
        labels[torch.rand(T) < ign_frac] = -100
        loss = chunked_linear_cross_entropy(x, w, labels, chunk=chunk)
        loss.backward()
        assert torch.isfinite(loss)
        x2 = x.detach().clone().requires_grad_(True)
        w2 = w.detach().clone().requires_grad_(True)
        if (labels != -100).any():
            ref = F.cross_entropy((x2 @ w2.t()).float(), labels, ignore_index=-100)
            ref.backward()
            assert torch.allclose(loss, ref, rtol=1e-5, atol=1e-5)
            assert torch.allclose(x.grad, x2.grad, rtol=1e-4, atol=1e-5)
            assert torch.allclose(w.grad, w2.grad, rtol=1e-4, atol=1e-5)
        else:
            assert loss.item() == 0.0
            assert torch.all(x.grad == 0) and torch.all(w.grad == 0)
