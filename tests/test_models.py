"""Model-level numerics: chunked fused CE vs the unfused reference."""
import torch
import torch.nn.functional as F

from learning_at_home_amd.models import TransformerDMoE, TransformerDMoEConfig
from learning_at_home_amd.ops.fused_ce import chunked_linear_cross_entropy


class TestChunkedCE:
    def test_loss_and_grads_match_unfused(self):
        torch.manual_seed(0)
        T, D, V = 37, 16, 50
        x = torch.randn(T, D, requires_grad=True)
        w = torch.randn(V, D, requires_grad=True)
        labels = torch.randint(0, V, (T,))
        loss = chunked_linear_cross_entropy(x, w, labels, chunk=8)
        loss.backward()

        x2 = x.detach().clone().requires_grad_(True)
        w2 = w.detach().clone().requires_grad_(True)
        ref = F.cross_entropy((x2 @ w2.t()).float(), labels)
        ref.backward()
        assert torch.allclose(loss, ref, atol=1e-5)
        assert torch.allclose(x.grad, x2.grad, atol=1e-5)
        assert torch.allclose(w.grad, w2.grad, atol=1e-4)

    def test_ignore_index_matches_unfused(self):
        # padded labels (-100) must contribute neither loss nor gradient,
        # and the mean must be over valid tokens only
        torch.manual_seed(2)
        T, D, V = 41, 16, 50
        x = torch.randn(T, D, requires_grad=True)
        w = torch.randn(V, D, requires_grad=True)
        labels = torch.randint(0, V, (T,))
        labels[::3] = -100  # ~1/3 padded, spread across chunks
        loss = chunked_linear_cross_entropy(x, w, labels, chunk=8)
        loss.backward()

        x2 = x.detach().clone().requires_grad_(True)
        w2 = w.detach().clone().requires_grad_(True)
        ref = F.cross_entropy((x2 @ w2.t()).float(), labels, ignore_index=-100)
        ref.backward()
        assert torch.allclose(loss, ref, atol=1e-5)
        assert torch.allclose(x.grad, x2.grad, atol=1e-5)
        assert torch.allclose(w.grad, w2.grad, atol=1e-4)
        # ignored tokens get exactly zero dx
        assert torch.all(x.grad[::3] == 0)

    def test_all_labels_ignored_is_finite(self):
        x = torch.randn(8, 16, requires_grad=True)
        w = torch.randn(50, 16, requires_grad=True)
        labels = torch.full((8,), -100)
        loss = chunked_linear_cross_entropy(x, w, labels, chunk=4)
        loss.backward()
        assert loss.item() == 0.0
        assert torch.all(x.grad == 0) and torch.all(w.grad == 0)

    def test_model_loss_finite_and_trains(self):
        torch.manual_seed(1)
        cfg = TransformerDMoEConfig(vocab_size=128, d_model=32, n_heads=2, n_layers=1,
                                    seq_len=16, num_experts=4, k_best=2, ffn_dim=64)
        model = TransformerDMoE(cfg)
        ids = torch.randint(0, 128, (2, 17))
        loss = model.loss(ids[:, :-1], ids[:, 1:], fused_ce=True)
        loss.backward()
        assert torch.isfinite(loss)
        assert model.embed.weight.grad is not None  # tied lm_head got grads


def test_flash_attention_cpu_fallback_matches_sdpa():
    """On CPU (or non-qualifying shapes) flash_attention must be exactly the
    torch sdpa causal path, differentiable."""
    import torch.nn.functional as F

    from learning_at_home_amd.ops.functional import flash_attention, flash_attention_fwd

    torch.manual_seed(0)
    q = torch.randn(2, 2, 32, 16, requires_grad=True)
    k = torch.randn(2, 2, 32, 16, requires_grad=True)
    v = torch.randn(2, 2, 32, 16, requires_grad=True)
    out = flash_attention(q, k, v)
    ref = F.scaled_dot_product_attention(q, k, v, is_causal=True)
    assert torch.equal(out, ref)
    out.sum().backward()
    assert q.grad is not None and torch.isfinite(q.grad).all()
    with torch.no_grad():
        assert torch.allclose(flash_attention_fwd(q.detach(), k.detach(), v.detach()), ref)


def test_can_fuse_ffn_infer_gates_on_cpu():
    from learning_at_home_amd.models import FFNExpert
    from learning_at_home_amd.ops.functional import can_fuse_ffn_infer

    e = FFNExpert(128, 256)
    assert not can_fuse_ffn_infer(torch.randn(4, 128), e)  # CPU / fp32 -> no fuse


def test_generate_kv_cache_matches_full_recompute():
    """Greedy decode with per-block KV caches must produce exactly the same
    tokens as argmax over a full forward at every step."""
    from learning_at_home_amd.models import TransformerDMoE, TransformerDMoEConfig

    torch.manual_seed(0)
    cfg = TransformerDMoEConfig(vocab_size=128, d_model=64, n_heads=4, n_layers=2,
                                seq_len=32, num_experts=4, k_best=2, ffn_dim=96)
    model = TransformerDMoE(cfg).eval()  # dense FFN factory: deterministic
    prompt = torch.randint(0, 128, (2, 5))
    out = model.generate(prompt, max_new_tokens=6)
    assert out.shape == (2, 11)
    assert torch.equal(out[:, :5], prompt)
    # reference: re-run the FULL sequence through forward at each step
    ref = prompt
    with torch.no_grad():
        for _ in range(6):
            logits = model(ref)
            ref = torch.cat([ref, logits[:, -1].argmax(-1, keepdim=True)], dim=1)
    assert torch.equal(out, ref)


def test_generate_respects_seq_len_bound():
    from learning_at_home_amd.models import TransformerDMoE, TransformerDMoEConfig

    cfg = TransformerDMoEConfig(vocab_size=64, d_model=32, n_heads=2, n_layers=1,
                                seq_len=8, num_experts=2, k_best=1, ffn_dim=48)
    model = TransformerDMoE(cfg).eval()
    out = model.generate(torch.randint(0, 64, (1, 6)), max_new_tokens=10)
    assert out.shape[1] <= cfg.seq_len


def test_cpu_memorization_loss_falls():
    """End-to-end training dynamics on CPU: a tiny Transformer-DMoE must
    memorize one fixed batch (loss strictly falls over a few dozen steps).
    Guards the gating/dispatch/combine/aux wiring the GPU path shares."""
    from learning_at_home_amd.models import TransformerDMoE, TransformerDMoEConfig
    from learning_at_home_amd.parallel import ExpertParallelMoE

    torch.manual_seed(0)
    cfg = TransformerDMoEConfig(vocab_size=64, d_model=48, n_heads=4, n_layers=2,
                                seq_len=16, num_experts=4, k_best=2, ffn_dim=96)
    model = TransformerDMoE(cfg, lambda i, c: ExpertParallelMoE(
        d_model=c.d_model, ffn_dim=c.ffn_dim, num_experts=c.num_experts,
        k=c.k_best, router_noise_std=0.0))
    opt = torch.optim.SGD(model.parameters(), lr=0.05)
    ids = torch.randint(0, 64, (4, 17))
    inputs, labels = ids[:, :-1], ids[:, 1:]
    losses = []
    for _ in range(40):
        loss = model.loss(inputs, labels)
        opt.zero_grad(); loss.backward(); opt.step()
        losses.append(float(loss.detach()))
    assert losses[-1] < losses[0] - 0.5, f"no learning: {losses[0]:.3f} -> {losses[-1]:.3f}"


def test_flash_attention_env_flag_cpu(monkeypatch):
    """LAH_FLASH_ATTN=1 routes block attention through flash_attention (sdpa
    fallback on CPU — outputs identical to the flag-off path)."""
    from learning_at_home_amd.models import TransformerDMoE, TransformerDMoEConfig

    torch.manual_seed(0)
    cfg = TransformerDMoEConfig(vocab_size=64, d_model=32, n_heads=2, n_layers=1,
                                seq_len=8, num_experts=2, k_best=1, ffn_dim=48)
    model = TransformerDMoE(cfg).eval()
    ids = torch.randint(0, 64, (2, 8))
    with torch.no_grad():
        base = model(ids)
        monkeypatch.setenv("LAH_FLASH_ATTN", "1")
        flagged = model(ids)
    assert torch.equal(base, flagged)


def test_synthetic_markov_structure_is_learned():
    """Abbreviated version of experiments/synthetic_lm_train.py: CE on a
    fixed Markov chain must fall well below the unigram floor within 40
    steps (the model is learning transitions, not token frequencies)."""
    import math

    from learning_at_home_amd.models import TransformerDMoE, TransformerDMoEConfig
    from learning_at_home_amd.parallel import ExpertParallelMoE

    torch.manual_seed(0)
    V = 32
    trans = torch.softmax(4.0 * torch.randn(V, V), dim=-1)

    def sample(batch, seq):
        s = torch.empty(batch, seq + 1, dtype=torch.long)
        s[:, 0] = torch.randint(0, V, (batch,))
        for t in range(seq):
            s[:, t + 1] = torch.multinomial(trans[s[:, t]], 1).squeeze(-1)
        return s[:, :-1], s[:, 1:]

    cfg = TransformerDMoEConfig(vocab_size=V, d_model=48, n_heads=4, n_layers=1,
                                seq_len=32, num_experts=4, k_best=2, ffn_dim=96)
    model = TransformerDMoE(cfg, lambda i, c: ExpertParallelMoE(
        d_model=c.d_model, ffn_dim=c.ffn_dim, num_experts=c.num_experts,
        k=c.k_best, router_noise_std=0.1))
    opt = torch.optim.Adam(model.parameters(), lr=5e-3)
    first = last = None
    for step in range(40):
        inputs, labels = sample(12, cfg.seq_len)
        loss = model.loss(inputs, labels)
        opt.zero_grad(); loss.backward(); opt.step()
        if step == 0:
            first = float(loss.detach())
        last = float(loss.detach())
    floor = math.log(V)
    assert first > floor - 0.7  # starts near the unigram floor
    assert last < floor - 0.8, f"no structure learned: {first:.3f} -> {last:.3f} (ln V={floor:.3f})"


def test_graph_decoder_static_path_matches_generate_cpu():
    """GraphDecoder's static-state step (eager on CPU) must reproduce the
    dynamic-cache model.generate greedy tokens."""
    import torch as _t

    from learning_at_home_amd.models import GraphDecoder, TransformerDMoE, TransformerDMoEConfig

    _t.manual_seed(5)
    cfg = TransformerDMoEConfig(vocab_size=97, d_model=32, n_heads=2, n_layers=2,
                                seq_len=40, num_experts=1, k_best=1, ffn_dim=64)
    model = TransformerDMoE(cfg).eval()  # dense FFN fallback
    prompt = _t.randint(0, 97, (1, 6))
    with _t.no_grad():
        ref = model.generate(prompt, max_new_tokens=12)
    dec = GraphDecoder(model, batch_size=1, use_graph=False)
    out = dec.generate(prompt, max_new_tokens=12)
    assert out.shape == ref.shape
    assert _t.equal(out, ref), (out.tolist(), ref.tolist())
