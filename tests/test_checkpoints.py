"""Checkpoint/resume tests (SURVEY.md §5 "Checkpoint / resume"): per-expert
torch.save maps under checkpoint_dir/<uid>/, checkpoint_last, restore."""
import torch

from learning_at_home_amd.models import FFNExpert
from learning_at_home_amd.runtime import ExpertBackend
from learning_at_home_amd.server import CheckpointSaver, load_experts, store_experts
from learning_at_home_amd.utils import BatchTensorDescriptor


def _make_backend(uid="ffn.0", d=32):
    expert = FFNExpert(d)
    opt = torch.optim.SGD(expert.parameters(), lr=0.1)
    return ExpertBackend(uid, expert, opt, args_schema=(BatchTensorDescriptor.make(d),))


class TestCheckpoints:
    def test_store_load_roundtrip(self, tmp_path):
        b = _make_backend()
        # train one async step so optimizer state exists
        x = torch.randn(4, 32)
        b.backward(x, torch.randn(4, 32))
        store_experts({"ffn.0": b}, tmp_path)
        assert (tmp_path / "ffn.0" / "checkpoint_last.pt").exists()
        timestamped = [p for p in (tmp_path / "ffn.0").iterdir() if p.name.startswith("checkpoint_2")]
        assert timestamped, "no timestamped checkpoint written"

        fresh = _make_backend()
        assert not torch.equal(
            fresh.expert.fc1.weight.detach(), b.expert.fc1.weight.detach()
        )
        restored = load_experts({"ffn.0": fresh}, tmp_path)
        assert restored == 1
        assert torch.equal(fresh.expert.fc1.weight.detach(), b.expert.fc1.weight.detach())
        assert fresh.update_count == b.update_count

    def test_checkpoint_format_is_plain_torch_save(self, tmp_path):
        """Contract: the file is a torch.save dict with 'model'/'optimizer'
        keys (byte-level compatibility target per BASELINE.json)."""
        b = _make_backend()
        store_experts({"ffn.0": b}, tmp_path)
        blob = torch.load(tmp_path / "ffn.0" / "checkpoint_last.pt", weights_only=False)
        assert set(blob.keys()) >= {"model", "optimizer"}
        assert "fc1.weight" in blob["model"]

    def test_saver_thread(self, tmp_path):
        b = _make_backend()
        saver = CheckpointSaver({"ffn.0": b}, tmp_path, update_period=0.2)
        saver.start()
        import time

        deadline = time.monotonic() + 5
        while time.monotonic() < deadline:
            if (tmp_path / "ffn.0" / "checkpoint_last.pt").exists():
                break
            time.sleep(0.05)
        saver.stop()
        assert (tmp_path / "ffn.0" / "checkpoint_last.pt").exists()

    def test_load_missing_is_noop(self, tmp_path):
        b = _make_backend()
        assert load_experts({"ffn.0": b}, tmp_path) == 0


class TestExpertStackInterop:
    def test_stack_checkpoint_loads_into_rpc_server_expert(self, tmp_path):
        """EP shard checkpoints and socket-server checkpoints share one
        format: a stack-written file restores an FFNExpert byte-for-byte."""
        import torch
        from learning_at_home_amd.parallel import ExpertStack

        torch.manual_seed(5)
        stack = ExpertStack(2, 32, 128)  # ffn_dim matches FFNExpert(32)'s 4*d default
        stack.save_checkpoints(tmp_path, "ffn", first_global_idx=4)
        assert (tmp_path / "ffn.5" / "checkpoint_last.pt").exists()

        b = _make_backend(uid="ffn.5", d=32)
        assert load_experts({"ffn.5": b}, tmp_path) == 1
        x = torch.randn(3, 32)
        with torch.no_grad():
            y_module = b.expert(x)
            ref = torch.nn.functional.gelu(x @ stack.W1[1].t() + stack.W1_bias[1],
                                           approximate="tanh") @ stack.W2[1].t() + stack.W2_bias[1]
        assert torch.allclose(y_module, ref, atol=1e-6)

    def test_stack_roundtrip(self, tmp_path):
        import torch
        from learning_at_home_amd.parallel import ExpertStack

        torch.manual_seed(6)
        src = ExpertStack(2, 32, 64)
        src.save_checkpoints(tmp_path, "rt")
        dst = ExpertStack(2, 32, 64)
        assert dst.load_checkpoints(tmp_path, "rt") == 2
        assert torch.equal(src.W1.detach(), dst.W1.detach())
        assert torch.equal(src.W2_bias.detach(), dst.W2_bias.detach())


class TestExpertStackOptimizerResume:
    """VERDICT r1 missing #5: EP-shard checkpoints must persist optimizer
    state so a restart is a true RESUME (identical loss trajectory), not a
    weight-only reload that silently resets momentum / Adam moments."""

    def _train_steps(self, moe, opt, steps, seed=123):
        losses = []
        g = torch.Generator().manual_seed(seed)
        for _ in range(steps):
            x = torch.randn(16, 8, generator=g)
            y = moe(x)
            loss = y.pow(2).mean() + (moe.aux_loss if moe.aux_loss is not None else 0.0)
            loss.backward()
            opt.step()
            losses.append(float(loss))
        return losses

    def _fresh(self, seed=7):
        from learning_at_home_amd.parallel import ExpertAdam, ExpertParallelMoE

        torch.manual_seed(seed)
        moe = ExpertParallelMoE(d_model=8, ffn_dim=16, num_experts=4, k=2,
                                router_noise_std=0.0, aux_loss_coef=0.01)
        moe.gate.weight.requires_grad_(False)  # experts-only training: the
        # per-uid checkpoint covers expert + optimizer state, not the gate
        opt = ExpertAdam(moe.experts, lr=5e-2)
        return moe, opt

    def test_adam_resume_trajectory_matches_uninterrupted(self, tmp_path):
        moe_a, opt_a = self._fresh()
        losses_ref = self._train_steps(moe_a, opt_a, 8)

        moe_b, opt_b = self._fresh()
        losses_b1 = self._train_steps(moe_b, opt_b, 4)
        assert losses_b1 == losses_ref[:4]
        moe_b.experts.save_checkpoints(tmp_path, "res", optimizer=opt_b,
                                       update_count=opt_b.step_count)
        del moe_b, opt_b  # "kill" the shard

        moe_c, opt_c = self._fresh(seed=7)  # same gate init via the seed
        assert moe_c.experts.load_checkpoints(tmp_path, "res", optimizer=opt_c) == 4
        assert opt_c.step_count == 4, "step counter not restored"
        # data stream continues where run A's step 5 started
        g = torch.Generator().manual_seed(123)
        for _ in range(4):
            torch.randn(16, 8, generator=g)  # replay consumed draws
        losses_c = []
        for _ in range(4):
            x = torch.randn(16, 8, generator=g)
            y = moe_c(x)
            loss = y.pow(2).mean() + (moe_c.aux_loss if moe_c.aux_loss is not None else 0.0)
            loss.backward()
            opt_c.step()
            losses_c.append(float(loss))
        for got, want in zip(losses_c, losses_ref[4:]):
            assert abs(got - want) < 1e-6, (
                f"resumed trajectory diverged: {losses_c} vs {losses_ref[4:]}")

    def test_weight_only_reload_diverges(self, tmp_path):
        """Sanity check that the test can actually tell the difference: a
        reload WITHOUT optimizer state must NOT reproduce the trajectory
        (Adam moments reset → different updates)."""
        moe_a, opt_a = self._fresh()
        losses_ref = self._train_steps(moe_a, opt_a, 8)

        moe_b, opt_b = self._fresh()
        self._train_steps(moe_b, opt_b, 4)
        moe_b.experts.save_checkpoints(tmp_path, "noopt", optimizer=opt_b)

        moe_c, opt_c = self._fresh(seed=7)
        moe_c.experts.load_checkpoints(tmp_path, "noopt")  # weights only
        g = torch.Generator().manual_seed(123)
        for _ in range(4):
            torch.randn(16, 8, generator=g)
        diverged = False
        for i in range(4):
            x = torch.randn(16, 8, generator=g)
            y = moe_c(x)
            loss = y.pow(2).mean() + (moe_c.aux_loss if moe_c.aux_loss is not None else 0.0)
            loss.backward()
            opt_c.step()
            if abs(float(loss) - losses_ref[4 + i]) > 1e-7:
                diverged = True
        assert diverged, "weight-only reload unexpectedly matched — test is vacuous"

    def test_fused_sgd_momentum_roundtrip(self, tmp_path):
        """FusedExpertSGD's momentum buffers and step counter (the SR seed
        source) survive a save/load cycle. (The kernel step itself is
        GPU-only; the state plumbing is what's under test here.)"""
        from learning_at_home_amd.parallel import ExpertStack, FusedExpertSGD

        torch.manual_seed(11)
        st = ExpertStack(3, 8, 16)
        opt = FusedExpertSGD(st, lr=0.1, momentum=0.9)
        for buf in opt.momentum_bufs:
            buf.normal_()
        opt.step_count = 17
        st.save_checkpoints(tmp_path, "sgdm", optimizer=opt, update_count=17)

        st2 = ExpertStack(3, 8, 16)
        opt2 = FusedExpertSGD(st2, lr=0.1, momentum=0.9)
        assert st2.load_checkpoints(tmp_path, "sgdm", optimizer=opt2) == 3
        assert opt2.step_count == 17
        for b1, b2 in zip(opt.momentum_bufs, opt2.momentum_bufs):
            assert torch.equal(b1, b2)


class TestServerRestoreOnStart:
    def test_server_create_restores_from_checkpoint_dir(self, tmp_path):
        """Restart journey: a server saves per-expert checkpoints; a FRESH
        Server.create pointed at the same dir serves the SAME weights."""
        from learning_at_home_amd import Server

        torch.manual_seed(0)
        s1 = Server.create(1, "ffn", 32, expert_prefix="ck", no_dht=True,
                           checkpoint_dir=tmp_path, start=True)
        try:
            x = torch.randn(3, 32)
            with torch.no_grad():
                y1 = s1.expert_backends["ck.0"].expert(x)
            store_experts(s1.expert_backends, tmp_path)
        finally:
            s1.shutdown()

        torch.manual_seed(99)  # different init: restore must override it
        s2 = Server.create(1, "ffn", 32, expert_prefix="ck", no_dht=True,
                           checkpoint_dir=tmp_path, start=True)
        try:
            with torch.no_grad():
                y2 = s2.expert_backends["ck.0"].expert(x)
            assert torch.allclose(y1, y2, atol=1e-6), "weights not restored"
        finally:
            s2.shutdown()
