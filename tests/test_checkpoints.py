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
