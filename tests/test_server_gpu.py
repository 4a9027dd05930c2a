"""GPU server-path tests: Runtime owns cuda:0, experts on device, RPC
round-trip through host staging (SURVEY.md §4 (b) 1-GPU integration)."""
import pytest
import torch

pytestmark = pytest.mark.gpu

from learning_at_home_amd import DHT, RemoteExpert, Server
from learning_at_home_amd.models import FFNExpert


def test_server_on_gpu_roundtrip():
    dht = DHT(start=True)
    server = Server.create(2, "ffn", 256, expert_prefix="gpu", initial_peers=[dht.endpoint],
                           device=torch.device("cuda:0"), announce_period=2.0, start=True)
    client_dht = DHT(initial_peers=[dht.endpoint], start=True)
    try:
        (ep,) = client_dht.get_experts(["gpu.0"])
        assert ep is not None
        remote = RemoteExpert("gpu.0", *ep)
        x = torch.randn(4, 256, requires_grad=True)
        y = remote(x)
        assert y.shape == (4, 256) and torch.isfinite(y).all()
        # numerics vs the server's own module
        backend = server.expert_backends["gpu.0"]
        with torch.no_grad():
            ref = backend.expert(x.detach().to("cuda:0")).cpu()
        assert torch.allclose(y.detach(), ref, atol=1e-3)
        y.sum().backward()
        assert x.grad is not None and torch.isfinite(x.grad).all()
        assert backend.update_count >= 0
    finally:
        server.shutdown()
        if server.dht is not None:
            server.dht.shutdown()
        client_dht.shutdown()
        dht.shutdown()


def test_expert_stack_checkpoint_roundtrip_gpu(tmp_path):
    """GPU ExpertStack (with compute copies) -> checkpoint -> fresh GPU stack:
    bf16/transposed copies refreshed on import."""
    from learning_at_home_amd.parallel import ExpertStack

    torch.manual_seed(3)
    src = ExpertStack(2, 128, 256).to("cuda:0")
    src.init_gpu_state(master_weights=False)  # SR mode: masters on CPU
    src.save_checkpoints(tmp_path, "g")
    dst = ExpertStack(2, 128, 256).to("cuda:0")
    dst.init_gpu_state(master_weights=True)
    assert dst.load_checkpoints(tmp_path, "g") == 2
    assert torch.allclose(dst.W1c.float().cpu(), src.W1c.float().cpu(), atol=1e-2)
    assert torch.equal(dst.W1t.cpu(), dst.W1c.transpose(1, 2).contiguous().cpu())
