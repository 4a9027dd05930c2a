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


def test_fused_ffn_infer_matches_reference():
    """Serving hot path: ExpertBackend.forward of a bf16 gelu FFNExpert on GPU
    must take the fused MFMA route and match the fp32 torch reference."""
    from learning_at_home_amd.ops.functional import can_fuse_ffn_infer, fused_ffn_infer
    from learning_at_home_amd.runtime import ExpertBackend
    from learning_at_home_amd.utils import BatchTensorDescriptor

    torch.manual_seed(0)
    d, ffn = 256, 512
    expert = FFNExpert(d, ffn).to("cuda:0", torch.bfloat16)
    for M in (1, 4, 37, 64, 129):  # ragged token counts exercise the 64-row tail tiles
        x = torch.randn(M, d, device="cuda:0", dtype=torch.bfloat16)
        assert can_fuse_ffn_infer(x, expert)
        y = fused_ffn_infer(x, expert.fc1.weight, expert.fc1.bias,
                            expert.fc2.weight, expert.fc2.bias)
        with torch.no_grad():
            ref = FFNExpert(d, ffn).to("cuda:0")
            ref.load_state_dict({k: v.float() for k, v in expert.state_dict().items()})
            y_ref = ref(x.float())
        assert y.shape == (M, d)
        assert torch.allclose(y.float(), y_ref, atol=2e-2, rtol=2e-2), \
            f"M={M}: max err {(y.float() - y_ref).abs().max().item()}"
    # the backend routes through the fused path (same numbers, tuple shape)
    backend = ExpertBackend("fused", expert, torch.optim.SGD(expert.parameters(), lr=0.0),
                            args_schema=(BatchTensorDescriptor.make(d, dtype="bfloat16"),))
    x = torch.randn(37, d, device="cuda:0", dtype=torch.bfloat16)
    (y_backend,) = backend.forward(x)
    y_direct = fused_ffn_infer(x, expert.fc1.weight, expert.fc1.bias,
                               expert.fc2.weight, expert.fc2.bias)
    assert torch.equal(y_backend, y_direct)
    # misaligned dims fall back cleanly
    odd = FFNExpert(96, 200).to("cuda:0", torch.bfloat16)
    assert not can_fuse_ffn_infer(torch.randn(4, 96, device="cuda:0", dtype=torch.bfloat16), odd)


def test_bf16_server_fp32_client_roundtrip():
    """End-to-end serving: bf16 GPU-hosted experts (fused MFMA forward path)
    behind the RPC socket, fp32 client — dtype casts at the backend boundary."""
    dht = DHT(start=True)
    server = Server.create(1, "ffn", 256, expert_prefix="bf16srv",
                           initial_peers=[dht.endpoint],
                           device=torch.device("cuda:0"), dtype=torch.bfloat16,
                           announce_period=2.0, start=True)
    client_dht = DHT(initial_peers=[dht.endpoint], start=True)
    try:
        (ep,) = client_dht.get_experts(["bf16srv.0"])
        assert ep is not None
        remote = RemoteExpert("bf16srv.0", *ep)
        x = torch.randn(37, 256, requires_grad=True)  # ragged M -> tail tiles
        y = remote(x)
        assert y.dtype == torch.float32 and y.shape == (37, 256)
        backend = server.expert_backends["bf16srv.0"]
        with torch.no_grad():
            ref = FFNExpert(256).to("cuda:0")
            ref.load_state_dict({k: v.float() for k, v in backend.expert.state_dict().items()})
            y_ref = ref(x.detach().to("cuda:0")).cpu()
        assert torch.allclose(y.detach(), y_ref, atol=0.05, rtol=0.05)
        y.sum().backward()
        assert x.grad is not None and x.grad.dtype == torch.float32
        assert torch.isfinite(x.grad).all()
    finally:
        server.shutdown()
        if server.dht is not None:
            server.dht.shutdown()
        client_dht.shutdown()
        dht.shutdown()
