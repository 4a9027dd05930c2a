"""Runtime-layer unit tests (TaskPool batching min/max,
ordering; ExpertBackend numerics vs calling the module directly)."""
import threading
import time

import pytest

import torch

from learning_at_home_amd.models import FFNExpert
from learning_at_home_amd.runtime import ExpertBackend, Runtime, TaskPool
from learning_at_home_amd.utils import BatchTensorDescriptor


class TestTaskPool:
    def test_batches_multiple_tasks(self):
        calls = []

        def process(x):
            calls.append(x.shape[0])
            return (x * 2,)

        pool = TaskPool(process, "p", max_batch_size=64)
        futures = [pool.submit_task(torch.full((4, 8), float(i))) for i in range(3)]
        n = pool.process_batch()
        assert n == 12 and calls == [12], "tasks were not batched together"
        for i, f in enumerate(futures):
            (out,) = f.result(timeout=1)
            assert torch.equal(out, torch.full((4, 8), 2.0 * i)), "outputs mis-split"

    def test_max_batch_size_respected(self):
        sizes = []
        pool = TaskPool(lambda x: (sizes.append(x.shape[0]) or (x,)), "p", max_batch_size=10)
        for _ in range(3):
            pool.submit_task(torch.zeros(6, 2))
        pool.process_batch()  # only one 6-row task fits under 10
        assert sizes == [6]
        pool.process_batch()
        assert sizes == [6, 6]

    def test_oversized_task_still_processed(self):
        pool = TaskPool(lambda x: (x,), "p", max_batch_size=4)
        f = pool.submit_task(torch.zeros(9, 2))
        assert pool.process_batch() == 9
        assert f.result(1)[0].shape == (9, 2)

    def test_priority_is_oldest_task(self):
        p1 = TaskPool(lambda x: (x,), "a")
        p2 = TaskPool(lambda x: (x,), "b")
        p1.submit_task(torch.zeros(1, 1))
        time.sleep(0.01)
        p2.submit_task(torch.zeros(1, 1))
        assert p1.priority < p2.priority
        assert TaskPool(lambda x: (x,), "c").priority == float("inf")

    def test_exception_propagates_to_futures(self):
        def boom(x):
            raise ValueError("kaboom")

        pool = TaskPool(boom, "p")
        f = pool.submit_task(torch.zeros(2, 2))
        pool.process_batch()
        try:
            f.result(1)
            assert False, "expected exception"
        except ValueError:
            pass


class TestExpertBackendNumerics:
    def test_forward_equals_module(self):
        torch.manual_seed(0)
        expert = FFNExpert(16)
        b = ExpertBackend("e", expert, torch.optim.SGD(expert.parameters(), lr=0.1),
                          args_schema=(BatchTensorDescriptor.make(16),))
        x = torch.randn(5, 16)
        (out,) = b.forward(x)
        with torch.no_grad():
            assert torch.equal(out, expert(x))

    def test_backward_grad_and_step(self):
        torch.manual_seed(1)
        expert = FFNExpert(16)
        ref = FFNExpert(16)
        ref.load_state_dict(expert.state_dict())
        b = ExpertBackend("e", expert, torch.optim.SGD(expert.parameters(), lr=0.1),
                          args_schema=(BatchTensorDescriptor.make(16),))
        x = torch.randn(3, 16)
        g = torch.randn(3, 16)
        (grad_x,) = b.backward(x, g)
        # reference: autograd on the pristine copy
        xr = x.clone().requires_grad_(True)
        ref(xr).backward(g)
        assert torch.allclose(grad_x, xr.grad, atol=1e-5)
        # async SGD applied exactly once
        assert b.update_count == 1
        expected_w = {n: p - 0.1 * g2 for (n, p), g2 in
                      zip(ref.named_parameters(),
                          [q.grad for q in ref.parameters()])}
        for n, p in expert.named_parameters():
            assert torch.allclose(p.detach(), expected_w[n].detach(), atol=1e-5), n


class TestRuntimeLoop:
    def test_runtime_drains_pools(self):
        torch.manual_seed(2)
        expert = FFNExpert(8)
        b = ExpertBackend("e", expert, torch.optim.SGD(expert.parameters(), lr=0.1),
                          args_schema=(BatchTensorDescriptor.make(8),))
        rt = Runtime({"e": b})
        rt.run_in_background()
        try:
            futures = [b.forward_pool.submit_task(torch.randn(2, 8)) for _ in range(5)]
            outs = [f.result(timeout=5) for f in futures]
            assert all(o[0].shape == (2, 8) for o in outs)
            # samples_processed is a monitoring counter incremented AFTER the
            # batch's futures resolve — eventually consistent, so poll briefly
            deadline = time.monotonic() + 2.0
            while rt.samples_processed < 10 and time.monotonic() < deadline:
                time.sleep(0.01)
            assert rt.samples_processed == 10
        finally:
            rt.shutdown()


class TestExpertBackendDtypeBoundary:
    def test_bf16_hosted_expert_fp32_client(self):
        """A server may host experts in bf16 (the GPU serving fast path);
        fp32 clients must see fp32 outputs and fp32 input grads, with values
        matching an fp32 twin within bf16 tolerance."""
        import copy

        torch.manual_seed(0)
        d = 64
        expert32 = FFNExpert(d, 128)
        expert16 = copy.deepcopy(expert32).to(torch.bfloat16)
        opt = torch.optim.SGD(expert16.parameters(), lr=0.0)
        backend = ExpertBackend(
            "bf16", expert16, opt,
            args_schema=(BatchTensorDescriptor.make(d),),
        )
        # outputs_schema reflects the CLIENT-side dtype (fp32), not the host dtype
        assert backend.outputs_schema[0].torch_dtype == torch.float32

        x = torch.randn(5, d)
        (y,) = backend.forward(x)
        assert y.dtype == torch.float32
        with torch.no_grad():
            y_ref = expert32(x)
        assert torch.allclose(y, y_ref, atol=0.05, rtol=0.05)

        dy = torch.randn_like(y_ref)
        (gx,) = backend.backward(x, dy)
        assert gx.dtype == torch.float32
        x_ref = x.detach().requires_grad_(True)
        expert32(x_ref).backward(dy)
        assert torch.allclose(gx, x_ref.grad, atol=0.05, rtol=0.05)


class TestRuntimeFaultIsolation:
    def test_bad_batch_poisons_only_its_futures(self):
        """An expert that raises must fail THAT batch's futures and leave the
        Runtime serving subsequent requests (server survives poison inputs)."""

        class FlakyExpert(torch.nn.Module):
            def __init__(self):
                super().__init__()
                self.scale = torch.nn.Parameter(torch.ones(1))

            def forward(self, x):
                if bool((x > 9000).any()):
                    raise RuntimeError("poison batch")
                return x * self.scale

        expert = FlakyExpert()
        b = ExpertBackend("flaky", expert, torch.optim.SGD(expert.parameters(), lr=0.1),
                          args_schema=(BatchTensorDescriptor.make(4),))
        rt = Runtime({"flaky": b})
        rt.run_in_background()
        try:
            bad = b.forward_pool.submit_task(torch.full((1, 4), 10000.0))
            with pytest.raises(Exception, match="poison"):
                bad.result(timeout=5)
            good = b.forward_pool.submit_task(torch.ones(2, 4))
            (out,) = good.result(timeout=5)
            assert torch.allclose(out, torch.ones(2, 4))
        finally:
            rt.shutdown()


class TestMinBatchAccumulation:
    def test_min_batch_size_waits_then_timeout_flushes(self):
        """A pool with min_batch_size accumulates small tasks; a lone task is
        still flushed once the batching timeout elapses (latency bound)."""
        expert = FFNExpert(8)
        b = ExpertBackend("mb", expert, torch.optim.SGD(expert.parameters(), lr=0.1),
                          args_schema=(BatchTensorDescriptor.make(8),))
        pool = b.forward_pool
        pool.min_batch_size, pool.timeout = 4, 0.4
        rt = Runtime({"mb": b})
        rt.run_in_background()
        try:
            t0 = time.monotonic()
            lone = pool.submit_task(torch.randn(1, 8))
            lone.result(timeout=5)
            waited = time.monotonic() - t0
            assert waited >= 0.3, f"lone task flushed too early ({waited:.3f}s)"

            # enough samples -> processed promptly, no timeout wait
            t0 = time.monotonic()
            futs = [pool.submit_task(torch.randn(2, 8)) for _ in range(2)]
            for f in futs:
                f.result(timeout=5)
            assert time.monotonic() - t0 < 0.3, "full batch should not wait"
        finally:
            rt.shutdown()
