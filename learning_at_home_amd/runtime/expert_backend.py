"""ExpertBackend — one trainable expert behind stateless-RPC-shaped
forward/backward (SURVEY.md §2 "ExpertBackend", §3.1 call stack).

Key reference semantics preserved exactly:
- `forward` keeps NO activations between calls;
- `backward` RE-RUNS forward under `enable_grad` on the inputs the client
  sends back (recompute-in-backward), runs autograd, returns grad_inputs;
- after every backward batch the expert's own optimizer steps immediately
  (asynchronous / delayed-gradient SGD — no global barrier anywhere).

MI355X note: this RPC-shaped backend serves the socket path (cross-host /
CPU parity, config #1) with plain PyTorch modules; the intra-node hot path
holds experts STACKED in parallel/ExpertStack and runs the hand-written
CDNA4 grouped-FFN kernels (ops/hip) — see parallel/expert_shard.py.
"""
from __future__ import annotations

from typing import Any, Dict, Optional, Sequence, Tuple

import torch
import torch.nn as nn

from ..utils import BatchTensorDescriptor, get_logger
from .task_pool import TaskPool

logger = get_logger(__name__)


class ExpertBackend:
    def __init__(
        self,
        name: str,
        expert: nn.Module,
        optimizer: torch.optim.Optimizer,
        *,
        args_schema: Sequence[BatchTensorDescriptor],
        outputs_schema: Optional[Sequence[BatchTensorDescriptor]] = None,
        max_batch_size: int = 16384,
        grad_clip: Optional[float] = None,
    ):
        self.name = name
        self.expert = expert
        self.optimizer = optimizer
        self.args_schema = tuple(args_schema)
        self.grad_clip = grad_clip
        self.update_count = 0
        # hosting dtype: clients always speak their own dtype (usually fp32);
        # inputs are cast to the expert's parameter dtype at this boundary and
        # outputs/grads cast back, so a server may host experts in bf16 (the
        # GPU serving fast path) transparently to every client
        first_param = next(iter(expert.parameters()), None)
        self._param_dtype = first_param.dtype if first_param is not None else None

        if outputs_schema is None:
            with torch.no_grad():
                dummy = [schema.make_batch(1) for schema in self.args_schema]
                device = next(expert.parameters()).device if any(True for _ in expert.parameters()) else torch.device("cpu")
                dummy = [d.to(device) for d in dummy]
                out = self.forward(*dummy)  # same boundary casts clients will see
                outputs_schema = tuple(BatchTensorDescriptor.from_tensor(o) for o in out)
        self.outputs_schema = tuple(outputs_schema)

        self.forward_pool = TaskPool(self.forward, f"{name}_forward", max_batch_size)
        self.backward_pool = TaskPool(self.backward, f"{name}_backward", max_batch_size)

    # ------------------------------------------------------------------
    def forward(self, *inputs: torch.Tensor) -> Tuple[torch.Tensor, ...]:
        """Inference-mode forward. No state is kept (backward recomputes).

        Serving hot path: a gelu FFNExpert hosted bf16 on a GPU runs through
        the fused MFMA NT kernels (ops.functional.fused_ffn_infer) — two
        launches, fused bias/GELU — instead of four torch ops; backward
        recompute stays on autograd (it needs the graph anyway)."""
        client_dtype = inputs[0].dtype if inputs and inputs[0].is_floating_point() else None
        with torch.no_grad():
            inputs = tuple(self._to_host_dtype(x) for x in inputs)
            if len(inputs) == 1:
                from ..ops.functional import can_fuse_ffn_infer, fused_ffn_infer

                if can_fuse_ffn_infer(inputs[0], self.expert):
                    e = self.expert
                    y = fused_ffn_infer(inputs[0], e.fc1.weight, e.fc1.bias,
                                        e.fc2.weight, e.fc2.bias)
                    return (self._to_client_dtype(y, client_dtype),)
            outputs = self.expert(*inputs)
        outputs = outputs if isinstance(outputs, tuple) else (outputs,)
        return tuple(self._to_client_dtype(o, client_dtype) for o in outputs)

    def _to_host_dtype(self, x: torch.Tensor) -> torch.Tensor:
        if (self._param_dtype is not None and x.is_floating_point()
                and self._param_dtype.is_floating_point and x.dtype != self._param_dtype):
            return x.to(self._param_dtype)
        return x

    @staticmethod
    def _to_client_dtype(x: torch.Tensor, dtype) -> torch.Tensor:
        return x.to(dtype) if dtype is not None and x.is_floating_point() and x.dtype != dtype else x

    def backward(self, *inputs_and_grads: torch.Tensor) -> Tuple[torch.Tensor, ...]:
        """Recompute forward under enable_grad, backprop the provided output
        grads, step this expert's optimizer, and return input grads."""
        num_inputs = len(self.args_schema)
        inputs, grad_outputs = inputs_and_grads[:num_inputs], inputs_and_grads[num_inputs:]
        client_dtypes = [x.dtype if x.is_floating_point() else None for x in inputs]
        with torch.enable_grad():
            detached = [self._to_host_dtype(x.detach()).requires_grad_(x.dtype.is_floating_point)
                        for x in inputs]
            outputs = self.expert(*detached)
            outputs = outputs if isinstance(outputs, tuple) else (outputs,)
            assert len(outputs) == len(grad_outputs), (
                f"{self.name}: got {len(grad_outputs)} grad tensors for {len(outputs)} outputs"
            )
            torch.autograd.backward(
                outputs, grad_tensors=[g.to(o.dtype) for g, o in zip(grad_outputs, outputs)],
                retain_graph=False, create_graph=False,
            )
        grad_inputs = tuple(
            self._to_client_dtype(x.grad if x.grad is not None else torch.zeros_like(x), dt)
            for x, dt in zip(detached, client_dtypes)
        )
        self.apply_gradients()
        return grad_inputs

    def apply_gradients(self) -> None:
        """One asynchronous optimizer step on this expert's own parameters."""
        if self.grad_clip is not None:
            torch.nn.utils.clip_grad_norm_(self.expert.parameters(), self.grad_clip)
        self.optimizer.step()
        self.optimizer.zero_grad(set_to_none=True)
        self.update_count += 1

    # ------------------------------------------------------------------
    def get_info(self) -> Dict[str, Any]:
        return {
            "name": self.name,
            "args_schema": [s.to_dict() for s in self.args_schema],
            "outputs_schema": [s.to_dict() for s in self.outputs_schema],
        }

    def get_pools(self) -> Tuple[TaskPool, TaskPool]:
        return self.forward_pool, self.backward_pool

    # checkpointing (SURVEY.md §5 "Checkpoint / resume": byte-compatible torch.save maps)
    def state_dict(self) -> Dict[str, Any]:
        return {
            "model": self.expert.state_dict(),
            "optimizer": self.optimizer.state_dict(),
            "update_count": self.update_count,
        }

    def load_state_dict(self, state: Dict[str, Any]) -> None:
        self.expert.load_state_dict(state["model"])
        if state.get("optimizer"):  # an EP-shard checkpoint has no torch-optim state
            self.optimizer.load_state_dict(state["optimizer"])
        self.update_count = int(state.get("update_count", 0))
