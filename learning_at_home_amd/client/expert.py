"""RemoteExpert — client-side proxy module for one expert (SURVEY.md §2
"RemoteExpert"): forward() issues a "fwd" RPC; the autograd backward issues a
"bwd" RPC carrying (inputs, grad_outputs) — matching the reference's
stateless recompute-in-backward server contract.

Fault-tolerance semantics: a failed/timed-out backward RPC returns ZERO input
grads instead of raising (the MoE layer drops dead experts and renormalizes —
SURVEY.md §5 "Failure detection"); a failed forward raises, and the MoE layer
catches it per-expert.
"""
from __future__ import annotations

import queue
import socket
import threading
from typing import Dict, Optional, Tuple

import torch
import torch.nn as nn

from ..utils import (
    BatchTensorDescriptor,
    Endpoint,
    PytorchSerializer,
    get_logger,
    recv_message,
    send_message,
)

logger = get_logger(__name__)

_DEFAULT_TIMEOUT = 30.0


class _SocketPool:
    """Persistent per-endpoint TCP connections, one outstanding request each."""

    def __init__(self) -> None:
        self._free: Dict[Endpoint, queue.SimpleQueue] = {}
        self._lock = threading.Lock()

    def _get_queue(self, endpoint: Endpoint) -> queue.SimpleQueue:
        with self._lock:
            if endpoint not in self._free:
                self._free[endpoint] = queue.SimpleQueue()
            return self._free[endpoint]

    def request(self, endpoint: Endpoint, kind: str, payload: bytes,
                timeout: float = _DEFAULT_TIMEOUT) -> bytes:
        q = self._get_queue(endpoint)
        pooled = True
        try:
            sock = q.get_nowait()
        except queue.Empty:
            pooled = False
            sock = socket.create_connection(endpoint, timeout=timeout)
            sock.setsockopt(socket.IPPROTO_TCP, socket.TCP_NODELAY, 1)
        try:
            sock.settimeout(timeout)
            send_message(sock, kind, payload)
            resp_kind, resp_payload = recv_message(sock)
        except BaseException:
            try:
                sock.close()
            except OSError:
                pass
            # a pooled socket may simply be stale (server restarted between
            # calls): retry ONCE on a fresh connection before giving up — a
            # fresh connection failing means the server is really down
            if pooled:
                sock = socket.create_connection(endpoint, timeout=timeout)
                sock.setsockopt(socket.IPPROTO_TCP, socket.TCP_NODELAY, 1)
                try:
                    sock.settimeout(timeout)
                    send_message(sock, kind, payload)
                    resp_kind, resp_payload = recv_message(sock)
                except BaseException:
                    try:
                        sock.close()
                    except OSError:
                        pass
                    raise
            else:
                raise
        q.put(sock)
        if resp_kind == "error":
            raise RuntimeError(f"remote expert error: {resp_payload.decode(errors='replace')}")
        return resp_payload


_pool = _SocketPool()


def rpc_call(endpoint: Endpoint, kind: str, obj, timeout: float = _DEFAULT_TIMEOUT):
    payload = PytorchSerializer.dumps(obj)
    return PytorchSerializer.loads(_pool.request(endpoint, kind, payload, timeout))


class _RemoteModuleCall(torch.autograd.Function):
    @staticmethod
    def forward(ctx, uid: str, endpoint: Endpoint, timeout: float, backward_timeout: float,
                *inputs: torch.Tensor):
        inputs_cpu = tuple(x.detach().cpu() for x in inputs)
        ctx.uid, ctx.endpoint = uid, endpoint
        ctx.backward_timeout = backward_timeout
        ctx.device = inputs[0].device if inputs else torch.device("cpu")
        ctx.save_for_backward(*inputs_cpu)
        outputs = rpc_call(endpoint, "fwd", [uid, list(inputs_cpu)], timeout)
        return tuple(o.to(ctx.device) for o in outputs)

    @staticmethod
    def backward(ctx, *grad_outputs: torch.Tensor):
        inputs_cpu = ctx.saved_tensors
        grads_cpu = [g.detach().cpu() for g in grad_outputs]
        try:
            grad_inputs = rpc_call(
                ctx.endpoint, "bwd", [ctx.uid, list(inputs_cpu) + grads_cpu], ctx.backward_timeout
            )
            grad_inputs = tuple(g.to(ctx.device) for g in grad_inputs)
        except Exception as e:  # noqa: BLE001  — drop dead expert, zero grads
            logger.warning(f"backward RPC to {ctx.uid}@{ctx.endpoint} failed ({e!r}); zero grads")
            grad_inputs = tuple(torch.zeros_like(x, device=ctx.device) for x in inputs_cpu)
        return (None, None, None, None, *grad_inputs)


class RemoteExpert(nn.Module):
    """Calls an expert hosted on another server as if it were a local module.
    `RemoteExpert(uid, host, port)(x)` is differentiable end-to-end."""

    def __init__(self, uid: str, host: str, port: int,
                 timeout: float = _DEFAULT_TIMEOUT, backward_timeout: float = _DEFAULT_TIMEOUT):
        super().__init__()
        self.uid = uid
        self.endpoint: Endpoint = (host, port)
        self.timeout, self.backward_timeout = timeout, backward_timeout
        self._info: Optional[dict] = None

    @property
    def info(self) -> dict:
        if self._info is None:
            self._info = rpc_call(self.endpoint, "info", self.uid, self.timeout)
        return self._info

    @property
    def args_schema(self) -> Tuple[BatchTensorDescriptor, ...]:
        return tuple(BatchTensorDescriptor.from_dict(d) for d in self.info["args_schema"])

    @property
    def outputs_schema(self) -> Tuple[BatchTensorDescriptor, ...]:
        return tuple(BatchTensorDescriptor.from_dict(d) for d in self.info["outputs_schema"])

    def forward(self, *args: torch.Tensor):
        outputs = _RemoteModuleCall.apply(self.uid, self.endpoint, self.timeout,
                                          self.backward_timeout, *args)
        return outputs[0] if len(outputs) == 1 else outputs

    def extra_repr(self) -> str:
        return f"uid={self.uid}, endpoint={self.endpoint}"
