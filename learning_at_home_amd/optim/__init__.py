"""Decentralized optimizer wrappers (SURVEY.md §2 "Optimizers"): local steps
+ periodic group averaging. Mirrors hivemind's DecentralizedSGD /
CollaborativeOptimizer shape: wrap any torch optimizer; every
`averaging_period` local steps, run one averager round over the model
parameters (asynchronously — training never blocks on a barrier)."""
from __future__ import annotations

from typing import Optional, Union

import torch
import torch.distributed as dist

from ..averaging import DecentralizedAverager, RcclAverager
from ..utils import get_logger
from .master_sgd import MasterSGD

logger = get_logger(__name__)


class DecentralizedSGD:
    """optimizer wrapper: `opt = DecentralizedSGD(params, lr=..., dht=dht,
    prefix='myrun', averaging_period=10)`; call .step() like a normal
    optimizer — every `averaging_period` steps it also runs group averaging."""

    def __init__(
        self,
        params,
        *,
        dht=None,
        prefix: str = "decentralized_sgd",
        averaging_period: int = 10,
        target_group_size: int = 8,
        opt_cls=torch.optim.SGD,
        averager: Optional[Union[DecentralizedAverager, RcclAverager]] = None,
        average_gradients: bool = False,
        **opt_kwargs,
    ):
        params = list(params)
        self.opt = opt_cls(params, **opt_kwargs)
        self.params = [p for group in self.opt.param_groups for p in group["params"]]
        self.averaging_period = averaging_period
        self.local_step = 0
        if averager is not None:
            self.averager = averager
        elif dht is not None:
            self.averager = DecentralizedAverager(
                self.params, dht, prefix=prefix, target_group_size=target_group_size, start=True
            )
        else:
            self.averager = None
        # hivemind CollaborativeOptimizer-style synchronous gradient averaging:
        # every step, grads are all-reduced (AVG) across the group BEFORE the
        # local optimizer applies them. RCCL tensor-plane only; grads must keep
        # stable storage, so zero_grad is forced to set_to_none=False.
        self.average_gradients = average_gradients
        if average_gradients:
            assert dist.is_initialized(), "average_gradients needs torch.distributed"
        self._grad_averager: Optional[RcclAverager] = None

    def zero_grad(self, set_to_none: bool = True) -> None:
        if self.average_gradients:
            set_to_none = False  # keep grad storage stable for the fused buffer
        self.opt.zero_grad(set_to_none=set_to_none)

    def step(self, closure=None):
        if self.average_gradients:
            grads = [p.grad for p in self.params if p.grad is not None]
            if grads:
                if self._grad_averager is None or len(self._grad_averager.tensors) != len(grads):
                    self._grad_averager = RcclAverager(grads, dtype=torch.float32)
                self._grad_averager.step(async_op=False)
        if self.averager is not None and isinstance(self.averager, RcclAverager):
            self.averager.wait()  # ensure any in-flight round has written back
        if isinstance(self.averager, DecentralizedAverager):
            with self.averager.get_tensors():
                result = self.opt.step(closure)
        else:
            result = self.opt.step(closure)
        self.local_step += 1
        if self.averager is not None and self.local_step % self.averaging_period == 0:
            if isinstance(self.averager, RcclAverager):
                self.averager.step(async_op=True)
            else:
                self.averager.step()
        return result

    @property
    def param_groups(self):
        return self.opt.param_groups

    def state_dict(self):
        return self.opt.state_dict()

    def load_state_dict(self, sd):
        self.opt.load_state_dict(sd)

    def shutdown(self) -> None:
        if isinstance(self.averager, DecentralizedAverager):
            self.averager.shutdown()


CollaborativeOptimizer = DecentralizedSGD  # alias (hivemind naming)

__all__ = ["DecentralizedSGD", "CollaborativeOptimizer", "MasterSGD"]
