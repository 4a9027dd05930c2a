"""Expert-parallel tensor plane: RCCL all-to-all over xGMI."""

from .expert_shard import ExpertAdam, ExpertParallelMoE, ExpertStack, FusedExpertSGD

__all__ = ["ExpertAdam", "ExpertParallelMoE", "ExpertStack", "FusedExpertSGD"]
