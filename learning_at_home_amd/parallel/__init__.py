"""Expert-parallel tensor plane: RCCL all-to-all over xGMI."""

from .expert_shard import ExpertParallelMoE, ExpertStack, FusedExpertSGD

__all__ = ["ExpertParallelMoE", "ExpertStack", "FusedExpertSGD"]
