"""Model families: expert blocks hosted by servers (SURVEY.md §2 "Server.create
builds num_experts of expert_cls") and the flagship Transformer-DMoE used by
bench.py (BASELINE.json configs #2/#3)."""

from .experts import FFNExpert, TransformerExpert, NopExpert, name_to_block
from .transformer_dmoe import TransformerDMoE, TransformerDMoEConfig, DMoETransformerBlock
from .graph_decode import GraphDecoder

__all__ = [
    "FFNExpert",
    "TransformerExpert",
    "NopExpert",
    "name_to_block",
    "TransformerDMoE",
    "TransformerDMoEConfig",
    "DMoETransformerBlock",
    "GraphDecoder",
]
