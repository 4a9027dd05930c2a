"""Client layer: autograd-integrated remote expert calls."""

from .expert import RemoteExpert, rpc_call
from .moe import RemoteMixtureOfExperts
from .beam_search import MoEBeamSearcher

__all__ = ["RemoteExpert", "RemoteMixtureOfExperts", "MoEBeamSearcher", "rpc_call"]
