"""Decentralized parameter averaging (SURVEY.md §2 "DecentralizedAverager",
§3.4): DHT matchmaking + butterfly (reduce-scatter/all-gather-by-parts) group
all-reduce, Moshpit-style iterated rounds.

Two transports, one semantics:
- `DecentralizedAverager` (averager.py): socket part-exchange — works across
  hosts and on CPU (reference parity; elasticity tests);
- `RcclAverager` (rccl.py): intra-node GPU groups — RCCL reduce-scatter +
  all-gather on a side HIP stream, overlapped with compute (the MI355X path).
"""

from .averager import DecentralizedAverager
from .rccl import RcclAverager

__all__ = ["DecentralizedAverager", "RcclAverager"]
