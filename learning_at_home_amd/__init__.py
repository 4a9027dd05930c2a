"""learning_at_home_amd — MI355X-native decentralized Mixture-of-Experts training engine.

A from-scratch rebuild of the Learning@home / hivemind capability set
(reference: mryab/learning-at-home; see SURVEY.md — the reference mount was
empty at survey time, so parity targets are cited as SURVEY.md sections, not
file:line) designed MI355X-first:

- control plane: asyncio Kademlia DHT (discovery + liveness leases)
- tensor plane: RCCL all-to-all over xGMI (token dispatch/return),
  RCCL reduce-scatter + all-gather on a side HIP stream (averaging)
- compute: hand-written CDNA4 HIP kernels (MFMA/LDS-tiled grouped expert
  FFN fwd/bwd, fused top-k gating + softmax + load-balance loss, combine,
  fused optimizer step) — no Triton, no CUDA shims.

Public API (contract per SURVEY.md §1/§2 and BASELINE.json north_star):
  RemoteExpert, RemoteMixtureOfExperts, DHT, DecentralizedAverager,
  ExpertBackend, Server, TaskPool, Runtime.
"""

__version__ = "0.1.0"

from .dht import DHT
from .runtime import ExpertBackend, Runtime, TaskPool
from .server import Server
from .client import MoEBeamSearcher, RemoteExpert, RemoteMixtureOfExperts
from .averaging import DecentralizedAverager
from .optim import DecentralizedSGD, MasterSGD

# reference-prototype codename aliases (the prototype called these
# TesseractServer / TesseractNetwork / TesseractRuntime)
TesseractServer = Server
TesseractNetwork = DHT
from .runtime import Runtime as TesseractRuntime  # noqa: E402

__all__ = [
    "DHT",
    "ExpertBackend",
    "Runtime",
    "TaskPool",
    "Server",
    "RemoteExpert",
    "RemoteMixtureOfExperts",
    "MoEBeamSearcher",
    "DecentralizedAverager",
    "DecentralizedSGD",
    "__version__",
]
