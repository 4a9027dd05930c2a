"""MI355X kernel suite (SURVEY.md §2 "hot-op" inventory).

The HIP extension (_lah_hip) is built IN-TREE by `python setup.py build_ext
--inplace` (or __graft_entry__.build()). Policy (per round contract): on a
GPU box the HIP path is mandatory — ops FAIL LOUDLY if the extension is
missing; on CPU the pure-torch fallbacks in functional.py serve the
multi-process gloo tests and numerics references.
"""
import torch

from ..utils import get_logger

logger = get_logger(__name__)

try:
    from . import _lah_hip as _C  # noqa: N811

    HAS_HIP = True
except ImportError as e:
    _C = None
    HAS_HIP = False
    _import_error = e
    if torch.cuda.is_available():
        raise RuntimeError(
            "learning_at_home_amd: GPU present but the HIP extension is not built. "
            "Run `PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace`."
        ) from e


def require_hip() -> None:
    if not HAS_HIP:
        raise RuntimeError(
            "HIP extension _lah_hip not available (build with "
            "`PYTORCH_ROCM_ARCH=gfx950 python setup.py build_ext --inplace`)"
        )


from . import functional  # noqa: E402

__all__ = ["_C", "HAS_HIP", "require_hip", "functional"]
