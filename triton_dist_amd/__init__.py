"""triton_dist_amd — MI355X-native compute–communication-overlap framework.

A from-scratch CDNA4 (gfx950) implementation of the capabilities of
ByteDance-Seed/Triton-distributed: symmetric memory over hipIpc/xGMI,
tile-granular notify/wait primitives, fused AllGather-GEMM /
GEMM-ReduceScatter / GEMM-AllReduce / MoE dispatch-combine kernels
(hand-written HIP + MFMA, no Triton/MLIR), TP/EP/SP layers, Qwen3-family
models, and a serving Engine — with RCCL used only for bootstrap and
golden-reference collectives.
"""
from __future__ import annotations

__version__ = "0.1.0"

from .utils.distributed import (  # noqa: F401
    initialize_distributed,
    finalize_distributed,
    dist_print,
    rank,
    world_size,
    has_gpu,
)
from .runtime.symm_mem import (  # noqa: F401
    SymmHeap,
    SymmBuffer,
    init_symm_heap,
    get_heap,
    shutdown_heap,
)


def _load_native():
    """Import the native module, building it in-tree if stale/missing."""
    try:
        from . import _C  # noqa: F401
        return _C
    except ImportError:
        from .build import build

        build()
        from . import _C  # noqa: F401
        return _C


# The native module is REQUIRED on GPU hosts (no silent eager fallback);
# on CPU-only hosts ops fall back to torch reference implementations and
# _C stays optional (it still cross-compiles fine without a GPU).
import torch as _torch

if _torch.cuda.is_available():
    _C = _load_native()
else:
    try:
        _C = _load_native()
    except Exception:  # pragma: no cover - CPU-only host without hipcc
        _C = None
