"""sboxgates_amd — MI355X-native S-box circuit-minimization engine.

A from-scratch implementation of the capabilities of dansarie/sboxgates
(Kwan's iterative gate search + exhaustive 3/5/7-input LUT search, gates.xsd
XML persistence, C/CUDA/DOT codegen) built MI355X-first: the candidate scans
are hand-written CDNA4 (gfx950) HIP kernels, and multi-GPU runs use one
process per GPU coordinated with torch.distributed over RCCL/xGMI.

The native engine lives in the in-tree extension `_core` (host C++ +
HIP kernels). This package is orchestration: model (S-box) loading, op
wrappers, the distributed layer, and utilities.
"""

try:
    from . import _core
except ImportError as e:  # pragma: no cover
    raise ImportError(
        "sboxgates_amd native extension is not built. Run `make ext` in the "
        "repo root (hipcc cross-compiles gfx950 without a GPU). Original "
        f"error: {e}"
    ) from e

from ._core import (  # noqa: F401
    MAX_GATES,
    NO_GATE,
    Engine,
    Options,
    PyDistCtx,
    State,
    gpu_available,
    gpu_count,
)

__version__ = "1.0.0"
