"""Hot-op dispatch: hand-written CDNA4 HIP kernels with eager CPU references.

See `evotorch_amd/ops/dispatch.py` for the routing contract and
`evotorch_amd/ops/hip/` for the gfx950 kernel sources.
"""

from .dispatch import (
    affine_from_noise,
    cma_update_c_,
    clipup_step_,
    domination_counts,
    es_gradients,
    fused_adam_step_,
    hip_available,
    hip_required,
    load_hip,
    pareto_ranks,
    potrf_tile_,
    sample_gaussian,
    snes_gradients,
)

__all__ = [
    "affine_from_noise",
    "cma_update_c_",
    "clipup_step_",
    "es_gradients",
    "fused_adam_step_",
    "hip_available",
    "hip_required",
    "load_hip",
    "sample_gaussian",
    "snes_gradients",
    "pareto_ranks",
    "potrf_tile_",
    "domination_counts",
]
