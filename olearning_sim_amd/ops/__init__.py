"""HIP/CDNA4 kernels for the hot per-client ops, with CPU fallbacks.

The extension is built in-tree by ``python -m olearning_sim_amd.ops.build``
(or ``__graft_entry__.build()``) into ``olearning_sim_amd/ops/_hip_ops.so``
for gfx950 only.  On a GPU box the HIP path is mandatory: if a CUDA/HIP
tensor reaches one of these ops and the extension is missing, we raise —
a silent eager fallback on the GPU would invalidate every benchmark.
"""

from .fused import (
    hip_ops_available,
    load_hip_ops,
    fused_sgd_update,
    weighted_delta_accum,
    apply_aggregate,
    cross_entropy_fwd_bwd,
)

__all__ = [
    "hip_ops_available", "load_hip_ops", "fused_sgd_update",
    "weighted_delta_accum", "apply_aggregate", "cross_entropy_fwd_bwd",
]
