"""Build the in-tree HIP extension for gfx950.

Usage: python -m olearning_sim_amd.ops.build
Produces olearning_sim_amd/ops/_hip_ops.so (travels with the repo
snapshot to the GPU box; no JIT cache involved).
"""

from __future__ import annotations

import os
import shutil
import sys


def build(verbose: bool = True) -> str:
    os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
    here = os.path.dirname(os.path.abspath(__file__))
    csrc = os.path.join(here, "csrc")
    build_dir = os.path.join(here, "_build")
    os.makedirs(build_dir, exist_ok=True)
    sources = [os.path.join(csrc, f) for f in
               ("bindings.cpp", "fused_update.hip", "aggregate.hip",
                "cross_entropy.hip", "groupnorm.hip", "client_conv.hip",
                "client_conv2.hip", "client_conv5.hip", "pool2x2.hip", "transpose.hip", "layernorm.hip", "pad2d.hip",
               "replicate.hip")]
    from torch.utils.cpp_extension import load
    mod_path = load(
        name="olsim_hip_ops",
        sources=sources,
        extra_cflags=["-O3"],
        extra_cuda_cflags=["-O3"],
        build_directory=build_dir,
        is_python_module=False,
        verbose=verbose,
    )
    built = os.path.join(build_dir, "olsim_hip_ops.so")
    target = os.path.join(here, "_hip_ops.so")
    if os.path.exists(built):
        shutil.copy2(built, target)
    return target


if __name__ == "__main__":
    path = build()
    print(f"built: {path}", file=sys.stderr)
