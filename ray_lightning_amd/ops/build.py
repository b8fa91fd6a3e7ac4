"""In-tree build of the HIP extensions for gfx950.

Builds two modules into ``ray_lightning_amd/ops/``:
- ``_hip_ops``    — CDNA4 kernels (pack/scale/fused optimizers)
- ``_rccl_comm``  — native RCCL communicator

Cross-compiles without a GPU (PYTORCH_ROCM_ARCH=gfx950); the built .so
files live in-tree so they travel with the repo snapshot to GPU boxes.

Run: ``python -m ray_lightning_amd.ops.build``
"""
from __future__ import annotations

import os
import shutil
import sys

_OPS_DIR = os.path.dirname(os.path.abspath(__file__))
_CSRC = os.path.join(_OPS_DIR, "csrc")
_BUILD = os.path.join(_OPS_DIR, "_build")


def build(verbose: bool = False) -> None:
    os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
    os.environ.setdefault("MAX_JOBS", "8")
    from torch.utils import cpp_extension

    os.makedirs(_BUILD, exist_ok=True)
    common_cflags = ["-O3", "-std=c++17"]

    for name, sources, ldflags in (
        ("_hip_ops", [os.path.join(_CSRC, "hip_ops.hip"),
                      os.path.join(_CSRC, "fused_bn.hip"),
                      os.path.join(_CSRC, "fused_ln.hip"),
                      os.path.join(_CSRC, "flash_attn.hip"),
                      os.path.join(_CSRC, "mfma_probe.hip")], []),
        ("_rccl_comm", [os.path.join(_CSRC, "rccl_comm.hip")],
         ["-L/opt/rocm/lib", "-lrccl"]),
        ("_lt_mlp", [os.path.join(_CSRC, "lt_mlp.hip")],
         ["-L/opt/rocm/lib", "-lhipblaslt"]),
    ):
        build_dir = os.path.join(_BUILD, name)
        os.makedirs(build_dir, exist_ok=True)
        cpp_extension.load(
            name=name,
            sources=sources,
            extra_cflags=common_cflags,
            extra_cuda_cflags=common_cflags,
            extra_ldflags=ldflags,
            extra_include_paths=["/opt/rocm/include"],
            build_directory=build_dir,
            verbose=verbose,
            is_python_module=False,
            keep_intermediates=True,
        )
        so = os.path.join(build_dir, f"{name}.so")
        if not os.path.exists(so):
            raise RuntimeError(f"build produced no {so}")
        shutil.copy2(so, os.path.join(_OPS_DIR, f"{name}.so"))
        print(f"[ops.build] built {name}.so "
              f"({os.path.getsize(so) // 1024} KiB)")


if __name__ == "__main__":
    build(verbose="-v" in sys.argv)
