"""In-tree build of the gfx950 HIP extension.

``python -m distegnn_amd.ops.build`` compiles csrc/*.hip + ext.cpp with
hipcc (--offload-arch=gfx950 via PYTORCH_ROCM_ARCH) into
``distegnn_amd/ops/_hip_ext.so``. The .so lives IN THE TREE so the gpurun
snapshot carries it to the GPU box (a JIT cache under ~/.cache would not
travel). hipcc cross-compiles fine on a CPU-only box.
"""

from __future__ import annotations

import os
import shutil
import sys

_OPS_DIR = os.path.dirname(os.path.abspath(__file__))
_CSRC = os.path.join(_OPS_DIR, "csrc")
_BUILD = os.path.join(_CSRC, "build")

SOURCES = [
    os.path.join(_CSRC, "ext.cpp"),
    os.path.join(_CSRC, "segment_reduce.hip"),
    os.path.join(_CSRC, "radius.hip"),
    os.path.join(_CSRC, "fused_edge.hip"),
    os.path.join(_CSRC, "fused_edge_bwd.hip"),
    os.path.join(_CSRC, "wgrad.hip"),
    os.path.join(_CSRC, "tall_linear.hip"),
    os.path.join(_CSRC, "fused_virtual.hip"),
    os.path.join(_CSRC, "elementwise.hip"),
    os.path.join(_CSRC, "cfconv.hip"),
]


def build(verbose: bool = False) -> str:
    os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
    os.makedirs(_BUILD, exist_ok=True)
    from torch.utils.cpp_extension import load

    load(
        name="_hip_ext",
        sources=SOURCES,
        build_directory=_BUILD,
        extra_cflags=["-O3"],
        extra_cuda_cflags=["-O3", "-std=c++17"],
        verbose=verbose,
        is_python_module=False,
        keep_intermediates=True,
    )
    built = os.path.join(_BUILD, "_hip_ext.so")
    target = os.path.join(_OPS_DIR, "_hip_ext.so")
    shutil.copy2(built, target)
    return target


if __name__ == "__main__":
    path = build(verbose="-v" in sys.argv)
    print(f"built {path}")
