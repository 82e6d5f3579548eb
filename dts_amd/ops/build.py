"""Build the HIP extension IN-TREE for gfx950.

`python -m dts_amd.ops.build` compiles dts_amd/ops/hip/*.hip with hipcc
(PYTORCH_ROCM_ARCH=gfx950 — cross-compiles fine on a GPU-less box) and
drops `_dts_hip.so` next to this file so the repo snapshot carries it to
the GPU box. Called by __graft_entry__.build().
"""

from __future__ import annotations

import os
import shutil
from pathlib import Path

OPS_DIR = Path(__file__).resolve().parent
HIP_DIR = OPS_DIR / "hip"
BUILD_DIR = HIP_DIR / "build"
TARGET = OPS_DIR / "_dts_hip.so"


def build(verbose: bool = True) -> Path:
    os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
    os.environ.setdefault("MAX_JOBS", str(os.cpu_count() or 4))
    BUILD_DIR.mkdir(parents=True, exist_ok=True)

    from torch.utils.cpp_extension import load

    sources = [
        str(HIP_DIR / "bindings.cpp"),
        str(HIP_DIR / "elementwise.hip"),
        str(HIP_DIR / "attention.hip"),
        str(HIP_DIR / "sampling.hip"),
        str(HIP_DIR / "gemv.hip"),
        str(HIP_DIR / "gemm_skinny.hip"),
    ]
    module = load(
        name="_dts_hip",
        sources=sources,
        build_directory=str(BUILD_DIR),
        extra_cflags=["-O3"],
        extra_cuda_cflags=["-O3", "--offload-arch=gfx950"],
        verbose=verbose,
        is_python_module=False,
        with_cuda=True,
    )
    built = BUILD_DIR / "_dts_hip.so"
    if not built.exists():
        raise RuntimeError(f"build produced no {built}")
    shutil.copy2(built, TARGET)
    print(f"built {TARGET}")
    return TARGET


if __name__ == "__main__":
    build()
