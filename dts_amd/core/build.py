"""Build the native core extension in-tree (CPU-only C++, no HIP)."""

from __future__ import annotations

import os
import shutil
from pathlib import Path

CORE_DIR = Path(__file__).resolve().parent
BUILD_DIR = CORE_DIR / "csrc" / "build"
TARGET = CORE_DIR / "_dts_core.so"


def build(verbose: bool = True) -> Path:
    os.environ.setdefault("MAX_JOBS", str(os.cpu_count() or 4))
    BUILD_DIR.mkdir(parents=True, exist_ok=True)
    from torch.utils.cpp_extension import load

    load(
        name="_dts_core",
        sources=[str(CORE_DIR / "csrc" / "core.cpp")],
        build_directory=str(BUILD_DIR),
        extra_cflags=["-O3", "-std=c++17"],
        verbose=verbose,
        is_python_module=False,
        with_cuda=False,
    )
    built = BUILD_DIR / "_dts_core.so"
    if not built.exists():
        raise RuntimeError(f"build produced no {built}")
    shutil.copy2(built, TARGET)
    print(f"built {TARGET}")
    return TARGET


if __name__ == "__main__":
    build()
