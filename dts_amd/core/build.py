"""Build the native core extension in-tree (CPU-only C++, no HIP)."""

from __future__ import annotations

import os
import shutil
from pathlib import Path

CORE_DIR = Path(__file__).resolve().parent
BUILD_DIR = CORE_DIR / "csrc" / "build"
TARGET = CORE_DIR / "_dts_core.so"


def build(verbose: bool = True) -> Path:
    """Set DTS_CORE_ASAN=1 to produce an AddressSanitizer build (run the
    test suite with LD_PRELOAD=libasan.so ASAN_OPTIONS=detect_leaks=0);
    profiles/asan_core.md records the clean pass."""
    os.environ.setdefault("MAX_JOBS", str(os.cpu_count() or 4))
    asan = os.environ.get("DTS_CORE_ASAN") == "1"
    ubsan = os.environ.get("DTS_CORE_UBSAN") == "1"
    if asan:
        build_dir = BUILD_DIR.parent / "build_asan"
    elif ubsan:
        build_dir = BUILD_DIR.parent / "build_ubsan"
    else:
        build_dir = BUILD_DIR
    build_dir.mkdir(parents=True, exist_ok=True)
    cflags = ["-O3", "-std=c++17"]
    ldflags = []
    if asan:
        cflags += ["-fsanitize=address", "-fno-omit-frame-pointer", "-g"]
        ldflags += ["-fsanitize=address"]
    if ubsan:
        # halt on the first UB report so the test run fails loudly
        cflags += [
            "-fsanitize=undefined",
            "-fno-sanitize-recover=all",
            "-fno-omit-frame-pointer",
            "-g",
        ]
        ldflags += ["-fsanitize=undefined"]
    from torch.utils.cpp_extension import load

    load(
        name="_dts_core",
        sources=[str(CORE_DIR / "csrc" / "core.cpp")],
        build_directory=str(build_dir),
        extra_cflags=cflags,
        extra_ldflags=ldflags,
        verbose=verbose,
        is_python_module=False,
        with_cuda=False,
    )
    built = build_dir / "_dts_core.so"
    if not built.exists():
        raise RuntimeError(f"build produced no {built}")
    shutil.copy2(built, TARGET)
    print(f"built {TARGET}")
    return TARGET


if __name__ == "__main__":
    build()
