"""Build the HIP extension IN-TREE for gfx950.

``python -m murmura_amd.ops.build`` compiles murmura_kernels.hip with hipcc
(via torch.utils.cpp_extension, PYTORCH_ROCM_ARCH=gfx950) and places
``_murmura_hip.so`` next to this file so it travels to the GPU box with the
repo snapshot (never a JIT cache under ~/.cache). hipcc cross-compiles without
a GPU, so this runs in the CPU-only container too.
"""

from __future__ import annotations

import os
import shutil
import sys
from pathlib import Path

PKG_DIR = Path(__file__).parent
SRC = PKG_DIR / "hip" / "murmura_kernels.hip"
BUILD_DIR = PKG_DIR / "_build"
OUT_SO = PKG_DIR / "_murmura_hip.so"


def build(verbose: bool = False) -> Path:
    os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
    os.environ.setdefault("MAX_JOBS", "8")
    from torch.utils import cpp_extension

    BUILD_DIR.mkdir(exist_ok=True)
    # stale-lock cleanup (a previously interrupted build leaves 'lock')
    lock = BUILD_DIR / "lock"
    if lock.exists():
        lock.unlink()
    cpp_extension.load(
        name="_murmura_hip",
        sources=[str(SRC)],
        build_directory=str(BUILD_DIR),
        extra_cuda_cflags=["-O3"],
        verbose=verbose,
        is_python_module=False,  # just build; we copy + import ourselves
    )
    built = BUILD_DIR / "_murmura_hip.so"
    if not built.exists():
        raise RuntimeError(f"build produced no {built}")
    shutil.copy2(built, OUT_SO)
    return OUT_SO


if __name__ == "__main__":
    so = build(verbose="-v" in sys.argv)
    print(f"built {so}")
