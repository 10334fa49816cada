"""Build the dcr_amd HIP extension for gfx950 (MI355X), in-tree.

`build_extension()` compiles dcr_amd/ops/hip/* with hipcc
(--offload-arch=gfx950 via PYTORCH_ROCM_ARCH) and places `_dcr_hip.so`
next to dcr_amd/ops/__init__.py so repo snapshots carry the binary
(gpurun ships in-tree .so files; a JIT cache under ~/.cache would not
travel). Driven by __graft_entry__.build().
"""
from __future__ import annotations

import os
import shutil
from pathlib import Path

HERE = Path(__file__).resolve().parent
HIP_DIR = HERE / "hip"
BUILD_DIR = HERE / "_build"
TARGET = HERE / "_dcr_hip.so"

SOURCES = [
    HIP_DIR / "bindings.cpp",
    HIP_DIR / "norms.hip",
    HIP_DIR / "norms_nhwc.hip",
    HIP_DIR / "elementwise.hip",
    HIP_DIR / "attention.hip",
    HIP_DIR / "conv_nhwc.hip",
    HIP_DIR / "conv_nhwc_bwd.hip",
    HIP_DIR / "gemm.hip",
]


def build_extension(verbose: bool = False, force: bool = False) -> Path:
    os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
    os.environ.setdefault("MAX_JOBS", str(min(16, os.cpu_count() or 4)))
    from torch.utils import cpp_extension

    BUILD_DIR.mkdir(parents=True, exist_ok=True)
    if force:
        for f in BUILD_DIR.glob("*"):
            if f.is_file():
                f.unlink()

    cpp_extension.load(
        name="_dcr_hip",
        sources=[str(s) for s in SOURCES],
        build_directory=str(BUILD_DIR),
        extra_cflags=["-O3"],
        extra_cuda_cflags=["-O3"],
        extra_include_paths=[str(HIP_DIR)],
        verbose=verbose,
        is_python_module=False,  # just build; we load from the copied path
    )
    built = BUILD_DIR / "_dcr_hip.so"
    if not built.exists():
        raise RuntimeError(f"build produced no {built}")
    shutil.copy2(built, TARGET)
    return TARGET


if __name__ == "__main__":
    print(build_extension(verbose=True))
