"""In-tree build of the gfx950 HIP extension.

Builds mpgcn_amd/ops/hip/*.hip with hipcc (via torch.utils.cpp_extension, which
drives hipcc for .hip sources under PYTORCH_ROCM_ARCH=gfx950) into
mpgcn_amd/ops/_build/_mpgcn_hip.so. The .so stays in-tree so it travels to GPU
boxes with a source snapshot; the loader (mpgcn_amd.ops) imports it directly
without re-invoking the build system.

Run:  python -m mpgcn_amd.ops.build
"""

from __future__ import annotations

import os
from pathlib import Path

HIP_DIR = Path(__file__).resolve().parent / "hip"
BUILD_DIR = Path(__file__).resolve().parent / "_build"
EXT_NAME = "_mpgcn_hip"
SOURCES = ["ext.hip", "axis_gemm.hip", "row_gemm.hip", "lstm.hip", "lstm_fused.hip", "red_gemm.hip", "elemwise.hip"]


def so_path() -> Path:
    return BUILD_DIR / f"{EXT_NAME}.so"


def needs_rebuild() -> bool:
    so = so_path()
    if not so.exists():
        return True
    so_mtime = so.stat().st_mtime
    deps = [HIP_DIR / s for s in SOURCES] + list(HIP_DIR.glob("*.hpp"))
    return any(d.stat().st_mtime > so_mtime for d in deps)


def build(verbose: bool = True) -> Path:
    os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
    # hipcc cross-compiles for gfx950 without a GPU present.
    os.environ.setdefault("TORCH_DONT_CHECK_COMPILER_ABI", "1")
    from torch.utils import cpp_extension

    BUILD_DIR.mkdir(parents=True, exist_ok=True)
    # ninja's depfiles miss header deps through torch's hipify step (a .hpp
    # edit silently left stale .o files linked into a "fresh" .so — measured
    # the hard way); clear the objects whenever any header is newer than one
    hpp_m = max((h.stat().st_mtime for h in HIP_DIR.glob("*.hpp")), default=0)
    for o in BUILD_DIR.glob("*.o"):
        if o.stat().st_mtime < hpp_m:
            for f in list(BUILD_DIR.glob("*.o")) + list(HIP_DIR.glob("*_hip.hip")):
                f.unlink(missing_ok=True)
            break
    cpp_extension.load(
        name=EXT_NAME,
        sources=[str(HIP_DIR / s) for s in SOURCES],
        build_directory=str(BUILD_DIR),
        extra_cflags=["-O3", "-std=c++17"],
        extra_cuda_cflags=["-O3", "-std=c++17"],
        verbose=verbose,
        is_python_module=False,  # just build; loader imports the .so itself
        keep_intermediates=True,
    )
    assert so_path().exists(), f"build produced no {so_path()}"
    return so_path()


if __name__ == "__main__":
    p = build()
    print(f"built {p}")
