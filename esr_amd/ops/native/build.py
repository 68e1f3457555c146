"""In-tree build of the gfx950 HIP extension.

Usage:  python -m esr_amd.ops.native.build

Builds esr_amd/ops/native/_esr_native.so with hipcc
(--offload-arch=gfx950) through torch.utils.cpp_extension so that the .so
travels with repo snapshots to GPU boxes (a JIT cache under ~/.cache would
not).  hipcc cross-compiles fine on CPU-only boxes.
"""

from __future__ import annotations

import os
import shutil
import sys
from pathlib import Path

HERE = Path(__file__).resolve().parent
SRC = HERE / "src"
SOURCES = [SRC / "bindings.cpp", SRC / "deform_conv.hip",
           SRC / "deform_conv_fused.hip", SRC / "gru_gates.hip",
           SRC / "event_ops.hip", SRC / "conv2d.hip", SRC / "redistribute.hip"]


def build(verbose: bool = False) -> str:
    os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
    os.environ.setdefault("MAX_JOBS", "8")
    from torch.utils.cpp_extension import load

    build_dir = HERE / "_build"
    build_dir.mkdir(exist_ok=True)
    mod = load(
        name="_esr_native",
        sources=[str(s) for s in SOURCES],
        extra_cflags=["-O3"],
        extra_cuda_cflags=["-O3", "--offload-arch=gfx950"],
        build_directory=str(build_dir),
        verbose=verbose,
        is_python_module=False,  # just build; we copy + import by path
    )
    so = build_dir / "_esr_native.so"
    if not so.exists():
        cands = list(build_dir.glob("_esr_native*.so"))
        if not cands:
            raise RuntimeError("extension build produced no .so")
        so = cands[0]
    dest = HERE / "_esr_native.so"
    shutil.copy2(so, dest)
    return str(dest)


if __name__ == "__main__":
    path = build(verbose="-v" in sys.argv)
    print(f"built {path}")
