"""Build the CDNA4 HIP extension in-tree.

Produces llmq_amd/ops/_hip_ops.so (tracked location, gitignored) so the
built artifact travels to GPU boxes with the repo snapshot. Cross-compiles
for gfx950 without a GPU (PYTORCH_ROCM_ARCH pins the arch).

Usage: python -m llmq_amd.ops.build [--verbose] [--force]
"""

from __future__ import annotations

import os
import shutil
import sys
from pathlib import Path

HERE = Path(__file__).parent
SRC = HERE / "hip" / "ops.hip"
OUT = HERE / "_hip_ops.so"
BUILD_DIR = HERE / "build"


def _sources_mtime() -> float:
    return max(p.stat().st_mtime for p in (HERE / "hip").iterdir() if p.is_file())


def build(verbose: bool = False, force: bool = False) -> Path:
    if OUT.is_file() and not force and OUT.stat().st_mtime >= _sources_mtime():
        return OUT
    # torch.utils.cpp_extension tracks only the listed sources (ops.hip), not
    # the files it #includes — touch the umbrella TU so edits to any kernel
    # file trigger a rebuild.
    SRC.touch()
    if force:
        shutil.rmtree(BUILD_DIR, ignore_errors=True)
    os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
    os.environ.setdefault("MAX_JOBS", "8")
    from torch.utils.cpp_extension import load

    BUILD_DIR.mkdir(exist_ok=True)

    def _load() -> None:
        load(
            name="llmq_amd_hip_ops",
            sources=[str(SRC)],
            extra_cflags=["-O3", "-std=c++20"],
            extra_cuda_cflags=["-O3", "-std=c++20"],
            build_directory=str(BUILD_DIR),
            is_python_module=False,
            verbose=verbose,
        )

    try:
        _load()
    except Exception:
        # A stale incremental object (e.g. after a template-signature change)
        # can link a .so with dangling device stubs — rebuild clean once.
        shutil.rmtree(BUILD_DIR, ignore_errors=True)
        BUILD_DIR.mkdir(exist_ok=True)
        _load()
    built = BUILD_DIR / "llmq_amd_hip_ops.so"
    if not built.is_file():
        candidates = list(BUILD_DIR.glob("*.so"))
        if not candidates:
            raise RuntimeError(f"build produced no .so under {BUILD_DIR}")
        built = candidates[0]
    shutil.copy2(built, OUT)
    return OUT


def build_ab_defer_on(verbose: bool = False) -> Path:
    """A/B artifact: same kernels with -DLLMQ_DEFER_ON (the T13 defer-max
    experiment — measured a LOSS, see profiles/r2_step5). Output name
    deliberately does NOT match the `_hip_ops*.so` auto-load glob —
    select it with LLMQ_OPS_SO=<path>."""
    out = HERE / "_ab_defer_on.so"
    if out.is_file() and out.stat().st_mtime >= _sources_mtime():
        return out
    SRC.touch()
    os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
    os.environ.setdefault("MAX_JOBS", "8")
    from torch.utils.cpp_extension import load

    bdir = HERE / "build_ab"
    shutil.rmtree(bdir, ignore_errors=True)
    bdir.mkdir(exist_ok=True)
    load(
        name="llmq_amd_hip_ops_ab",
        sources=[str(SRC)],
        extra_cflags=["-O3", "-std=c++20", "-DLLMQ_DEFER_ON"],
        extra_cuda_cflags=["-O3", "-std=c++20", "-DLLMQ_DEFER_ON"],
        build_directory=str(bdir),
        is_python_module=False,
        verbose=verbose,
    )
    built = next(bdir.glob("*.so"))
    shutil.copy2(built, out)
    return out


if __name__ == "__main__":
    if "--ab-defer-on" in sys.argv:
        out = build_ab_defer_on(verbose="--verbose" in sys.argv)
    else:
        out = build(verbose="--verbose" in sys.argv, force="--force" in sys.argv)
    print(f"built {out}")
