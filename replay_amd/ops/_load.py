"""Loader for the in-tree HIP extension (gfx950).

The extension is built in-tree by ``__graft_entry__.build()`` (so the .so
travels to the GPU box with the repo snapshot) via torch.utils.cpp_extension
with PYTORCH_ROCM_ARCH=gfx950.
"""

from __future__ import annotations

import importlib.util
import os
from pathlib import Path

_HERE = Path(__file__).parent
EXT_NAME = "replay_amd_hip"


def _so_path() -> Path:
    return _HERE / f"{EXT_NAME}.so"


def load_extension():
    so = _so_path()
    if not so.exists():
        raise FileNotFoundError(so)
    import torch  # noqa: F401  (registers torch symbols the .so needs)

    spec = importlib.util.spec_from_file_location(EXT_NAME, so)
    module = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(module)
    return module


def build_extension(verbose: bool = True):
    """Compile the HIP sources for gfx950 into replay_amd/ops/replay_amd_hip.so."""
    os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
    from torch.utils.cpp_extension import load

    src_dir = _HERE / "hip"
    # exclude torch-hipify's generated *_hip.hip copies of our sources
    sources = sorted(
        str(p) for p in src_dir.glob("*.hip") if not p.name.endswith("_hip.hip")
    ) + sorted(str(p) for p in src_dir.glob("*.cpp"))
    if not sources:
        raise RuntimeError(f"No HIP sources under {src_dir}")
    build_dir = _HERE / "build"
    build_dir.mkdir(exist_ok=True)
    module = load(
        name=EXT_NAME,
        sources=sources,
        extra_cflags=["-O3"],
        extra_cuda_cflags=["-O3", "--offload-arch=gfx950", "-std=c++17"],
        build_directory=str(build_dir),
        verbose=verbose,
    )
    # copy the built .so in-tree so it snapshots to the GPU box
    built = build_dir / f"{EXT_NAME}.so"
    if built.exists():
        import shutil

        shutil.copy2(built, _so_path())
    return module
