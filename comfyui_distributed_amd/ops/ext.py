"""HIP extension build/load.

The extension is built IN-TREE (``comfyui_distributed_amd/ops/_distgpu_hip.so``)
so the binary travels with the source checkout to GPU boxes. Building uses
torch's cpp_extension driving hipcc with ``--offload-arch=gfx950``; the
sources are native HIP (no hipify rewrites — the pass is a verified no-op on
them).

Loading policy (loud-failure, per the framework contract):
* on a machine with a GPU, a missing/broken extension raises
  :class:`KernelUnavailableError` — ops never silently fall back to eager;
* on CPU-only machines (CI) the Python fallbacks in dispatch.py are used.
"""

from __future__ import annotations

import os
from pathlib import Path

import torch

from ..utils.errors import KernelUnavailableError

_HERE = Path(__file__).resolve().parent
HIP_DIR = _HERE / "hip"
EXT_NAME = "_distgpu_hip"
SO_PATH = _HERE / f"{EXT_NAME}.so"

SOURCES = [
    "bindings.cpp",
    "norms.hip",
    "attention.hip",
    "tile_ops.hip",
    "mfma_selftest.hip",
    "conv.hip",
    "gemm.hip",
]

_ext = None
_load_error: Exception | None = None


def build(verbose: bool = True) -> Path:
    """Compile the extension for gfx950 into the package tree."""
    os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
    from torch.utils.cpp_extension import load

    sources = [str(HIP_DIR / s) for s in SOURCES if (HIP_DIR / s).exists()]
    build_dir = _HERE / "build"
    build_dir.mkdir(exist_ok=True)
    mod = load(
        name=EXT_NAME,
        sources=sources,
        build_directory=str(build_dir),
        extra_cuda_cflags=["-O3", "--offload-arch=gfx950"],
        verbose=verbose,
    )
    built = build_dir / f"{EXT_NAME}.so"
    if built.exists():
        import shutil

        shutil.copy2(built, SO_PATH)
    global _ext
    _ext = mod
    return SO_PATH


def _import_prebuilt():
    import importlib.util

    spec = importlib.util.spec_from_file_location(EXT_NAME, SO_PATH)
    mod = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(mod)
    return mod


def get_ext(required: bool | None = None):
    """Return the loaded extension module.

    ``required=None`` resolves to ``torch.cuda.is_available()``: on a GPU box
    the HIP path is mandatory.
    """
    global _ext, _load_error
    if _ext is not None:
        return _ext
    if required is None:
        required = torch.cuda.is_available()
    if _load_error is None and SO_PATH.exists():
        try:
            _ext = _import_prebuilt()
            return _ext
        except Exception as exc:  # noqa: BLE001
            _load_error = exc
    if required:
        raise KernelUnavailableError(
            f"HIP extension {EXT_NAME} is required on a GPU host but could not "
            f"be loaded (so={SO_PATH}, exists={SO_PATH.exists()}, "
            f"error={_load_error!r}). Build it with "
            f"`python -c 'import __graft_entry__; __graft_entry__.build()'`."
        )
    return None
