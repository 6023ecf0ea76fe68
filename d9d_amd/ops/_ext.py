"""In-tree HIP extension loader.

The extension is built IN-TREE (d9d_amd/csrc/_build/) so the .so ships to GPU
boxes with the repo snapshot. `build()` (cross-)compiles for gfx950 — works on
CPU-only machines; `get_ext()` imports the prebuilt .so without any rebuild
checks so a GPU box never recompiles.

On a GPU, ops FAIL LOUDLY if the extension is missing — there is no silent
eager fallback on the HIP path (CPU keeps a pure-torch reference path for
tests only).
"""

import importlib.util
import os
import sys
from pathlib import Path

_CSRC = Path(__file__).resolve().parent.parent / "csrc"
_BUILD_DIR = _CSRC / "_build"
_EXT_NAME = "d9d_amd_kernels"

_SOURCES = [
    "bindings.cpp",
    "rms_norm.hip",
    "silu_mul.hip",
    "stochastic.hip",
    "rope.hip",
    "moe_permute.hip",
    "moe_router.hip",
    "gmm.hip",
    "attention.hip",
    "cce.hip",
    "gdn.hip",
]

_ext_module = None


def _so_path() -> Path:
    return _BUILD_DIR / f"{_EXT_NAME}.so"


def build(verbose: bool = True):
    """Compile the extension for gfx950 (cross-compiles fine without a GPU)."""
    global _ext_module
    os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
    from torch.utils import cpp_extension

    _BUILD_DIR.mkdir(parents=True, exist_ok=True)
    sources = [str(_CSRC / s) for s in _SOURCES if (_CSRC / s).exists()]
    module = cpp_extension.load(
        name=_EXT_NAME,
        sources=sources,
        build_directory=str(_BUILD_DIR),
        extra_cflags=["-O3", "-std=c++17"],
        extra_cuda_cflags=["-O3", "-std=c++17", "--offload-arch=gfx950"],
        verbose=verbose,
        is_python_module=True,
    )
    _ext_module = module
    return module


def _load_prebuilt():
    global _ext_module
    if _ext_module is not None:
        return _ext_module
    so = _so_path()
    if not so.exists():
        return None
    import torch  # noqa: F401  — the .so resolves symbols against libtorch
    spec = importlib.util.spec_from_file_location(_EXT_NAME, so)
    module = importlib.util.module_from_spec(spec)
    try:
        spec.loader.exec_module(module)
    except ImportError:
        return None
    sys.modules[_EXT_NAME] = module
    _ext_module = module
    return module


def get_ext(required: bool = True):
    """Return the compiled extension module; on a GPU this must exist."""
    module = _load_prebuilt()
    if module is None and required:
        raise RuntimeError(
            "d9d_amd HIP extension not built. Run `python -c "
            '"from d9d_amd.ops import _ext; _ext.build()"` (or __graft_entry__.build()) '
            "before running on a GPU — there is no eager fallback on the HIP path."
        )
    return module


def has_ext() -> bool:
    return _load_prebuilt() is not None
