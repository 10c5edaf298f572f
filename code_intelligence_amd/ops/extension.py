"""In-tree HIP extension loader for gfx950.

The extension is compiled IN-TREE (``code_intelligence_amd/ops/_ci_hip.so``)
so the built artifact travels with the repo snapshot to GPU boxes; no JIT
cache under ~/.cache is relied upon.

Policy (per project rules): on a ROCm GPU the HIP kernels are the ONLY
compute path — if the extension is missing on a GPU machine, ops raise
loudly instead of silently falling back to eager PyTorch. On CPU-only
machines the pure-PyTorch reference implementations run (used by the
non-gpu test suite and the plumbing config of BASELINE.json).
"""
from __future__ import annotations

import importlib.util
import os
import sys
from pathlib import Path

import torch

_OPS_DIR = Path(__file__).resolve().parent
_CSRC = _OPS_DIR / "csrc"
_EXT_NAME = "_ci_hip"

_ext = None
_tried = False

SOURCES = [
    "bindings.cpp",
    "lstm_pointwise.hip",
    "lstm_gemm.hip",
    "lstm_gemv.hip",
    "pool.hip",
    "qrnn_pool.hip",
    "adam.hip",
    "ce.hip",
    "embedding.hip",
    "dropconnect.hip",
    "artar.hip",
    "fp8util.hip",
    "tokenizer.cpp",
]


def _find_so() -> Path | None:
    for cand in sorted(_OPS_DIR.glob(f"{_EXT_NAME}*.so")):
        return cand
    return None


def _warn_if_stale(so: Path) -> None:
    """A silently-stale binary is the worst failure mode of in-tree builds:
    edited kernels that never run. Warn (once, at load) when any csrc file
    is newer than the .so."""
    try:
        so_m = so.stat().st_mtime
        newer = [f.name for f in _CSRC.iterdir()
                 if f.suffix in (".hip", ".cpp", ".h")
                 and not f.name.endswith("_hip.hip")
                 and f.stat().st_mtime > so_m]
        if newer:
            import warnings
            warnings.warn(
                f"_ci_hip.so is older than kernel sources {newer}; "
                "rebuild with python -m code_intelligence_amd.ops.build",
                RuntimeWarning, stacklevel=3)
    except OSError:
        pass


def load(required: bool = False):
    """Import the in-tree .so. required=True -> raise if absent."""
    global _ext, _tried
    if _ext is not None:
        return _ext
    if _tried and not required:
        return None
    _tried = True
    so = _find_so()
    if so is not None:
        _warn_if_stale(so)
    if so is None:
        if required:
            raise RuntimeError(
                "code_intelligence_amd HIP extension (_ci_hip*.so) not built. "
                "Run `python -m code_intelligence_amd.ops.build` (or "
                "__graft_entry__.build()). GPU ops refuse to run without the "
                "native gfx950 kernels — no silent eager fallback.")
        return None
    spec = importlib.util.spec_from_file_location(_EXT_NAME, so)
    mod = importlib.util.module_from_spec(spec)
    # torch must be imported first so the extension finds libtorch symbols
    spec.loader.exec_module(mod)
    _ext = mod
    sys.modules[_EXT_NAME] = mod
    return _ext


def have() -> bool:
    return load(required=False) is not None


def require():
    return load(required=True)


def on_gpu(*tensors: torch.Tensor) -> bool:
    return any(t.is_cuda for t in tensors if isinstance(t, torch.Tensor))


def build(verbose: bool = True) -> Path:
    """Compile every HIP source for gfx950 into the in-tree .so."""
    os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
    from torch.utils.cpp_extension import load as jit_load

    build_dir = _OPS_DIR / "_build"
    build_dir.mkdir(exist_ok=True)
    # drop stale hipify copies (cpp_extension writes <name>_hip.hip next to
    # sources; ninja compiles THOSE, so stale ones shadow edits)
    for stale in _CSRC.glob("*_hip.hip"):
        stale.unlink()
    for stale in _CSRC.glob("*.prehip"):
        stale.unlink()
    sources = [str(_CSRC / s) for s in SOURCES if (_CSRC / s).exists()]
    mod = jit_load(
        name=_EXT_NAME,
        sources=sources,
        extra_cflags=["-O3", "-std=c++17"],
        extra_cuda_cflags=["-O3", "-std=c++17", "--offload-arch=gfx950"],
        build_directory=str(build_dir),
        verbose=verbose,
        is_python_module=True,
    )
    # copy the built .so in-tree so it ships with the repo snapshot
    built = Path(mod.__file__)
    dest = _OPS_DIR / built.name
    if built.resolve() != dest.resolve():
        import shutil

        shutil.copy2(built, dest)
    global _ext
    _ext = mod
    return dest
