"""Build-infrastructure analogue of the reference's ``common.jit`` package
(reference common/jit/{core,cpp_ext,env,utils}.py: a JIT compile-and-cache
pipeline for its CUDA extensions, consumed by _flex_flash_attn_jit.py).

The MI355X rebuild compiles ahead-of-time into ONE in-tree C-ABI library
(csrc/build.py -> magi_attention/_libs/libmagi_ffa.so, hipcc
--offload-arch=gfx950), so there is no runtime JIT cache: kernels are
selected by launcher dispatch inside the library, not by per-variant
compilation. This module exposes the operations a jit-cache user actually
needs — (re)build and locate the library — under the reference's module
name for surface parity.
"""
from __future__ import annotations

from pathlib import Path


def build(force: bool = False, verbose: bool = True) -> Path:
    """Compile (if stale) the native library; returns its path.
    AOT analogue of the reference's gen_jit_spec(...).build()."""
    from ..csrc.build import build as _build

    return _build(force=force, verbose=verbose)


def get_lib_path() -> Path:
    """Path of the built C-ABI library (may not exist yet; call build())."""
    from ..csrc.build import LIB

    return LIB


def is_built() -> bool:
    return get_lib_path().exists()


__all__ = ["build", "get_lib_path", "is_built"]
