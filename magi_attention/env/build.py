"""Build env flags (reference surface: env/build.py — knobs of ITS JIT
compile cache). This rebuild compiles ahead-of-time (csrc/build.py, hipcc);
the names map onto the AOT build: FORCE_JIT_BUILD / NO_BUILD_CACHE both
force a recompile, WORKSPACE_BASE relocates nothing (the library is
in-tree so it travels with the repo), and nvcc_threads reports the
parallelism hipcc is invoked with (single translation-unit link)."""
from __future__ import annotations

import os

from . import _get, _get_bool


def is_no_build_cache() -> bool:
    return _get_bool("MAGI_ATTENTION_NO_BUILD_CACHE")


def workspace_base_dir() -> str:
    base = _get("MAGI_ATTENTION_WORKSPACE_BASE", "")
    if base:
        return base
    from ..csrc.build import LIBDIR

    return str(LIBDIR)


def is_force_jit_build() -> bool:
    return _get_bool("MAGI_ATTENTION_FORCE_JIT_BUILD")


def is_build_verbose() -> bool:
    return _get_bool("MAGI_ATTENTION_BUILD_VERBOSE")


def is_build_debug() -> bool:
    return _get_bool("MAGI_ATTENTION_BUILD_DEBUG")


def nvcc_threads() -> str:
    return os.environ.get("MAGI_ATTENTION_NVCC_THREADS", "4")
