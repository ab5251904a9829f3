"""ctypes binding of the native C-ABI library (include/magi_ffa.h).

This is the host side of the drop-in kernel-module boundary (the role the
reference's JIT'd pybind module plays, flex_flash_attn.py:261-290). The HIP
extension is REQUIRED on a GPU box: any attempt to run a compute op without it
raises, never falls back to a CPU path.
"""
from __future__ import annotations

import ctypes
import os
from pathlib import Path

import torch

_LIB_PATH = Path(__file__).resolve().parent / "_libs" / "libmagi_ffa.so"
_lib: ctypes.CDLL | None = None
_load_error: Exception | None = None


class MagiFfaFwdArgs(ctypes.Structure):
    _fields_ = [
        ("q", ctypes.c_void_p),
        ("k", ctypes.c_void_p),
        ("v", ctypes.c_void_p),
        ("out", ctypes.c_void_p),
        ("lse", ctypes.c_void_p),
        ("q_ranges", ctypes.c_void_p),
        ("k_ranges", ctypes.c_void_p),
        ("attn_type_map", ctypes.c_void_p),
        ("locks", ctypes.c_void_p),
        ("max_logits", ctypes.c_void_p),
        ("qk_starts", ctypes.c_void_p),
        ("n_ranges", ctypes.c_int64),
        ("total_q", ctypes.c_int64),
        ("total_k", ctypes.c_int64),
        ("hq", ctypes.c_int32),
        ("hk", ctypes.c_int32),
        ("d", ctypes.c_int32),
        ("max_seqlen_q", ctypes.c_int32),
        ("softmax_scale", ctypes.c_float),
        ("softcap", ctypes.c_float),
        ("out_is_fp32", ctypes.c_int32),
        ("disable_atomic_reduction", ctypes.c_int32),
        ("cu_margin", ctypes.c_int32),
        ("stream", ctypes.c_void_p),
    ]


class MagiFfaBwdArgs(ctypes.Structure):
    _fields_ = [
        ("dout", ctypes.c_void_p),
        ("q", ctypes.c_void_p),
        ("k", ctypes.c_void_p),
        ("v", ctypes.c_void_p),
        ("out", ctypes.c_void_p),
        ("lse", ctypes.c_void_p),
        ("dq", ctypes.c_void_p),
        ("dk", ctypes.c_void_p),
        ("dv", ctypes.c_void_p),
        ("dpsum", ctypes.c_void_p),
        ("q_ranges", ctypes.c_void_p),
        ("k_ranges", ctypes.c_void_p),
        ("attn_type_map", ctypes.c_void_p),
        ("seg_starts", ctypes.c_void_p),
        ("n_ranges", ctypes.c_int64),
        ("total_q", ctypes.c_int64),
        ("total_k", ctypes.c_int64),
        ("hq", ctypes.c_int32),
        ("hk", ctypes.c_int32),
        ("d", ctypes.c_int32),
        ("max_seqlen_k", ctypes.c_int32),
        ("out_is_fp32", ctypes.c_int32),
        ("softmax_scale", ctypes.c_float),
        ("softcap", ctypes.c_float),
        ("cu_margin", ctypes.c_int32),
        ("stream", ctypes.c_void_p),
    ]


class MagiRangeOpArgs(ctypes.Structure):
    _fields_ = [
        ("input", ctypes.c_void_p),
        ("output", ctypes.c_void_p),
        ("in_ranges", ctypes.c_void_p),
        ("out_starts", ctypes.c_void_p),
        ("in_lse", ctypes.c_void_p),
        ("out_lse", ctypes.c_void_p),
        ("n_ranges", ctypes.c_int64),
        ("row_elems", ctypes.c_int64),
        ("total_rows", ctypes.c_int64),
        ("elem_size", ctypes.c_int32),
        ("n_heads", ctypes.c_int32),
        ("reduce_op", ctypes.c_int32),
        ("stream", ctypes.c_void_p),
    ]


class MagiSinkArgs(ctypes.Structure):
    _fields_ = [
        ("out", ctypes.c_void_p),
        ("lse", ctypes.c_void_p),
        ("sink", ctypes.c_void_p),
        ("dsink", ctypes.c_void_p),
        ("dpsum", ctypes.c_void_p),
        ("total_rows", ctypes.c_int64),
        ("n_heads", ctypes.c_int32),
        ("d", ctypes.c_int32),
        ("s_sink", ctypes.c_int32),
        ("ssh", ctypes.c_int32),
        ("out_is_fp32", ctypes.c_int32),
        ("stream", ctypes.c_void_p),
    ]


class MagiGrpCollPullArgs(ctypes.Structure):
    """mirrors magi_grpcoll_pull_args in csrc/grpcoll.hip (native grpcoll)"""

    _fields_ = [
        ("pieces", ctypes.c_void_p),
        ("n_pieces", ctypes.c_int32),
        ("row_elems", ctypes.c_int32),
        ("elem_size", ctypes.c_int32),
        ("peer_ptrs", ctypes.c_void_p * 8),
        ("peer_flags", ctypes.c_void_p * 8),
        ("wait_value", ctypes.c_int32),
        ("n_peers", ctypes.c_int32),
        ("dst", ctypes.c_void_p),
        ("reduce", ctypes.c_int32),
        ("stream", ctypes.c_void_p),
    ]


class MagiFfaIndexArgs(ctypes.Structure):
    """mirrors magi_ffa_index_args (index-attention token-gather forward)"""

    _fields_ = [
        ("q", ctypes.c_void_p),
        ("k", ctypes.c_void_p),
        ("v", ctypes.c_void_p),
        ("out", ctypes.c_void_p),
        ("lse", ctypes.c_void_p),
        ("indices_2d", ctypes.c_void_p),
        ("total_q", ctypes.c_int64),
        ("total_k", ctypes.c_int64),
        ("max_topk", ctypes.c_int32),
        ("hq", ctypes.c_int32),
        ("hk", ctypes.c_int32),
        ("d", ctypes.c_int32),
        ("softmax_scale", ctypes.c_float),
        ("softcap", ctypes.c_float),
        ("out_is_fp32", ctypes.c_int32),
        ("stream", ctypes.c_void_p),
    ]


class MagiCorrectArgs(ctypes.Structure):
    _fields_ = [
        ("out1", ctypes.c_void_p),
        ("lse1", ctypes.c_void_p),
        ("out2", ctypes.c_void_p),
        ("lse2", ctypes.c_void_p),
        ("total_rows", ctypes.c_int64),
        ("n_heads", ctypes.c_int32),
        ("d", ctypes.c_int32),
        ("stream", ctypes.c_void_p),
    ]


def _try_load() -> ctypes.CDLL | None:
    global _lib, _load_error
    if _lib is not None:
        return _lib
    try:
        lib = ctypes.CDLL(str(_LIB_PATH))
        for name, argtypes in [
            ("magi_ffa_fwd", [ctypes.POINTER(MagiFfaFwdArgs)]),
            ("magi_ffa_fwd_fp8", [ctypes.POINTER(MagiFfaFwdArgs)]),
            ("magi_ffa_fwd_index", [ctypes.POINTER(MagiFfaIndexArgs)]),
            ("magi_ffa_bwd", [ctypes.POINTER(MagiFfaBwdArgs)]),
            ("magi_ffa_bwd_dq", [ctypes.POINTER(MagiFfaBwdArgs)]),
            ("magi_ffa_bwd_dkv", [ctypes.POINTER(MagiFfaBwdArgs)]),
            ("magi_ffa_bwd_dv", [ctypes.POINTER(MagiFfaBwdArgs)]),
            ("magi_ffa_bwd_dk", [ctypes.POINTER(MagiFfaBwdArgs)]),
            ("magi_ffa_bwd_preprocess", [ctypes.POINTER(MagiFfaBwdArgs)]),
            ("magi_range_gather", [ctypes.POINTER(MagiRangeOpArgs)]),
            ("magi_range_reduce", [ctypes.POINTER(MagiRangeOpArgs)]),
            ("magi_correct_out_lse", [ctypes.POINTER(MagiCorrectArgs)]),
            ("magi_ffa_sink_postprocess", [ctypes.POINTER(MagiSinkArgs)]),
            ("magi_ffa_dsink", [ctypes.POINTER(MagiSinkArgs)]),
            (
                "magi_probe_mfma",
                [ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p],
            ),
            ("magi_ipc_get_handle", [ctypes.c_void_p, ctypes.c_void_p]),
            ("magi_ipc_open",
             [ctypes.c_void_p, ctypes.POINTER(ctypes.c_void_p)]),
            ("magi_ipc_close", [ctypes.c_void_p]),
            ("magi_grpcoll_signal",
             [ctypes.c_void_p, ctypes.c_int32, ctypes.c_void_p]),
            ("magi_grpcoll_wait",
             [ctypes.c_void_p, ctypes.c_int32, ctypes.c_void_p]),
            ("magi_grpcoll_ack",
             [ctypes.c_void_p, ctypes.c_int32, ctypes.c_void_p]),
            ("magi_grpcoll_pull", [ctypes.POINTER(MagiGrpCollPullArgs)]),
            ("magi_ipc_base",
             [ctypes.c_void_p, ctypes.POINTER(ctypes.c_void_p),
              ctypes.POINTER(ctypes.c_uint64)]),
        ]:
            fn = getattr(lib, name)
            fn.argtypes = argtypes
            fn.restype = ctypes.c_int
        lib.magi_ffa_abi_version.restype = ctypes.c_int
        _lib = lib
    except OSError as e:  # missing .so or missing HIP runtime
        _load_error = e
        return None
    return _lib


def lib() -> ctypes.CDLL:
    l = _try_load()
    if l is None:
        raise RuntimeError(
            f"magi_attention native library not available at {_LIB_PATH} "
            f"(build it with `python -m magi_attention.csrc.build`). "
            f"The HIP kernel path is mandatory on GPU - no fallback. "
            f"Original error: {_load_error}"
        )
    return l


def is_available() -> bool:
    return _try_load() is not None


def ptr(t: torch.Tensor | None) -> ctypes.c_void_p:
    if t is None:
        return ctypes.c_void_p(0)
    return ctypes.c_void_p(t.data_ptr())


def current_stream_ptr() -> ctypes.c_void_p:
    return ctypes.c_void_p(torch.cuda.current_stream().cuda_stream)


def check(rc: int, what: str) -> None:
    if rc != 0:
        raise RuntimeError(f"{what} failed with code {rc}")
