# Env-flag system (reference magi_attention/env/{general,comm}.py — names kept
# verbatim; flags are snapshotted into DistAttnRuntimeKey so flips produce new
# plans, reference dist_attn_runtime_mgr.py:79-87).
from __future__ import annotations

import os
from typing import Any


def _get(name: str, default: str) -> str:
    return os.environ.get(name, default)


def _get_bool(name: str, default: bool = False) -> bool:
    return _get(name, "1" if default else "0") in ("1", "true", "TRUE", "True")


def _get_int(name: str, default: int) -> int:
    return int(_get(name, str(default)))


# ---- general (reference env/general.py) ----
def is_sanity_check_enable() -> bool:
    return _get_bool("MAGI_ATTENTION_SANITY_CHECK")


def is_deterministic_mode_enable() -> bool:
    return _get_bool("MAGI_ATTENTION_DETERMINISTIC_MODE")


def is_cpp_backend_enable() -> bool:
    return _get_bool("MAGI_ATTENTION_CPP_BACKEND")


def is_auto_range_merge_enable() -> bool:
    return _get_bool("MAGI_ATTENTION_AUTO_RANGE_MERGE")


def kernel_backend() -> str:
    return _get("MAGI_ATTENTION_KERNEL_BACKEND", "ffa")


def min_chunks_per_rank() -> int:
    return _get_int("MAGI_ATTENTION_MIN_CHUNKS_PER_RANK", 2)


def dist_attn_runtime_dict_size() -> int:
    return _get_int("MAGI_ATTENTION_DIST_ATTN_RUNTIME_DICT_SIZE", 100)


def log_level() -> str:
    return _get("MAGI_ATTENTION_LOG_LEVEL", "WARNING")


# ---- comm (reference env/comm.py) ----
def is_hierarchical_comm_enable() -> bool:
    return _get_bool("MAGI_ATTENTION_HIERARCHICAL_COMM")


def is_bwd_cosched() -> bool:
    """Default OFF since the late-r2 head-major dkv gate: the fused dkv's
    per-head XCD-L2 affinity (99.1 vs 109.4 ms solo) is diluted when dq
    waves co-reside, so serializing the passes now wins the step A/B
    (205.7 vs 208.9 ms at 64k; earlier in r2, pre-gate, co-scheduling had
    won 209.6 vs 218.3). MAGI_BWD_COSCHED=1 re-enables the side-stream
    co-schedule."""
    return _get("MAGI_BWD_COSCHED", "0") == "1"


def bwd_dkv_mode() -> str:
    """"auto" (default): fused dK+dV kernel for long ranges (r2: 124.5 ms vs
    the dv+dk split's 127.4 at 64k, one less launch + one less Q/dO staging
    stream), split for short ranges (the fused W4 variant is occupancy-1).
    MAGI_BWD_SPLIT_DKV=1 forces the split; MAGI_BWD_FUSED_DKV=1 forces fused.
    """
    if _get_bool("MAGI_BWD_SPLIT_DKV"):
        return "split"
    if _get_bool("MAGI_BWD_FUSED_DKV"):
        return "fused"
    return "auto"


def is_bwd_split_dkv(max_seqlen_k: int = 0, head_dim: int = 128) -> bool:
    mode = bwd_dkv_mode()
    if head_dim == 192:
        return True  # fused + D=192 exceeds the 160 KB LDS (its V tiles)
    if mode == "auto":
        # r2: the fused W8 kernel wins from 1k ranges up (A/B on 2k varlen:
        # fused 0.92 ms vs dv+dk 1.04); below that the W4 split is safer
        return max_seqlen_k < 1024
    return mode == "split"


def ffa_forward_sm_margin() -> int:
    return _get_int("MAGI_ATTENTION_FFA_FORWARD_SM_MARGIN", 8)


def ffa_backward_sm_margin() -> int:
    return _get_int("MAGI_ATTENTION_FFA_BACKWARD_SM_MARGIN", 8)


def is_qo_comm_enable() -> bool:
    return _get_bool("MAGI_ATTENTION_QO_COMM")


def is_forward_high_precision_reduce_enable() -> bool:
    return _get_bool("MAGI_ATTENTION_FORWARD_HIGH_PRECISION_REDUCE")


def is_backward_high_precision_reduce_enable() -> bool:
    return _get_bool("MAGI_ATTENTION_BACKWARD_HIGH_PRECISION_REDUCE")


def is_bwd_hide_tail_reduce() -> bool:
    """MAGI_ATTENTION_BWD_HIDE_TAIL_REDUCE=1: overlap the tail-stage dKV
    group-reduce with the local-grad dtype casts instead of waiting first
    (reference env/general.py:233, dist_attn.py:2503)."""
    return _get_bool("MAGI_ATTENTION_BWD_HIDE_TAIL_REDUCE")


def is_native_grpcoll_enable() -> bool:
    return _get_bool("MAGI_ATTENTION_NATIVE_GRPCOLL")


def snapshot() -> tuple[tuple[str, Any], ...]:
    """Frozen snapshot of every flag that affects plan construction."""
    return (
        ("deterministic", is_deterministic_mode_enable()),
        ("kernel_backend", kernel_backend()),
        ("auto_range_merge", is_auto_range_merge_enable()),
        ("hierarchical_comm", is_hierarchical_comm_enable()),
        ("qo_comm", is_qo_comm_enable()),
        ("bwd_hide_tail_reduce", is_bwd_hide_tail_reduce()),
        ("fwd_hp_reduce", is_forward_high_precision_reduce_enable()),
        ("bwd_hp_reduce", is_backward_high_precision_reduce_enable()),
        ("fwd_sm_margin", ffa_forward_sm_margin()),
        ("bwd_sm_margin", ffa_backward_sm_margin()),
        ("native_grpcoll", is_native_grpcoll_enable()),
        ("min_chunks_per_rank", min_chunks_per_rank()),
    )


# reference-named submodule views (env/general.py, env/comm.py, env/build.py)
# — imported at the bottom: they re-import this module's flag functions
from . import build, comm, general  # noqa: E402,F401
