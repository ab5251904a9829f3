"""Reference-named view of the general env flags (reference surface:
env/general.py). The flat magi_attention.env module holds the primary
implementations; this module adds the reference's exact names and the few
flags whose mechanisms have no MI355X analogue (documented per function)."""
from __future__ import annotations

from . import (  # noqa: F401
    _get,
    _get_bool,
    _get_int,
    dist_attn_runtime_dict_size,
    is_auto_range_merge_enable,
    is_cpp_backend_enable,
    is_deterministic_mode_enable,
    is_sanity_check_enable,
    kernel_backend,
    log_level,
    min_chunks_per_rank,
)


def is_flatten_head_groups_enable() -> bool:
    """Reference: flatten GQA head groups into the token dim for some SM90
    schedules. The MI355X kernels index heads directly; accepted, unused."""
    return _get_bool("MAGI_ATTENTION_FLATTEN_HEAD_GROUPS")


def precision() -> str | None:
    """MAGI_ATTENTION_PRECISION override of the compute dtype."""
    return _get("MAGI_ATTENTION_PRECISION", "") or None


def is_cuda_device_max_connections_one() -> bool:
    """Reference: checks CUDA_DEVICE_MAX_CONNECTIONS=1 (stream-ordering
    hint). ROCm's analogue is GPU_MAX_HW_QUEUES; either counts here."""
    import os

    return (
        os.environ.get("CUDA_DEVICE_MAX_CONNECTIONS") == "1"
        or os.environ.get("GPU_MAX_HW_QUEUES") == "1"
    )


def is_profile_mode_enable() -> bool:
    return _get_bool("MAGI_ATTENTION_PROFILE_MODE")


def is_cat_gqa_enable() -> bool:
    """Reference: concatenate GQA heads for wider SM90 tiles. No MI355X
    analogue (the wave64 tiling covers heads directly); accepted, unused."""
    return _get_bool("MAGI_ATTENTION_CATGQA")


def dist_attn_backward_hide_tail_reduce() -> bool:
    from . import is_bwd_hide_tail_reduce

    return is_bwd_hide_tail_reduce()
