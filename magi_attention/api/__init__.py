# Public API exports (reference magi_attention/api/__init__.py:15-60).
from magi_attention.common import AttnForwardMeta
from magi_attention.common.enum import AttnMaskType, AttnOverlapMode
from magi_attention.common.ranges import AttnRanges
from magi_attention.config import (
    BSDispatchAlg,
    DispatchAlg,
    DispatchConfig,
    DistAttnConfig,
    DPDispatchAlg,
    GreedyOverlapAlg,
    GrpCollConfig,
    LBDispatchAlg,
    MinHeapDispatchAlg,
    OverlapAlg,
    OverlapConfig,
    SequentialDispatchAlg,
    SortedSequentialSelectAlg,
    ToppHeapDispatchAlg,
    UniformOverlapAlg,
)
from magi_attention.dist_attn_runtime_mgr import DistAttnRuntimeKey
from magi_attention.functional import flex_flash_attn_func

from .functools import (
    compute_pad_size,
    infer_attn_mask_from_cu_seqlens,
    infer_attn_mask_from_sliding_window,
    infer_varlen_mask_from_batch,
    squash_batch_dim,
)
from .magi_attn_interface import (
    DistAttnRuntimeDictManager,
    GeneralAttnMaskType,
    calc_attn,
    clear_cache,
    dispatch,
    dist_attn_runtime_dict_mgr,
    get_most_recent_key,
    get_position_ids,
    magi_attn_flex_dispatch,
    magi_attn_flex_key,
    magi_attn_varlen_dispatch,
    magi_attn_varlen_key,
    make_flex_key_for_new_mask_after_dispatch,
    make_varlen_key_for_new_mask_after_dispatch,
    roll,
    roll_simple,
    undispatch,
)

__all__ = [
    "magi_attn_varlen_key",
    "magi_attn_varlen_dispatch",
    "magi_attn_flex_key",
    "magi_attn_flex_dispatch",
    "dispatch",
    "undispatch",
    "roll",
    "roll_simple",
    "calc_attn",
    "clear_cache",
    "get_most_recent_key",
    "get_position_ids",
    "make_varlen_key_for_new_mask_after_dispatch",
    "make_flex_key_for_new_mask_after_dispatch",
    "compute_pad_size",
    "squash_batch_dim",
    "infer_varlen_mask_from_batch",
    "infer_attn_mask_from_sliding_window",
    "infer_attn_mask_from_cu_seqlens",
    "DistAttnRuntimeKey",
    "DistAttnRuntimeDictManager",
    "dist_attn_runtime_dict_mgr",
    "DistAttnConfig",
    "DispatchConfig",
    "OverlapConfig",
    "GrpCollConfig",
    "AttnMaskType",
    "AttnOverlapMode",
    "AttnRanges",
    "AttnForwardMeta",
    "GeneralAttnMaskType",
]
