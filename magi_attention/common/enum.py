# MI355X-native rebuild of the reference API surface
# (reference: magi_attention/common/enum.py — names and int maps kept).
from __future__ import annotations

from enum import Enum
from typing import Literal, Tuple, TypeAlias

import torch

GroupReduceOp: TypeAlias = Literal["sum", "avg", "lse"]
OutMaybeWithLSE: TypeAlias = "torch.Tensor | Tuple[torch.Tensor, torch.Tensor]"
AttnSinkLayout: TypeAlias = Literal["sh", "shd", "ssh"]


class AttnType(Enum):
    SELF_ATTN = "self_attn"
    CROSS_ATTN = "cross_attn"


class AttnRole(Enum):
    QUERY = "query"
    KEY = "key"
    VALUE = "value"


class AttnMaskType(Enum):
    """int map: 0=FULL, 1=CAUSAL, 2=INVCAUSAL, 3=BICAUSAL
    (reference common/enum.py:42-98; kernel semantics flex_flash_attn.py:1247-1341)."""

    FULL = "full"
    CAUSAL = "causal"
    BICAUSAL = "bi_causal"
    INVCAUSAL = "inv_causal"

    @classmethod
    def from_int_type(cls, int_type: int) -> "AttnMaskType":
        return (cls.FULL, cls.CAUSAL, cls.INVCAUSAL, cls.BICAUSAL)[int_type]

    def to_int_type(self) -> int:
        return {
            AttnMaskType.FULL: 0,
            AttnMaskType.CAUSAL: 1,
            AttnMaskType.INVCAUSAL: 2,
            AttnMaskType.BICAUSAL: 3,
        }[self]


class AttnOverlapMode(Enum):
    STATIC = "static"
    DYNAMIC = "dynamic"


class DispatchAlgType(Enum):
    LOWER_BOUND = "lower_bound"
    DYNAMIC_PROGRAMMING = "dynamic_programming"
    BINARY_SEARCH = "binary_search"
    MIN_HEAP = "min_heap"
    TOPP_HEAP = "topp_heap"
    BACKTRACKING_PRUNING = "backtracing_pruning"
    RANDOM_SELECT = "random_select"
    SEQUENTIAL_SELECT = "sequential_select"
    BATCH_TOPP_HEAP = "batch_topp_heap"
    SORTED_SEQUENTIAL_SELECT = "sorted_sequential_select"


class OverlapAlgType(Enum):
    UNIFORM = "uniform"
    GREEDY = "greedy"


class MagiAttentionKernelBackend(Enum):
    FFA = "ffa"
    SDPA = "sdpa"
    SDPA_OL = "sdpa_ol"
    FA4 = "fa4"


class MagiAttentionPrecision(Enum):
    BF16 = "bf16"
    FP16 = "fp16"
    FP32 = "fp32"
    FP64 = "fp64"


class DynamicAttnAlgType(Enum):
    """Reference common/enum.py — dynamic attn solver algorithm tags (the
    dynamic solver itself is a later-round item, SURVEY #17)."""

    BALANCED = "balanced"
    GREEDY = "greedy"


class GrpCollBufferName(Enum):
    """Reference common/enum.py — named native-grpcoll buffers (the RCCL
    a2av transport used here needs none; kept for import parity)."""

    GROUP_CAST = "group_cast"
    GROUP_REDUCE = "group_reduce"
