# Public config dataclasses (reference magi_attention/config.py:1-71 +
# meta/solver/dispatch_solver.py:359 DispatchConfig + overlap_solver.py
# OverlapConfig — names kept).
from __future__ import annotations

from dataclasses import dataclass, field

from . import env  # noqa: F401  (reference config.py:17 re-export)
from .common.enum import AttnOverlapMode, DispatchAlgType, OverlapAlgType


# ---- dispatch algorithms ----
@dataclass(frozen=True)
class DispatchAlg:
    type: DispatchAlgType = DispatchAlgType.MIN_HEAP


@dataclass(frozen=True)
class MinHeapDispatchAlg(DispatchAlg):
    type: DispatchAlgType = DispatchAlgType.MIN_HEAP


@dataclass(frozen=True)
class LBDispatchAlg(DispatchAlg):
    type: DispatchAlgType = DispatchAlgType.LOWER_BOUND


@dataclass(frozen=True)
class DPDispatchAlg(DispatchAlg):
    type: DispatchAlgType = DispatchAlgType.DYNAMIC_PROGRAMMING


@dataclass(frozen=True)
class BSDispatchAlg(DispatchAlg):
    type: DispatchAlgType = DispatchAlgType.BINARY_SEARCH


@dataclass(frozen=True)
class SequentialDispatchAlg(DispatchAlg):
    type: DispatchAlgType = DispatchAlgType.SEQUENTIAL_SELECT


@dataclass(frozen=True)
class SortedSequentialSelectAlg(DispatchAlg):
    type: DispatchAlgType = DispatchAlgType.SORTED_SEQUENTIAL_SELECT


@dataclass(frozen=True)
class ToppHeapDispatchAlg(DispatchAlg):
    """MinHeap with IOU-affinity tie-breaks: the top max(1, ceil(cp*top_p))
    least-loaded ranks are candidates, affinity picks among them (reference
    dispatch_solver.py:207,990)."""

    type: DispatchAlgType = DispatchAlgType.TOPP_HEAP
    top_p: float = 0.5


# ---- overlap algorithms ----
@dataclass(frozen=True)
class OverlapAlg:
    type: OverlapAlgType = OverlapAlgType.UNIFORM


@dataclass(frozen=True)
class UniformOverlapAlg(OverlapAlg):
    type: OverlapAlgType = OverlapAlgType.UNIFORM


@dataclass(frozen=True)
class GreedyOverlapAlg(OverlapAlg):
    type: OverlapAlgType = OverlapAlgType.GREEDY


@dataclass(frozen=True)
class DispatchConfig:
    chunk_size: int = 512
    alg: DispatchAlg = field(default_factory=MinHeapDispatchAlg)
    uneven_shard: bool = False


@dataclass(frozen=True)
class OverlapConfig:
    enable: bool = True
    mode: AttnOverlapMode = AttnOverlapMode.STATIC
    degree: int | None = 2
    min_chunk_size: int = 512
    max_num_chunks: int = 64
    alg: OverlapAlg = field(default_factory=UniformOverlapAlg)


@dataclass(frozen=True)
class GrpCollConfig:
    """Kept for API compatibility (native grpcoll is a later-round item)."""

    num_sms: int = 8


@dataclass(frozen=True)
class DistAttnConfig:
    dispatch_config: DispatchConfig = field(default_factory=DispatchConfig)
    overlap_config: OverlapConfig = field(default_factory=OverlapConfig)
    grpcoll_config: GrpCollConfig = field(default_factory=GrpCollConfig)
