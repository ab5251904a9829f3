# Planner layer (reference magi_attention/meta/_make_dispatch_meta.py:56 and
# _make_attn_meta.py:40 — same roles, MI355X-first rebuild).
from __future__ import annotations

from dataclasses import dataclass, field
from typing import List

from ..common.enum import AttnMaskType
from ..common.ranges import AttnRanges
from ..config import DistAttnConfig
from .attn_solver import DistAttnSolver
from .containers import CalcMeta, CommMeta
from .geometry import MaskSlice, slice_from_raw
from .solver.dispatch_solver import DispatchSolver


@dataclass
class DispatchMeta:
    """Chunk -> rank assignment (reference meta/collection/dispatch_meta.py:24)."""

    cp_size: int
    cp_rank: int
    chunk_size: int
    total_seqlen: int          # padded
    num_chunks: int
    partitions: List[List[int]]
    loads: List[float] = field(default_factory=list)

    @property
    def chunks_per_rank(self) -> int:
        return self.num_chunks // self.cp_size

    def host_ranges(self, rank: int) -> AttnRanges:
        rr = AttnRanges()
        for c in self.partitions[rank]:
            rr.append(
                AttnRanges.from_ranges(
                    [(c * self.chunk_size, (c + 1) * self.chunk_size)]
                )[0]
            )
        return rr.merge()


def normalize_slices(
    q_ranges: AttnRanges,
    k_ranges: AttnRanges,
    attn_mask_type,
) -> List[MaskSlice]:
    """Raw API triples -> aligned MaskSlices."""
    if not isinstance(attn_mask_type, (list, tuple)):
        attn_mask_type = [attn_mask_type] * len(q_ranges)
    out: List[MaskSlice] = []
    for qr, kr, t in zip(q_ranges, k_ranges, attn_mask_type):
        ti = t.to_int_type() if isinstance(t, AttnMaskType) else (
            AttnMaskType(t).to_int_type() if isinstance(t, str) else int(t)
        )
        out.extend(slice_from_raw(qr.start, qr.end, kr.start, kr.end, ti))
    return out


def make_dispatch_meta_from_qk_ranges(
    slices: List[MaskSlice],
    total_seqlen_padded: int,
    cp_size: int,
    cp_rank: int,
    dist_attn_config: DistAttnConfig,
) -> DispatchMeta:
    """Chunk the (padded) token space and assign chunks to ranks balancing
    per-chunk mask area (reference _make_dispatch_meta.py:56)."""
    from .geometry import area_in_rows

    chunk_size = dist_attn_config.dispatch_config.chunk_size
    assert total_seqlen_padded % (chunk_size * cp_size) == 0, (
        f"padded seqlen {total_seqlen_padded} not divisible by "
        f"chunk_size*cp = {chunk_size}*{cp_size}"
    )
    num_chunks = total_seqlen_padded // chunk_size
    workloads = []
    for c in range(num_chunks):
        a, b = c * chunk_size, (c + 1) * chunk_size
        workloads.append(float(sum(area_in_rows(sl, a, b) for sl in slices)))
    affinities = None
    from ..common.enum import DispatchAlgType

    if dist_attn_config.dispatch_config.alg.type == DispatchAlgType.TOPP_HEAP:
        # per-chunk k coverage for the IOU-affinity tie-break
        from .geometry import q_window

        affinities = []
        for c in range(num_chunks):
            a, b = c * chunk_size, (c + 1) * chunk_size
            rr = AttnRanges()
            for sl in slices:
                for sub in q_window(sl, a, b):
                    rr.append(AttnRanges.from_ranges([[sub.ks, sub.ke]])[0])
            affinities.append(rr.merge())
    sol = DispatchSolver(dist_attn_config.dispatch_config.alg).solve(
        workloads, cp_size, affinities=affinities,
        uneven_shard=dist_attn_config.dispatch_config.uneven_shard,
    )
    return DispatchMeta(
        cp_size=cp_size,
        cp_rank=cp_rank,
        chunk_size=chunk_size,
        total_seqlen=total_seqlen_padded,
        num_chunks=num_chunks,
        partitions=sol.partitions,
        loads=sol.loads,
    )


def make_attn_meta_from_dispatch_meta(
    slices: List[MaskSlice],
    dispatch_meta: DispatchMeta,
    dist_attn_config: DistAttnConfig,
) -> tuple[DistAttnSolver, CalcMeta, CommMeta]:
    """Reference _make_attn_meta.py:40: run the DistAttnSolver and emit this
    rank's calc/comm tables."""
    oc = dist_attn_config.overlap_config
    degree = (oc.degree or 1) if oc.enable else 1
    from ..common.enum import AttnOverlapMode, OverlapAlgType

    solver = DistAttnSolver(
        slices=slices,
        partitions=dispatch_meta.partitions,
        chunk_size=dispatch_meta.chunk_size,
        total_seqlen=dispatch_meta.total_seqlen,
        cp_size=dispatch_meta.cp_size,
        overlap_degree=degree,
        min_stage_tokens=oc.min_chunk_size,
        overlap_mode=("dynamic" if oc.enable
                      and oc.mode == AttnOverlapMode.DYNAMIC else "static"),
        overlap_alg=("greedy" if oc.alg.type == OverlapAlgType.GREEDY
                     else "uniform"),
        max_num_chunks=oc.max_num_chunks,
    )
    rank = dispatch_meta.cp_rank
    return solver, solver.make_calc_meta(rank), solver.make_comm_meta(rank)


def make_qo_meta_from_dispatch_meta(
    slices: List[MaskSlice],
    dispatch_meta: DispatchMeta,
    dist_attn_config: DistAttnConfig,
):
    """QO-comm plan (MAGI_ATTENTION_QO_COMM=1): run the SAME solver on the
    transposed mask — its remote-K machinery then produces remote-Q cast and
    partial-(out,lse)/dq reduce tables (reference env/comm.py:72)."""
    from .geometry import transpose_slice

    oc = dist_attn_config.overlap_config
    degree = (oc.degree or 1) if oc.enable else 1
    slices_t = [t for sl in slices for t in transpose_slice(sl)]
    solver_t = DistAttnSolver(
        slices=slices_t,
        partitions=dispatch_meta.partitions,
        chunk_size=dispatch_meta.chunk_size,
        total_seqlen=dispatch_meta.total_seqlen,
        cp_size=dispatch_meta.cp_size,
        overlap_degree=degree,
        min_stage_tokens=oc.min_chunk_size,
    )
    return solver_t.make_qo_comm_meta(dispatch_meta.cp_rank)


# reference-surface submodules + bucket factories (imported at the bottom:
# collection re-imports DispatchMeta from this partially-initialized module)
from . import collection, container  # noqa: E402,F401
from ._buckets import (  # noqa: E402,F401
    make_bucket_per_rank_from_qk_ranges,
    make_global_bucket_from_qk_ranges,
)
