# DistAttnSolver — per-rank host/remote slice split, overlap staging, and the
# group-cast/group-reduce transfer tables.
# (Reference: meta/solver/dist_attn_solver.py:206 DistAttnSolver.solve:297,
#  make_comm_meta:1667, make_calc_meta:1836; zero-redundancy invariant
#  dist_attn_solver.py:463-470: remote K need = slice k-ranges minus hosted.)
from __future__ import annotations

from dataclasses import dataclass
from typing import List, Tuple

from ..common.range import AttnRange
from ..common.ranges import AttnRanges
from .containers import (
    AttnArg,
    CalcMeta,
    CommMeta,
    GroupCastArg,
    GroupReduceArg,
    RowChunkMap,
)
from .geometry import MaskSlice, k_window, q_window


def _ranges_of(pieces: List[AttnRange]) -> AttnRanges:
    rr = AttnRanges()
    for p in pieces:
        rr.append(p.clone())
    return rr


def _split_stages(need: AttnRanges, degree: int, min_stage_tokens: int = 512
                  ) -> List[AttnRanges]:
    """Split the (merged, sorted) remote-need ranges into `degree` contiguous
    groups with balanced token counts (reference OverlapSolver Uniform alg)."""
    total = need.total_seqlen
    if total == 0 or degree <= 1:
        return [need] + [AttnRanges() for _ in range(degree - 1)]
    # do not create stages smaller than min_stage_tokens
    degree = max(1, min(degree, (total + min_stage_tokens - 1) // min_stage_tokens))
    per = (total + degree - 1) // degree
    out: List[AttnRanges] = []
    cur = AttnRanges()
    cur_n = 0
    for r in need.merge():
        s = r.start
        while s < r.end:
            take = min(per - cur_n, r.end - s)
            cur.append(AttnRange(s, s + take))
            cur_n += take
            s += take
            if cur_n == per:
                out.append(cur)
                cur = AttnRanges()
                cur_n = 0
    if cur_n:
        out.append(cur)
    while len(out) < degree:
        out.append(AttnRanges())
    return out


@dataclass
class RankAttnPlan:
    host_ranges: AttnRanges              # global rows hosted (merged)
    host_slices: List[MaskSlice]         # k in global coords, within host
    remote_slices: List[MaskSlice]       # k in global coords, remote
    stages_need: List[AttnRanges]        # per stage: global k ranges needed


class DistAttnSolver:
    """Builds per-rank CalcMeta/CommMeta from the global mask and dispatch.
    Deterministic pure-host computation: every rank derives the full table."""

    def __init__(
        self,
        slices: List[MaskSlice],
        partitions: List[List[int]],
        chunk_size: int,
        total_seqlen: int,
        cp_size: int,
        overlap_degree: int = 2,
        min_stage_tokens: int = 512,
    ):
        self.cp_size = cp_size
        self.chunk_size = chunk_size
        self.total_seqlen = total_seqlen
        self.plans: List[RankAttnPlan] = []

        host_ranges_all: List[AttnRanges] = []
        for r in range(cp_size):
            rr = AttnRanges()
            for c in partitions[r]:
                rr.append(AttnRange(c * chunk_size, (c + 1) * chunk_size))
            host_ranges_all.append(rr.merge())
        self.host_ranges_all = host_ranges_all

        max_deg = 1
        for r in range(cp_size):
            hr = host_ranges_all[r]
            rank_slices: List[MaskSlice] = []
            for piece in hr:
                for sl in slices:
                    rank_slices.extend(q_window(sl, piece.start, piece.end))
            host_slices: List[MaskSlice] = []
            remote_slices: List[MaskSlice] = []
            remote_need = AttnRanges()
            for sl in rank_slices:
                k_rng = AttnRanges.from_ranges([(sl.ks, sl.ke)])
                host_parts = k_rng.find_overlap_ranges(hr)
                hole_parts = k_rng.find_hole_ranges(hr)
                for p in host_parts:
                    host_slices.extend(k_window(sl, p.start, p.end))
                for p in hole_parts:
                    remote_slices.extend(k_window(sl, p.start, p.end))
                    remote_need.append(p.clone())
            remote_need = remote_need.merge()
            stages = _split_stages(remote_need, overlap_degree, min_stage_tokens)
            max_deg = max(max_deg, len(stages))
            self.plans.append(
                RankAttnPlan(hr, host_slices, remote_slices, stages)
            )
        # all ranks must run the same number of collective rounds
        self.overlap_degree = max_deg
        for p in self.plans:
            while len(p.stages_need) < max_deg:
                p.stages_need.append(AttnRanges())

    # ---------------- calc meta ----------------
    def make_calc_meta(self, rank: int) -> CalcMeta:
        plan = self.plans[rank]
        hr = plan.host_ranges

        def to_local_args(slices: List[MaskSlice], k_space: AttnRanges) -> AttnArg:
            qr, kr, tm = [], [], []
            area = 0
            mx = 0
            for sl in slices:
                lq = hr.make_range_local(AttnRange(sl.qs, sl.qe), is_self_merged=True)
                lk = k_space.make_range_local(AttnRange(sl.ks, sl.ke))
                qr.append((lq.start, lq.end))
                kr.append((lk.start, lk.end))
                tm.append(sl.t)
                area += sl.area()
                mx = max(mx, lq.end - lq.start)
            return AttnArg(qr, kr, tm, max_seqlen_q=mx, total_area=area)

        host_arg = to_local_args(plan.host_slices, hr)
        stage_args = []
        for s in range(self.overlap_degree):
            st = plan.stages_need[s]
            stage_slices: List[MaskSlice] = []
            for sl in plan.remote_slices:
                for piece in st:
                    stage_slices.extend(k_window(sl, piece.start, piece.end))
            stage_args.append(to_local_args(stage_slices, st))
        return CalcMeta(host_arg=host_arg, stage_args=stage_args)

    # ---------------- comm meta ----------------
    def make_comm_meta(self, rank: int) -> CommMeta:
        cp = self.cp_size
        casts: List[GroupCastArg] = []
        reduces: List[GroupReduceArg] = []
        my_hr = self.host_ranges_all[rank]
        L = my_hr.total_seqlen  # my local kv rows

        for s in range(self.overlap_degree):
            # overlap table: ov[o][r] = what owner o sends to dst r this stage
            ov = [
                [
                    self.plans[r].stages_need[s].find_overlap_ranges(
                        self.host_ranges_all[o]
                    )
                    for r in range(cp)
                ]
                for o in range(cp)
            ]
            # ---- cast: me as SENDER (owner) ----
            send_in: List[Tuple[int, int]] = []
            send_out: List[int] = []
            in_splits: List[int] = []
            cursor = 0
            for r in range(cp):
                pieces = ov[rank][r]
                tok = pieces.total_seqlen
                # k rows then v rows within this dst segment
                vcur = cursor + tok
                for p in pieces:
                    lp = my_hr.make_range_local(p, is_self_merged=True)
                    send_in.append((lp.start, lp.end))
                    send_out.append(cursor)
                    cursor += p.seqlen
                for p in pieces:
                    lp = my_hr.make_range_local(p, is_self_merged=True)
                    send_in.append((L + lp.start, L + lp.end))
                    send_out.append(vcur)
                    vcur += p.seqlen
                cursor = vcur
                in_splits.append(2 * tok)
            total_send = cursor
            # ---- cast: me as RECEIVER ----
            my_need = self.plans[rank].stages_need[s]
            S = my_need.total_seqlen
            recv_in: List[Tuple[int, int]] = []
            recv_out: List[int] = []
            out_splits: List[int] = []
            rcur = 0
            for o in range(cp):
                pieces = ov[o][rank]
                tok = pieces.total_seqlen
                for p in pieces:  # k rows
                    lp = my_need.make_range_local(p)
                    recv_in.append((rcur, rcur + p.seqlen))
                    recv_out.append(lp.start)
                    rcur += p.seqlen
                for p in pieces:  # v rows -> stage offset +S
                    lp = my_need.make_range_local(p)
                    recv_in.append((rcur, rcur + p.seqlen))
                    recv_out.append(S + lp.start)
                    rcur += p.seqlen
                out_splits.append(2 * tok)
            casts.append(
                GroupCastArg(
                    send_pack=RowChunkMap(send_in, send_out, total_send),
                    input_split_sizes=in_splits,
                    recv_unpack=RowChunkMap(recv_in, recv_out, rcur),
                    output_split_sizes=out_splits,
                    stage_tokens=S,
                )
            )
            # ---- reduce (bwd): reverse tables ----
            # me as SENDER of partial dkv (stage buffer rows -> owners)
            rs_in: List[Tuple[int, int]] = []
            rs_out: List[int] = []
            rs_splits: List[int] = []
            cursor = 0
            for o in range(cp):
                pieces = ov[o][rank]  # what I received from o = what I return
                tok = pieces.total_seqlen
                vcur = cursor + tok
                for p in pieces:
                    lp = my_need.make_range_local(p)
                    rs_in.append((lp.start, lp.end))
                    rs_out.append(cursor)
                    cursor += p.seqlen
                for p in pieces:
                    lp = my_need.make_range_local(p)
                    rs_in.append((S + lp.start, S + lp.end))
                    rs_out.append(vcur)
                    vcur += p.seqlen
                cursor = vcur
                rs_splits.append(2 * tok)
            # me as RECEIVER of partial dkv for my hosted rows (sum-reduce)
            rr_in: List[Tuple[int, int]] = []
            rr_out: List[int] = []
            rr_splits: List[int] = []
            rcur = 0
            for r in range(cp):
                pieces = ov[rank][r]
                tok = pieces.total_seqlen
                for p in pieces:  # dk rows
                    lp = my_hr.make_range_local(p, is_self_merged=True)
                    rr_in.append((rcur, rcur + p.seqlen))
                    rr_out.append(lp.start)
                    rcur += p.seqlen
                for p in pieces:  # dv rows
                    lp = my_hr.make_range_local(p, is_self_merged=True)
                    rr_in.append((rcur, rcur + p.seqlen))
                    rr_out.append(L + lp.start)
                    rcur += p.seqlen
                rr_splits.append(2 * tok)
            reduces.append(
                GroupReduceArg(
                    send_pack=RowChunkMap(rs_in, rs_out, cursor),
                    input_split_sizes=rs_splits,
                    recv_reduce=RowChunkMap(rr_in, rr_out, rcur),
                    output_split_sizes=rr_splits,
                    total_recv=rcur,
                )
            )
        return CommMeta(stages_cast=casts, stages_reduce=reduces)
