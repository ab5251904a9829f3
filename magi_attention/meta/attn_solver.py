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
    HierGroupCastArg,
    HierGroupReduceArg,
    RowChunkMap,
)
from .containers import QoCommMeta
from .geometry import MaskSlice, k_window, q_window, transpose_slice


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


def _split_stages_area(
    need: AttnRanges, remote_slices: List[MaskSlice], degree: int,
    min_stage_tokens: int,
) -> List[AttnRanges]:
    """Greedy area-balanced stage composition (reference GreedyOverlapAlg):
    cut the remote-need ranges into ~min_stage_tokens pieces, compute the
    mask AREA each unlocks, and min-heap the pieces into `degree` stages by
    accumulated area — stages then carry comparable COMPUTE, not comparable
    tokens (a token-balanced split under a causal mask gives the last stage
    far more area than the first)."""
    import heapq

    pieces: List[Tuple[int, int]] = []
    for r in need.merge():
        a = r.start
        while a < r.end:
            b = min(a + min_stage_tokens, r.end)
            pieces.append((a, b))
            a = b
    if not pieces:
        return [AttnRanges() for _ in range(degree)]

    def area_of(a: int, b: int) -> int:
        tot = 0
        for sl in remote_slices:
            for sub in k_window(sl, a, b):
                tot += sub.area()
        return tot

    degree = max(1, min(degree, len(pieces)))
    scored = sorted(((area_of(a, b), a, b) for a, b in pieces), reverse=True)
    heap = [(0.0, i) for i in range(degree)]
    heapq.heapify(heap)
    buckets: List[List[Tuple[int, int]]] = [[] for _ in range(degree)]
    for ar, a, b in scored:
        load, i = heapq.heappop(heap)
        buckets[i].append((a, b))
        heapq.heappush(heap, (load + ar, i))
    out = []
    for bk in buckets:
        rr = AttnRanges()
        for a, b in sorted(bk):
            rr.append(AttnRange(a, b))
        out.append(rr.merge())
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
        overlap_mode: str = "static",
        overlap_alg: str = "uniform",
        max_num_chunks: int = 64,
    ):
        """overlap_mode "dynamic" (reference AttnOverlapMode.DYNAMIC,
        overlap_solver.py): the per-rank stage count follows the remote-need
        size (one stage per ~4*min_stage_tokens, capped by max_num_chunks)
        instead of a fixed degree; all ranks still pad to the global max so
        every rank runs the same number of collective rounds (the reference
        all-reduces the degree, dist_attn_solver.py:994 — here every rank
        derives all plans deterministically, so it is a host max).
        overlap_alg "greedy" (reference GreedyOverlapAlg, overlap_solver.py:
        205): stages balance the CALC AREA they unlock, not the token count —
        min-heap assignment of min_stage_tokens-sized pieces by area."""
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
            deg = overlap_degree
            if overlap_mode == "dynamic":
                tok = remote_need.total_seqlen
                deg = max(1, min((tok + 4 * min_stage_tokens - 1)
                                 // (4 * min_stage_tokens),
                                 max_num_chunks))
            if overlap_alg == "greedy" and deg > 1:
                stages = _split_stages_area(
                    remote_need, remote_slices, deg, min_stage_tokens
                )
            else:
                stages = _split_stages(remote_need, deg, min_stage_tokens)
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

    def _overlap_table(self, s: int) -> List[List[AttnRanges]]:
        """ov[o][r] = the (merged) global k ranges owner o casts to rank r in
        stage s. Every rank derives the full table deterministically."""
        cp = self.cp_size
        return [
            [
                self.plans[r].stages_need[s].find_overlap_ranges(
                    self.host_ranges_all[o]
                )
                for r in range(cp)
            ]
            for o in range(cp)
        ]

    # ---------------- comm meta ----------------
    def make_comm_meta(self, rank: int, ncopies: int = 2) -> CommMeta:
        """ncopies: tensor copies packed through the same row tables — 2 for
        (K,V) (also reused for QO-comm's (q,do) and (lse,dpsum) casts), 1
        for single-tensor casts/reduces (QO-comm q / out / dq)."""
        cp = self.cp_size
        casts: List[GroupCastArg] = []
        reduces: List[GroupReduceArg] = []
        my_hr = self.host_ranges_all[rank]
        L = my_hr.total_seqlen  # my local kv rows

        for s in range(self.overlap_degree):
            ov = self._overlap_table(s)
            # ---- cast: me as SENDER (owner) ----
            send_in: List[Tuple[int, int]] = []
            send_out: List[int] = []
            in_splits: List[int] = []
            cursor = 0
            for r in range(cp):
                pieces = ov[rank][r]
                tok = pieces.total_seqlen
                # copy c's rows at source offset c*L within this dst segment
                for c in range(ncopies):
                    ccur = cursor + c * tok
                    for p in pieces:
                        lp = my_hr.make_range_local(p, is_self_merged=True)
                        send_in.append((c * L + lp.start, c * L + lp.end))
                        send_out.append(ccur)
                        ccur += p.seqlen
                cursor += ncopies * tok
                in_splits.append(ncopies * tok)
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
                for c in range(ncopies):
                    for p in pieces:
                        lp = my_need.make_range_local(p)
                        recv_in.append((rcur, rcur + p.seqlen))
                        recv_out.append(c * S + lp.start)
                        rcur += p.seqlen
                out_splits.append(ncopies * tok)
            casts.append(
                GroupCastArg(
                    send_pack=RowChunkMap(send_in, send_out, total_send),
                    input_split_sizes=in_splits,
                    recv_unpack=RowChunkMap(recv_in, recv_out, rcur),
                    output_split_sizes=out_splits,
                    stage_tokens=S,
                    ncopies=ncopies,
                )
            )
            # ---- reduce (bwd): reverse tables ----
            # me as SENDER of partials (stage buffer rows -> owners)
            rs_in: List[Tuple[int, int]] = []
            rs_out: List[int] = []
            rs_splits: List[int] = []
            cursor = 0
            for o in range(cp):
                pieces = ov[o][rank]  # what I received from o = what I return
                tok = pieces.total_seqlen
                for c in range(ncopies):
                    ccur = cursor + c * tok
                    for p in pieces:
                        lp = my_need.make_range_local(p)
                        rs_in.append((c * S + lp.start, c * S + lp.end))
                        rs_out.append(ccur)
                        ccur += p.seqlen
                cursor += ncopies * tok
                rs_splits.append(ncopies * tok)
            # me as RECEIVER of partials for my hosted rows (sum-reduce)
            rr_in: List[Tuple[int, int]] = []
            rr_out: List[int] = []
            rr_splits: List[int] = []
            rcur = 0
            for r in range(cp):
                pieces = ov[rank][r]
                tok = pieces.total_seqlen
                for c in range(ncopies):
                    for p in pieces:
                        lp = my_hr.make_range_local(p, is_self_merged=True)
                        rr_in.append((rcur, rcur + p.seqlen))
                        rr_out.append(c * L + lp.start)
                        rcur += p.seqlen
                rr_splits.append(ncopies * tok)
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

    # ---------------- native grpcoll pull tables ----------------
    def make_native_comm_meta(self, rank: int):
        """Per-stage PULL plans for the HIP-IPC transport (csrc/grpcoll.hip):
        cast: which rows of each peer's kv WINDOW land where in my stage
        buffer; reduce: which rows of each peer's partial-dKV window sum into
        my local dkv accumulator. All derived from the same overlap table as
        the a2av path (every rank computes all plans deterministically)."""
        from .containers import NativeStageMeta

        cp = self.cp_size
        my_hr = self.host_ranges_all[rank]
        L = my_hr.total_seqlen
        out = []
        stage_tokens_all = []
        for s in range(self.overlap_degree):
            ov = self._overlap_table(s)
            my_need = self.plans[rank].stages_need[s]
            S = my_need.total_seqlen
            cast = []
            for o in range(cp):
                L_o = self.host_ranges_all[o].total_seqlen
                for p_ in ov[o][rank]:
                    src = self.host_ranges_all[o].make_range_local(
                        p_, is_self_merged=True)
                    dst = my_need.make_range_local(p_)
                    n = p_.seqlen
                    cast.append((o, src.start, dst.start, n))          # k
                    cast.append((o, L_o + src.start, S + dst.start, n))  # v
            red = []
            for r in range(cp):
                S_r = self.plans[r].stages_need[s].total_seqlen
                for p_ in ov[rank][r]:
                    src = self.plans[r].stages_need[s].make_range_local(p_)
                    dst = my_hr.make_range_local(p_, is_self_merged=True)
                    n = p_.seqlen
                    red.append((r, src.start, dst.start, n))            # dk
                    red.append((r, S_r + src.start, L + dst.start, n))  # dv
            cast_consumers = sum(
                1 for r in range(cp) if ov[rank][r].total_seqlen > 0
            )
            reduce_consumers = sum(
                1 for o in range(cp) if ov[o][rank].total_seqlen > 0
            )
            out.append(NativeStageMeta(
                cast_pieces=cast, reduce_pieces=red, stage_tokens=S,
                cast_consumers=cast_consumers,
                reduce_consumers=reduce_consumers,
            ))
            stage_tokens_all.append(
                [self.plans[r].stages_need[s].total_seqlen for r in range(cp)]
            )
        return out, stage_tokens_all

    # ---------------- QO-comm (solver runs on the TRANSPOSED mask) --------
    def make_qo_calc_meta(self, rank: int) -> CalcMeta:
        """Stage args with the ORIGINAL orientation restored: this solver was
        built on transposed slices (q' = k), so host/remote slices transpose
        back before local-coordinate mapping. Stage q coords live in the
        stage-need buffer space; k coords in my hosted rows."""
        plan = self.plans[rank]
        hr = plan.host_ranges

        def restored(slices_t: List[MaskSlice], qspace: AttnRanges,
                     q_merged: bool) -> AttnArg:
            qr, kr, tm = [], [], []
            area = 0
            mx = 0
            for sl_t in slices_t:
                for o in transpose_slice(sl_t):  # back to original orientation
                    lq = (qspace.make_range_local(AttnRange(o.qs, o.qe),
                                                  is_self_merged=True)
                          if q_merged else
                          qspace.make_range_local(AttnRange(o.qs, o.qe)))
                    lk = hr.make_range_local(AttnRange(o.ks, o.ke),
                                             is_self_merged=True)
                    qr.append((lq.start, lq.end))
                    kr.append((lk.start, lk.end))
                    tm.append(o.t)
                    area += o.area()
                    mx = max(mx, lq.end - lq.start)
            return AttnArg(qr, kr, tm, max_seqlen_q=mx, total_area=area)

        host_arg = restored(plan.host_slices, hr, True)
        stage_args = []
        for s in range(self.overlap_degree):
            st = plan.stages_need[s]
            stage_slices: List[MaskSlice] = []
            for sl in plan.remote_slices:
                for piece in st:
                    stage_slices.extend(k_window(sl, piece.start, piece.end))
            stage_args.append(restored(stage_slices, st, False))
        return CalcMeta(host_arg=host_arg, stage_args=stage_args)

    def make_qo_comm_meta(self, rank: int) -> QoCommMeta:
        m1 = self.make_comm_meta(rank, ncopies=1)
        m2 = self.make_comm_meta(rank, ncopies=2)
        return QoCommMeta(
            calc=self.make_qo_calc_meta(rank),
            stages_cast1=m1.stages_cast,
            stages_cast2=m2.stages_cast,
            stages_reduce1=m1.stages_reduce,
        )

    # ---------------- hierarchical comm meta ----------------
    def make_hier_comm_meta(
        self, rank: int, ws_intra: int, ws_inter: int
    ) -> Tuple[List[HierGroupCastArg], List[HierGroupReduceArg]]:
        """Hierarchical (2D-mesh) realisation of the same stages. Rank layout
        is row-major over (inter, intra): node(r)=r//ws_intra, loc(r)=r%ws_intra
        (reference comm_meta.py:227 intra=mesh dim 1, inter=mesh dim 0).

        Cast: a row needed by several ranks of a remote node crosses the
        inter-node wire ONCE, to the proxy with the owner's local rank, which
        forwards it intra-node. Reduce mirrors this: partials for a remote
        owner are first SUMMED on the in-node proxy, then one node-sum crosses
        per contributing node. Unlike the reference
        (_group_collective_hier.py:319 all_gather_object at call time), every
        table is derived here at plan time from the global overlap table.
        """
        assert ws_intra * ws_inter == self.cp_size
        wi, wn = ws_intra, ws_inter
        n0, l0 = rank // wi, rank % wi

        def g(n: int, l: int) -> int:
            return n * wi + l

        def union(ranges_list: List[AttnRanges]) -> AttnRanges:
            u = AttnRanges()
            for rr in ranges_list:
                for p in rr:
                    u.append(p.clone())
            return u.merge()

        my_hr = self.host_ranges_all[rank]
        L = my_hr.total_seqlen
        casts: List[HierGroupCastArg] = []
        reduces: List[HierGroupReduceArg] = []

        for s in range(self.overlap_degree):
            ov = self._overlap_table(s)
            my_need = self.plans[rank].stages_need[s]
            S = my_need.total_seqlen

            def loc_host(p):
                lp = my_hr.make_range_local(p, is_self_merged=True)
                return lp.start, lp.end

            def loc_need(p):
                lp = my_need.make_range_local(p)
                return lp.start, lp.end

            # ---------- cast phase 1: pre-intra (same-node direct) ----------
            in_r, out_s, in_sp = [], [], []
            cur = 0
            for l in range(wi):
                pieces = ov[rank][g(n0, l)]
                tok = pieces.total_seqlen
                vcur = cur + tok
                for p in pieces:
                    a, b = loc_host(p)
                    in_r.append((a, b)); out_s.append(cur); cur += b - a
                for p in pieces:
                    a, b = loc_host(p)
                    in_r.append((L + a, L + b)); out_s.append(vcur); vcur += b - a
                cur = vcur
                in_sp.append(2 * tok)
            pre_send = RowChunkMap(in_r, out_s, cur)

            in_r, out_s, out_sp = [], [], []
            rcur = 0
            for l in range(wi):
                pieces = ov[g(n0, l)][rank]
                tok = pieces.total_seqlen
                for p in pieces:
                    a, _ = loc_need(p)
                    in_r.append((rcur, rcur + p.seqlen)); out_s.append(a)
                    rcur += p.seqlen
                for p in pieces:
                    a, _ = loc_need(p)
                    in_r.append((rcur, rcur + p.seqlen)); out_s.append(S + a)
                    rcur += p.seqlen
                out_sp.append(2 * tok)
            pre = GroupCastArg(
                send_pack=pre_send, input_split_sizes=in_sp,
                recv_unpack=RowChunkMap(in_r, out_s, rcur),
                output_split_sizes=out_sp, stage_tokens=S,
            )

            # ---------- cast phase 2: inter (one dedup copy per node) -------
            # send: union over the destination node's ranks -> proxy (n, l0)
            in_r, out_s, in_sp = [], [], []
            cur = 0
            for n in range(wn):
                u = (union([ov[rank][g(n, l)] for l in range(wi)])
                     if n != n0 else AttnRanges())
                tok = u.total_seqlen
                vcur = cur + tok
                for p in u:
                    a, b = loc_host(p)
                    in_r.append((a, b)); out_s.append(cur); cur += b - a
                for p in u:
                    a, b = loc_host(p)
                    in_r.append((L + a, L + b)); out_s.append(vcur); vcur += b - a
                cur = vcur
                in_sp.append(2 * tok)
            inter_send = RowChunkMap(in_r, out_s, cur)
            inter_in_splits = in_sp

            # recv (me as proxy): from owner (n, l0): union over my node's dsts
            proxy_u = [
                (union([ov[g(n, l0)][g(n0, l)] for l in range(wi)])
                 if n != n0 else AttnRanges())
                for n in range(wn)
            ]
            inter_out_splits = [2 * u.total_seqlen for u in proxy_u]
            inter_total_recv = sum(inter_out_splits)
            off = [0] * wn
            acc = 0
            for n in range(wn):
                off[n] = acc
                acc += inter_out_splits[n]

            # ---------- cast phase 3: post-intra (proxy -> final dst) -------
            in_r, out_s, in_sp = [], [], []
            cur = 0
            for l in range(wi):
                tok = sum(
                    ov[g(n, l0)][g(n0, l)].total_seqlen
                    for n in range(wn) if n != n0
                )
                vcur = cur + tok
                for pass_v in (False, True):
                    c = vcur if pass_v else cur
                    for n in range(wn):
                        if n == n0:
                            continue
                        u = proxy_u[n]
                        U = u.total_seqlen
                        base = off[n] + (U if pass_v else 0)
                        for p in ov[g(n, l0)][g(n0, l)]:
                            lp = u.make_range_local(p)
                            in_r.append((base + lp.start, base + lp.end))
                            out_s.append(c); c += p.seqlen
                    if pass_v:
                        vcur = c
                    else:
                        cur = c
                cur = vcur
                in_sp.append(2 * tok)
            post_send = RowChunkMap(in_r, out_s, cur)
            post_in_splits = in_sp

            in_r, out_s, out_sp = [], [], []
            rcur = 0
            for l in range(wi):
                tok = sum(
                    ov[g(n, l)][rank].total_seqlen
                    for n in range(wn) if n != n0
                )
                for pass_v in (False, True):
                    for n in range(wn):
                        if n == n0:
                            continue
                        for p in ov[g(n, l)][rank]:
                            a, _ = loc_need(p)
                            in_r.append((rcur, rcur + p.seqlen))
                            out_s.append(S + a if pass_v else a)
                            rcur += p.seqlen
                out_sp.append(2 * tok)
            casts.append(HierGroupCastArg(
                pre=pre,
                inter_send_pack=inter_send,
                inter_in_splits=inter_in_splits,
                inter_out_splits=inter_out_splits,
                inter_total_recv=inter_total_recv,
                post_send_pack=post_send,
                post_in_splits=post_in_splits,
                post_recv_unpack=RowChunkMap(in_r, out_s, rcur),
                post_out_splits=out_sp,
                stage_tokens=S,
            ))

            # ---------- reduce phase 1: pre-intra -------------------------
            # send to local rank l: partials for owner (n, l) for EVERY node n
            # (n == n0: l is the owner itself; else: l is the owner's proxy)
            in_r, out_s, in_sp = [], [], []
            cur = 0
            for l in range(wi):
                tok = sum(ov[g(n, l)][rank].total_seqlen for n in range(wn))
                vcur = cur + tok
                for pass_v in (False, True):
                    c = vcur if pass_v else cur
                    for n in range(wn):
                        for p in ov[g(n, l)][rank]:
                            a, b = loc_need(p)
                            if pass_v:
                                a, b = S + a, S + b
                            in_r.append((a, b)); out_s.append(c); c += b - a
                    if pass_v:
                        vcur = c
                    else:
                        cur = c
                cur = vcur
                in_sp.append(2 * tok)
            r_pre_send = RowChunkMap(in_r, out_s, cur)
            r_pre_in_splits = in_sp

            # recv from local rank lc: direct partials (owner me, n == n0) sum
            # into my dkv; proxied partials (owner (n, l0), n != n0) sum into
            # the proxy buffer laid out per node at off[n] (same proxy_u).
            d_in, d_out = [], []
            x_in, x_out = [], []
            out_sp = []
            rcur = 0
            for lc in range(wi):
                tok = sum(
                    ov[g(n, l0)][g(n0, lc)].total_seqlen for n in range(wn)
                )
                for pass_v in (False, True):
                    for n in range(wn):
                        u = proxy_u[n]
                        U = u.total_seqlen
                        for p in ov[g(n, l0)][g(n0, lc)]:
                            if n == n0:
                                a, _b = loc_host(p)
                                d_in.append((rcur, rcur + p.seqlen))
                                d_out.append(L + a if pass_v else a)
                            else:
                                lp = u.make_range_local(p)
                                base = off[n] + (U if pass_v else 0)
                                x_in.append((rcur, rcur + p.seqlen))
                                x_out.append(base + lp.start)
                            rcur += p.seqlen
                out_sp.append(2 * tok)
            proxy_rows = inter_total_recv  # same layout as the cast proxy buf

            # ---------- reduce phase 2: inter (node-sum -> owner) ----------
            # send: the proxy buffer itself, segment n = [2*U_n] rows
            r_inter_in_splits = list(inter_out_splits)
            # recv: node-sums for MY hosted rows from each contributing node
            in_r2, out_s2, out_sp2 = [], [], []
            rcur2 = 0
            for n in range(wn):
                u = (union([ov[rank][g(n, l)] for l in range(wi)])
                     if n != n0 else AttnRanges())
                tok = u.total_seqlen
                for pass_v in (False, True):
                    for p in u:
                        a, _b = loc_host(p)
                        in_r2.append((rcur2, rcur2 + p.seqlen))
                        out_s2.append(L + a if pass_v else a)
                        rcur2 += p.seqlen
                out_sp2.append(2 * tok)
            reduces.append(HierGroupReduceArg(
                pre_send_pack=r_pre_send,
                pre_in_splits=r_pre_in_splits,
                pre_recv_direct=RowChunkMap(d_in, d_out, rcur),
                pre_recv_proxy=RowChunkMap(x_in, x_out, proxy_rows),
                pre_out_splits=out_sp,
                pre_total_recv=rcur,
                proxy_rows=proxy_rows,
                inter_in_splits=r_inter_in_splits,
                inter_recv_reduce=RowChunkMap(in_r2, out_s2, rcur2),
                inter_out_splits=out_sp2,
                inter_total_recv=rcur2,
            ))
        return casts, reduces
