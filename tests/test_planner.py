"""Planner end-to-end checks (CPU, single process): the dispatch + attn solver
tables must reproduce the global mask exactly and the comm tables must be a
consistent a2av exchange. The distributed runtime is only a transport for
what is validated here."""
import random

import pytest
import torch

from magi_attention.config import DispatchConfig, DistAttnConfig, OverlapConfig
from magi_attention.meta import (
    make_attn_meta_from_dispatch_meta,
    make_dispatch_meta_from_qk_ranges,
    normalize_slices,
)
from magi_attention.meta.geometry import to_dense
from magi_attention.common.ranges import AttnRanges
from oracle import make_attn_mask, ref_attn
from oracle.ref_attn import merge_out_lse


def build_case(seed, total=1024, cp=4, chunk=128, degree=2):
    rng = random.Random(seed)
    # varlen docs with random types
    n_docs = rng.randrange(2, 6)
    cuts = sorted(rng.sample(range(64, total - 1), n_docs - 1))
    bounds = [0] + cuts + [total]
    q_ranges, k_ranges, types = [], [], []
    for a, b in zip(bounds, bounds[1:]):
        q_ranges.append([a, b])
        k_ranges.append([a, b])
        types.append(rng.choice([0, 1, 1, 2, 3]))
    # plus a couple of cross-doc slices — kept PAIR-DISJOINT from existing
    # slices (overlapping (q,k) pairs are invalid input: the kernel merges
    # per-slice softmax contributions, so a pair present twice would be
    # double-counted; reference semantics require disjoint pair sets)
    occupied = make_attn_mask(total, total, q_ranges, k_ranges, types)
    for _ in range(rng.randrange(0, 3)):
        a = rng.randrange(0, total - 64)
        b = rng.randrange(a + 32, min(a + 400, total) + 1)
        c = rng.randrange(0, total - 64)
        d = rng.randrange(c + 32, min(c + 400, total) + 1)
        t = rng.choice([0, 1, 2, 3])
        cand = make_attn_mask(total, total, [[a, b]], [[c, d]], [t])
        if (cand & occupied).any():
            continue
        occupied |= cand
        q_ranges.append([a, b])
        k_ranges.append([c, d])
        types.append(t)
    cfg = DistAttnConfig(
        dispatch_config=DispatchConfig(chunk_size=chunk),
        overlap_config=OverlapConfig(degree=degree, min_chunk_size=64),
    )
    slices = normalize_slices(
        AttnRanges.from_ranges(q_ranges), AttnRanges.from_ranges(k_ranges), types
    )
    return q_ranges, k_ranges, types, slices, cfg


@pytest.mark.parametrize("seed", range(6))
@pytest.mark.parametrize("cp", [1, 2, 4])
def test_planner_covers_global_mask(seed, cp):
    total, chunk = 1024, 128
    q_ranges, k_ranges, types, slices, cfg = build_case(seed, total, cp, chunk)
    dm = make_dispatch_meta_from_qk_ranges(slices, total, cp, 0, cfg)
    ref_mask = make_attn_mask(total, total, q_ranges, k_ranges, types)

    cover = torch.zeros(total, total, dtype=torch.int32)
    for r in range(cp):
        dm.cp_rank = r
        solver, calc, comm = make_attn_meta_from_dispatch_meta(slices, dm, cfg)
        plan = solver.plans[r]
        hr = plan.host_ranges
        # host slices are in global coords in the plan
        for sl in plan.host_slices + plan.remote_slices:
            cover += to_dense([sl], total, total).int()
        # stage args must cover all remote slices exactly once:
        st_tokens = sum(s.total_seqlen for s in plan.stages_need)
        # zero-redundancy: remote need == union of remote slice k ranges minus host
        need = AttnRanges()
        for sl in plan.remote_slices:
            need.append(AttnRanges.from_ranges([(sl.ks, sl.ke)])[0])
        assert st_tokens == need.merge().total_seqlen
        # comm table shape consistency
        for s in range(comm.overlap_degree):
            cast = comm.stages_cast[s]
            assert sum(cast.output_split_sizes) == cast.recv_unpack.total_rows
            assert sum(cast.input_split_sizes) == cast.send_pack.total_rows
    # each (q,k) allowed pair computed exactly once across all ranks
    assert torch.equal(cover.bool(), ref_mask)
    assert int(cover.max()) <= 1


@pytest.mark.parametrize("seed", range(4))
@pytest.mark.parametrize("cp,degree", [(2, 1), (4, 2), (4, 3)])
def test_planner_simulated_distributed_attention(seed, cp, degree):
    """Simulate the full CP forward with the planner's own tables (gather +
    a2av emulation + per-stage oracle attention + lse merge) and compare to
    the global oracle."""
    total, chunk = 1024, 128
    hq, hk, d = 4, 2, 32
    q_ranges, k_ranges, types, slices, cfg = build_case(seed, total, cp, chunk,
                                                        degree)
    g = torch.Generator().manual_seed(seed)
    q = torch.randn(total, hq, d, generator=g, dtype=torch.float64)
    k = torch.randn(total, hk, d, generator=g, dtype=torch.float64)
    v = torch.randn(total, hk, d, generator=g, dtype=torch.float64)

    ref_mask = make_attn_mask(total, total, q_ranges, k_ranges, types)
    ref_o, ref_lse = ref_attn(q, k, v, ref_mask)

    dm = make_dispatch_meta_from_qk_ranges(slices, total, cp, 0, cfg)

    solvers = []
    for r in range(cp):
        dm.cp_rank = r
        solvers.append(make_attn_meta_from_dispatch_meta(slices, dm, cfg))

    # precompute per-rank local rows
    def rows_of(rr):
        idx = []
        for p in rr:
            idx.extend(range(p.start, p.end))
        return torch.tensor(idx, dtype=torch.long)

    host_rows = [rows_of(solvers[r][0].host_ranges_all[r]) for r in range(cp)]

    # ---- emulate the a2av exchange per stage and check the stage buffers ----
    deg = solvers[0][2].overlap_degree
    for r in range(cp):
        assert solvers[r][2].overlap_degree == deg

    stage_kv = [[None] * deg for _ in range(cp)]
    for s in range(deg):
        # build send buffers
        send_buf = []
        for r in range(cp):
            _, _, comm = solvers[r]
            cast = comm.stages_cast[s]
            kv_local = torch.cat(
                [k[host_rows[r]], v[host_rows[r]]], dim=0
            )  # [2L, hk, d]
            buf = torch.zeros(cast.send_pack.total_rows, hk, d, dtype=k.dtype)
            for (a, b), o in zip(cast.send_pack.in_ranges, cast.send_pack.out_starts):
                buf[o:o + (b - a)] = kv_local[a:b]
            send_buf.append(buf)
        # exchange: dst r gets from each src o the o->r segment
        for r in range(cp):
            _, _, comm = solvers[r]
            cast = comm.stages_cast[s]
            recv = []
            for o in range(cp):
                ocast = solvers[o][2].stages_cast[s]
                start = sum(ocast.input_split_sizes[:r])
                recv.append(send_buf[o][start:start + ocast.input_split_sizes[r]])
                assert ocast.input_split_sizes[r] == cast.output_split_sizes[o]
            recv = torch.cat(recv) if recv else torch.zeros(0, hk, d)
            S = cast.stage_tokens
            st = torch.zeros(2 * S, hk, d, dtype=k.dtype)
            for (a, b), o in zip(cast.recv_unpack.in_ranges,
                                 cast.recv_unpack.out_starts):
                st[o:o + (b - a)] = recv[a:b]
            stage_kv[r][s] = st
            # check against direct gather of the stage's global ranges
            srows = rows_of(solvers[r][0].plans[r].stages_need[s])
            if len(srows):
                assert torch.equal(st[:S], k[srows])
                assert torch.equal(st[S:], v[srows])

    # ---- per-rank staged attention + merge, then compare ----
    for r in range(cp):
        solver, calc, comm = solvers[r]
        rows = host_rows[r]
        ql = q[rows]
        kl, vl = k[rows], v[rows]
        L = len(rows)
        outs, lses = [], []

        def run(arg, kk, vv):
            if arg.is_empty():
                return None
            m = make_attn_mask(
                L, kk.shape[0], arg.q_ranges, arg.k_ranges, arg.attn_type_map
            )
            return ref_attn(ql, kk, vv, m)

        res = run(calc.host_arg, kl, vl)
        if res:
            outs.append(res[0])
            lses.append(res[1])
        for s in range(deg):
            S = comm.stages_cast[s].stage_tokens
            if S == 0:
                continue
            res = run(calc.stage_args[s], stage_kv[r][s][:S], stage_kv[r][s][S:])
            if res:
                outs.append(res[0])
                lses.append(res[1])
        if not outs:
            continue
        out_m, lse_m = merge_out_lse(outs, lses)
        torch.testing.assert_close(
            out_m.to(torch.float64), ref_o[rows], atol=1e-6, rtol=1e-6
        )
        fin = torch.isfinite(ref_lse[rows])
        torch.testing.assert_close(
            lse_m.float()[fin], ref_lse[rows][fin], atol=1e-5, rtol=1e-5
        )


def test_merge_ranges_reference_example():
    """merge_ranges on the reference's own docstring example
    (flex_flash_attn.py:110-149)."""
    import torch

    from magi_attention.functional import merge_ranges

    outer = torch.tensor([[20, 30], [10, 20], [10, 20], [20, 30]],
                         dtype=torch.int32)
    inner = torch.tensor([[100, 110], [120, 130], [140, 150], [160, 170]],
                         dtype=torch.int32)
    tm = torch.tensor([0, 1, 0, 0], dtype=torch.int32)
    merged, so, si, st, inv, count = merge_ranges(outer, inner, tm)
    assert count.item() == 2
    assert merged[:2].tolist() == [[10, 20], [20, 30]]
    assert merged[2:].tolist() == [[0, 0], [0, 0]]  # zero padding
    assert so.tolist() == [[10, 20], [10, 20], [20, 30], [20, 30]]
    # inner/type rows follow their outer rows (stable within equal outers)
    assert si.tolist() == [[120, 130], [140, 150], [100, 110], [160, 170]]
    assert st.tolist() == [1, 0, 0, 0]
    assert inv.tolist() == [0, 0, 1, 1]

    from magi_attention.functional.flex_flash_attn import _seg_starts

    starts = _seg_starts(inv, 4)
    assert starts.tolist() == [0, 2, 4, 4, 4]  # padded tail = empty segments


def test_solver_coverage_fuzz():
    """Seeded random masks: the union of every rank's host+stage slices must
    equal the global mask exactly (coverage, no duplication), for random
    range sets across cp sizes."""
    import random

    import torch

    from magi_attention.common.ranges import AttnRanges
    from magi_attention.config import (
        DispatchConfig,
        DistAttnConfig,
        OverlapConfig,
    )
    from magi_attention.meta import (
        make_attn_meta_from_dispatch_meta,
        make_dispatch_meta_from_qk_ranges,
        normalize_slices,
    )
    from oracle import make_attn_mask

    rng = random.Random(1234)
    for trial in range(12):
        total = rng.choice([256, 384, 512])
        # slices must be pairwise cell-disjoint (the reference's mask
        # semantics: duplicated coverage would double-count in the softmax) —
        # resample any slice that overlaps the accumulated area
        n = rng.randint(1, 6)
        qr, kr, tt = [], [], []
        area = torch.zeros(total, total, dtype=torch.bool)
        for _ in range(n):
            for _attempt in range(20):
                a = rng.randrange(0, total - 32)
                b = rng.randrange(a + 16, min(a + 256, total) + 1)
                c = rng.randrange(0, total - 32)
                d_ = rng.randrange(c + 16, min(c + 256, total) + 1)
                t = rng.choice([0, 1, 2, 3])
                sub = make_attn_mask(total, total, [[a, b]], [[c, d_]], [t])
                if not (area & sub).any():
                    area |= sub
                    qr.append([a, b])
                    kr.append([c, d_])
                    tt.append(t)
                    break
        if not qr:
            continue
        cp = rng.choice([2, 4])
        cfg = DistAttnConfig(
            dispatch_config=DispatchConfig(chunk_size=32),
            overlap_config=OverlapConfig(
                degree=rng.choice([1, 2]), min_chunk_size=16
            ),
        )
        slices = normalize_slices(
            AttnRanges.from_ranges(qr), AttnRanges.from_ranges(kr), tt
        )
        want = make_attn_mask(total, total, qr, kr, tt)
        got = torch.zeros_like(want, dtype=torch.int32)
        for rank in range(cp):
            dm = make_dispatch_meta_from_qk_ranges(slices, total, cp, rank, cfg)
            solver, calc, _ = make_attn_meta_from_dispatch_meta(slices, dm, cfg)
            hr = solver.plans[rank].host_ranges
            # host slices: q local -> global via host ranges; k local == global
            # offsets within hosted/stage spaces
            def add(arg, k_space):
                rows = []
                for piece in hr:
                    rows.extend(range(piece.start, piece.end))
                ksp = []
                for piece in k_space:
                    ksp.extend(range(piece.start, piece.end))
                for (qs, qe), (ks, ke), t in zip(
                    arg.q_ranges, arg.k_ranges, arg.attn_type_map
                ):
                    sub = make_attn_mask(
                        qe - qs, ke - ks, [[0, qe - qs]], [[0, ke - ks]], [t]
                    )
                    for i in range(qe - qs):
                        for j in range(ke - ks):
                            if sub[i, j]:
                                got[rows[qs + i], ksp[ks + j]] += 1
            add(calc.host_arg, hr)
            for s, arg in enumerate(calc.stage_args):
                add(arg, solver.plans[rank].stages_need[s])
        assert torch.equal(got.bool(), want), f"trial {trial}: coverage mismatch"
        assert (got <= 1).all(), f"trial {trial}: duplicated area"


# ---- r2 solver depth: algs / overlap modes / uneven shards ----

def _alg_by_name(name):
    from magi_attention.config import (
        LBDispatchAlg, MinHeapDispatchAlg, SequentialDispatchAlg,
        SortedSequentialSelectAlg, ToppHeapDispatchAlg,
    )
    return {
        "minheap": MinHeapDispatchAlg(), "seq": SequentialDispatchAlg(),
        "sortedseq": SortedSequentialSelectAlg(),
        "topp": ToppHeapDispatchAlg(), "lb": LBDispatchAlg(),
    }[name]


@pytest.mark.parametrize("alg", ["minheap", "seq", "sortedseq", "topp", "lb"])
@pytest.mark.parametrize("mode_alg", [("static", "uniform"),
                                      ("dynamic", "uniform"),
                                      ("static", "greedy"),
                                      ("dynamic", "greedy")])
@pytest.mark.parametrize("seed", [1, 3])
def test_planner_alg_mode_matrix(alg, mode_alg, seed):
    """Every (dispatch alg x overlap mode x overlap alg) combination must
    still satisfy exact coverage + zero redundancy (the invariant of
    test_planner_covers_global_mask)."""
    from magi_attention.common.enum import AttnOverlapMode
    from magi_attention.config import GreedyOverlapAlg, UniformOverlapAlg

    mode, oalg = mode_alg
    total, cp, chunk = 1024, 4, 128
    q_ranges, k_ranges, types, slices, _ = build_case(seed, total, cp, chunk)
    cfg = DistAttnConfig(
        dispatch_config=DispatchConfig(chunk_size=chunk, alg=_alg_by_name(alg)),
        overlap_config=OverlapConfig(
            degree=2, min_chunk_size=64,
            mode=(AttnOverlapMode.DYNAMIC if mode == "dynamic"
                  else AttnOverlapMode.STATIC),
            alg=(GreedyOverlapAlg() if oalg == "greedy"
                 else UniformOverlapAlg()),
        ),
    )
    dm = make_dispatch_meta_from_qk_ranges(slices, total, cp, 0, cfg)
    ref_mask = make_attn_mask(total, total, q_ranges, k_ranges, types)
    cover = torch.zeros(total, total, dtype=torch.int32)
    for r in range(cp):
        dm.cp_rank = r
        solver, calc, comm = make_attn_meta_from_dispatch_meta(slices, dm, cfg)
        plan = solver.plans[r]
        for sl in plan.host_slices + plan.remote_slices:
            cover += to_dense([sl], total, total).int()
        need = AttnRanges()
        for sl in plan.remote_slices:
            need.append(AttnRanges.from_ranges([(sl.ks, sl.ke)])[0])
        st_tokens = sum(s.total_seqlen for s in plan.stages_need)
        assert st_tokens == need.merge().total_seqlen, "zero redundancy"
    assert torch.equal(cover.bool(), ref_mask)
    assert int(cover.max()) <= 1


def test_dispatch_solver_uneven_shards():
    from magi_attention.meta.solver.dispatch_solver import DispatchSolver
    from magi_attention.config import MinHeapDispatchAlg, SequentialDispatchAlg

    w = [5.0, 1.0, 3.0, 2.0, 8.0, 1.0, 4.0]  # 7 chunks, cp=3
    for alg in (MinHeapDispatchAlg(), SequentialDispatchAlg()):
        sol = DispatchSolver(alg).solve(w, 3, uneven_shard=True)
        counts = sorted(len(p) for p in sol.partitions)
        assert counts == [2, 2, 3]
        assert sorted(c for p in sol.partitions for c in p) == list(range(7))
        for p, l in zip(sol.partitions, sol.loads):
            assert abs(sum(w[c] for c in p) - l) < 1e-9


def test_topp_heap_affinity_reduces_spread():
    """IOU affinity must group chunks touching the same k rows onto the same
    rank when loads are comparable."""
    from magi_attention.meta.solver.dispatch_solver import DispatchSolver
    from magi_attention.config import ToppHeapDispatchAlg

    w = [1.0] * 8
    aff = [AttnRanges.from_ranges([[0, 100]]) for _ in range(4)] + \
          [AttnRanges.from_ranges([[100, 200]]) for _ in range(4)]
    sol = DispatchSolver(ToppHeapDispatchAlg(top_p=1.0)).solve(
        w, 2, affinities=aff)
    for p in sol.partitions:
        groups = {0 if c < 4 else 1 for c in p}
        assert len(groups) == 1, f"affinity should not mix groups: {sol.partitions}"
