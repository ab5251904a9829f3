"""Hierarchical 2D-mesh comm (reference _group_collective_hier.py,
MAGI_ATTENTION_HIERARCHICAL_COMM): pre-intra direct + one deduplicated
inter-node copy via the same-local-rank proxy + post-intra forward; reduce
mirrors with in-node partial sums. Validated against the flat collectives on
the same solver plan (gloo, ws=4 as a 2x2 mesh), plus a host-only dedup
traffic assertion and an end-to-end API run through a 2D DeviceMesh."""
import os
import socket

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from oracle import make_attn_mask, ref_attn


def _free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return p


# masks with heavy cross-rank k sharing (full pieces) and causal structure
MASK = dict(
    total=512,
    q_ranges=[[0, 192], [192, 384], [384, 512]],
    k_ranges=[[0, 512], [0, 384], [128, 512]],
    types=[0, 1, 2],
)


def _subgroups_2x2():
    """All ranks collectively create the 2x2 subgroups; returns mine."""
    rank = dist.get_rank()
    intra_groups = [dist.new_group([0, 1]), dist.new_group([2, 3])]
    inter_groups = [dist.new_group([0, 2]), dist.new_group([1, 3])]
    return intra_groups[rank // 2], inter_groups[rank % 2]


def _worker_collectives(rank, ws, port, *_):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["MAGI_ATTENTION_HIERARCHICAL_COMM"] = "1"
    dist.init_process_group("gloo", rank=rank, world_size=ws)
    try:
        from magi_attention.comm.primitive import (
            group_cast,
            group_reduce,
            hier_group_cast,
            hier_group_reduce,
        )
        from magi_attention.common.ranges import AttnRanges
        from magi_attention.config import (
            DispatchConfig,
            DistAttnConfig,
            OverlapConfig,
        )
        from magi_attention.dist_attn_runtime_mgr import (
            DistAttnRuntimeMgr,
            init_dist_attn_runtime_key,
        )

        intra, inter = _subgroups_2x2()
        cfg = DistAttnConfig(
            dispatch_config=DispatchConfig(chunk_size=64),
            overlap_config=OverlapConfig(degree=2, min_chunk_size=32),
        )
        key = init_dist_attn_runtime_key(
            AttnRanges.from_ranges(MASK["q_ranges"]),
            AttnRanges.from_ranges(MASK["k_ranges"]),
            MASK["types"], MASK["total"], MASK["total"], 0, 64, 4, 2, 32,
            dist.group.WORLD, cfg,
        )
        mgr = DistAttnRuntimeMgr(
            key, dist.group.WORLD, cfg, mesh_groups=(intra, inter, 2, 2)
        )
        cm = mgr.runtime.comm_meta
        assert cm.stages_cast_hier is not None
        L = mgr.runtime.total_local_q
        g = torch.Generator().manual_seed(100 + rank)
        # integer-valued floats: sums are exact, bitwise comparison is fair
        kv_local = torch.randint(
            -64, 64, (2 * L, 2, 8), generator=g
        ).float()
        for s in range(cm.overlap_degree):
            flat = group_cast(
                kv_local, cm.stages_cast[s], dist.group.WORLD
            ).wait_post_process()
            hier = hier_group_cast(
                kv_local, cm.stages_cast_hier[s], intra, inter
            ).wait_post_process()
            assert torch.equal(flat, hier), f"stage {s} cast mismatch"

            S = cm.stages_cast[s].stage_tokens
            partial = torch.randint(
                -64, 64, (2 * S, 2, 8), generator=g
            ).float()
            dst_flat = torch.zeros(2 * L, 2, 8)
            dst_hier = torch.zeros(2 * L, 2, 8)
            group_reduce(
                partial, dst_flat, cm.stages_reduce[s], dist.group.WORLD
            ).wait_post_process()
            hier_group_reduce(
                partial, dst_hier, cm.stages_reduce_hier[s], intra, inter
            ).wait_post_process()
            assert torch.equal(dst_flat, dst_hier), f"stage {s} reduce mismatch"
    finally:
        dist.destroy_process_group()


def test_hier_vs_flat_collectives():
    port = _free_port()
    mp.spawn(_worker_collectives, args=(4, port), nprocs=4, join=True)


def test_hier_dedup_traffic():
    """A row needed by both ranks of a remote node must cross the inter wire
    once: with a full mask every rank needs every hosted row, so flat sends
    each hosted row to 2 remote ranks while hier sends it once per node."""
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

    total, cp = 512, 4
    slices = normalize_slices(
        AttnRanges.from_ranges([[0, total]]),
        AttnRanges.from_ranges([[0, total]]),
        [0],
    )
    cfg = DistAttnConfig(
        dispatch_config=DispatchConfig(chunk_size=64),
        overlap_config=OverlapConfig(degree=1, min_chunk_size=32),
    )
    dm = make_dispatch_meta_from_qk_ranges(slices, total, cp, 0, cfg)
    solver, _, comm = make_attn_meta_from_dispatch_meta(slices, dm, cfg)
    casts_h, _ = solver.make_hier_comm_meta(0, 2, 2)
    for s in range(len(comm.stages_cast)):
        flat_inter_rows = sum(
            comm.stages_cast[s].input_split_sizes[r] for r in (2, 3)
        )
        hier_inter_rows = sum(casts_h[s].inter_in_splits)
        assert hier_inter_rows * 2 == flat_inter_rows, (
            f"stage {s}: hier={hier_inter_rows} flat={flat_inter_rows}"
        )
        assert hier_inter_rows > 0


def _worker_e2e(rank, ws, port, *_):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["MAGI_ATTENTION_HIERARCHICAL_COMM"] = "1"
    dist.init_process_group("gloo", rank=rank, world_size=ws)
    try:
        from torch.distributed.device_mesh import init_device_mesh

        import magi_attention.functional.dist_attn as da
        from magi_attention.api import (
            calc_attn,
            dispatch,
            magi_attn_flex_key,
            undispatch,
        )
        from magi_attention.common.ranges import AttnRanges
        from magi_attention.config import (
            DispatchConfig,
            DistAttnConfig,
            OverlapConfig,
        )
        from tests.dist_backend import OracleBackend

        da.register_test_attn_backend(OracleBackend)
        mesh = init_device_mesh(
            "cpu", (2, 2), mesh_dim_names=("inter", "intra")
        )
        total, hq, hk, d = MASK["total"], 4, 2, 32
        g = torch.Generator().manual_seed(31)
        q = torch.randn(total, hq, d, generator=g, dtype=torch.float64)
        k = torch.randn(total, hk, d, generator=g, dtype=torch.float64)
        v = torch.randn(total, hk, d, generator=g, dtype=torch.float64)
        dout = torch.randn(total, hq, d, generator=g, dtype=torch.float64)
        cfg = DistAttnConfig(
            dispatch_config=DispatchConfig(chunk_size=64),
            overlap_config=OverlapConfig(degree=2, min_chunk_size=32),
        )
        key = magi_attn_flex_key(
            AttnRanges.from_ranges(MASK["q_ranges"]),
            AttnRanges.from_ranges(MASK["k_ranges"]),
            MASK["types"], total, total, hq, hk, d,
            cp_group_or_mesh=mesh, dist_attn_config=cfg,
        )
        from magi_attention.api.magi_attn_interface import (
            dist_attn_runtime_dict_mgr,
        )

        assert (
            dist_attn_runtime_dict_mgr[key].runtime.use_hier
        ), "hier path not active through the mesh API"
        ql = dispatch(q, key).requires_grad_(True)
        kl = dispatch(k, key).requires_grad_(True)
        vl = dispatch(v, key).requires_grad_(True)
        out_l, _ = calc_attn(ql, kl, vl, key)
        out_full = undispatch(out_l, key)
        mask = make_attn_mask(
            total, total, MASK["q_ranges"], MASK["k_ranges"], MASK["types"]
        )
        ref_o, _ = ref_attn(q, k, v, mask)
        torch.testing.assert_close(out_full, ref_o, atol=1e-5, rtol=1e-4)

        dout_l = dispatch(dout, key)
        (out_l * dout_l).sum().backward()
        qg = q.clone().requires_grad_(True)
        kg = k.clone().requires_grad_(True)
        vg = v.clone().requires_grad_(True)
        ro, _ = ref_attn(qg, kg, vg, mask)
        (ro * dout).sum().backward()
        from magi_attention.api import get_position_ids

        pos = get_position_ids(key)
        pad = key.pad_size
        for got, ref, h in ((ql.grad, qg.grad, hq), (kl.grad, kg.grad, hk),
                            (vl.grad, vg.grad, hk)):
            ref_pad = torch.cat(
                [ref, torch.zeros(pad, h, d, dtype=ref.dtype)]
            )
            torch.testing.assert_close(got, ref_pad[pos], atol=1e-5, rtol=1e-4)
    finally:
        dist.destroy_process_group()


def test_hier_end_to_end_mesh_api():
    port = _free_port()
    mp.spawn(_worker_e2e, args=(4, port), nprocs=4, join=True)


def _worker_hier_det(rank, ws, port, *_):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["MAGI_ATTENTION_HIERARCHICAL_COMM"] = "1"
    os.environ["MAGI_ATTENTION_DETERMINISTIC_MODE"] = "1"
    dist.init_process_group("gloo", rank=rank, world_size=ws)
    try:
        from torch.distributed.device_mesh import init_device_mesh

        import magi_attention.functional.dist_attn as da
        from magi_attention.api import calc_attn, dispatch, magi_attn_flex_key
        from magi_attention.common.ranges import AttnRanges
        from magi_attention.config import (
            DispatchConfig,
            DistAttnConfig,
            OverlapConfig,
        )
        from tests.dist_backend import OracleBackend

        da.register_test_attn_backend(OracleBackend)
        mesh = init_device_mesh("cpu", (2, 2),
                                mesh_dim_names=("inter", "intra"))
        total, hq, hk, d = MASK["total"], 4, 2, 32
        g = torch.Generator().manual_seed(77)
        q = torch.randn(total, hq, d, generator=g, dtype=torch.float64)
        k = torch.randn(total, hk, d, generator=g, dtype=torch.float64)
        v = torch.randn(total, hk, d, generator=g, dtype=torch.float64)
        dout = torch.randn(total, hq, d, generator=g, dtype=torch.float64)
        cfg = DistAttnConfig(
            dispatch_config=DispatchConfig(chunk_size=64),
            overlap_config=OverlapConfig(degree=2, min_chunk_size=32),
        )

        def run():
            key = magi_attn_flex_key(
                AttnRanges.from_ranges(MASK["q_ranges"]),
                AttnRanges.from_ranges(MASK["k_ranges"]),
                MASK["types"], total, total, hq, hk, d,
                cp_group_or_mesh=mesh, dist_attn_config=cfg,
            )
            ql = dispatch(q, key).requires_grad_(True)
            kl = dispatch(k, key).requires_grad_(True)
            vl = dispatch(v, key).requires_grad_(True)
            out_l, _ = calc_attn(ql, kl, vl, key)
            (out_l * dispatch(dout, key)).sum().backward()
            return (out_l.detach().clone(), ql.grad.clone(),
                    kl.grad.clone(), vl.grad.clone())

        a, b = run(), run()
        for x, y, name in zip(a, b, ["out", "dq", "dk", "dv"]):
            assert torch.equal(x, y), f"hier+det {name} not repeatable"
        dist.barrier()
    finally:
        dist.destroy_process_group()


def test_hier_plus_deterministic_repeatable():
    """Hierarchical comm composed with deterministic mode: two identical runs
    must be bitwise identical (plan + reduce order both fixed). One retry for
    gloo's occasional std::terminate teardown race under mp.spawn."""
    for attempt in range(2):
        try:
            port = _free_port()
            mp.spawn(_worker_hier_det, args=(4, port), nprocs=4, join=True)
            return
        except mp.ProcessExitedException:
            if attempt == 1:
                raise
