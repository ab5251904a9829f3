"""Multi-process (gloo, CPU) tests of the full API flow:
key -> dispatch -> calc_attn -> undispatch -> backward, vs the global oracle.
The attention math runs through the test oracle backend (tests/dist_backend.py)
so what is validated here is the planner + comm + runtime machinery — the HIP
kernel itself is validated on GPU in tests/test_ffa_gpu.py."""
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


CASES = {
    "varlen_causal": dict(
        total=768,
        q_ranges=[[0, 256], [256, 640], [640, 768]],
        k_ranges=[[0, 256], [256, 640], [640, 768]],
        types=[1, 1, 1],
    ),
    "mixed_types": dict(
        total=640,
        q_ranges=[[0, 200], [200, 512], [512, 640]],
        k_ranges=[[0, 200], [200, 512], [512, 640]],
        types=[0, 1, 3],
    ),
    "sliding_window": None,  # built in-proc from the sliding-window helper
    # reference test_pipeline-style configs (scaled down):
    "q_overlap": dict(  # continuous multi-masks with q overlap
        total=640,
        q_ranges=[[0, 256], [0, 256], [256, 640], [256, 640]],
        k_ranges=[[0, 256], [256, 448], [256, 640], [0, 128]],
        types=[1, 0, 1, 0],
    ),
    "full_from_pieces": dict(  # full mask assembled from small pieces
        total=512,
        q_ranges=[[0, 128], [128, 384], [384, 512]],
        k_ranges=[[0, 512], [0, 512], [0, 512]],
        types=[0, 0, 0],
    ),
    "unaligned_total": dict(  # exercises compute_pad_size > 0 (padded rows)
        total=700,
        q_ranges=[[0, 300], [300, 700]],
        k_ranges=[[0, 300], [300, 700]],
        types=[1, 1],
    ),
}


def _worker(rank, ws, port, case_name, degree, q_data):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=ws)
    try:
        import magi_attention.functional.dist_attn as da
        from magi_attention.api import (
            calc_attn,
            dispatch,
            magi_attn_flex_key,
            undispatch,
        )
        from magi_attention.common.range import AttnRange
        from magi_attention.common.ranges import AttnRanges
        from magi_attention.config import (
            DispatchConfig,
            DistAttnConfig,
            OverlapConfig,
        )
        from tests.dist_backend import OracleBackend

        da.register_test_attn_backend(OracleBackend)

        if case_name == "sliding_window":
            from magi_attention.api import infer_attn_mask_from_sliding_window

            total = 768
            qr1, kr1, tt1 = infer_attn_mask_from_sliding_window(
                AttnRange(0, 512), AttnRange(0, 512), (127, 0)
            )
            qr2, kr2, tt2 = infer_attn_mask_from_sliding_window(
                AttnRange(512, 768), AttnRange(512, 768), (63, 32)
            )
            q_ranges = AttnRanges()
            q_ranges.extend(qr1)
            q_ranges.extend(qr2)
            k_ranges = AttnRanges()
            k_ranges.extend(kr1)
            k_ranges.extend(kr2)
            types = tt1 + tt2
            case = dict(
                total=total,
                q_ranges=[[r.start, r.end] for r in q_ranges],
                k_ranges=[[r.start, r.end] for r in k_ranges],
                types=[t.to_int_type() for t in types],
            )
        else:
            case = CASES[case_name]

        total = case["total"]
        hq, hk, d = (8, 2, 64) if case_name == "mixed_types" else (4, 2, 32)
        g = torch.Generator().manual_seed(17)
        q = torch.randn(total, hq, d, generator=g, dtype=torch.float64)
        k = torch.randn(total, hk, d, generator=g, dtype=torch.float64)
        v = torch.randn(total, hk, d, generator=g, dtype=torch.float64)
        dout = torch.randn(total, hq, d, generator=g, dtype=torch.float64)

        cfg = DistAttnConfig(
            dispatch_config=DispatchConfig(chunk_size=64),
            overlap_config=OverlapConfig(degree=degree, min_chunk_size=32),
        )
        key = magi_attn_flex_key(
            AttnRanges.from_ranges(case["q_ranges"]),
            AttnRanges.from_ranges(case["k_ranges"]),
            case["types"],
            total, total, hq, hk, d,
            cp_group_or_mesh=dist.group.WORLD,
            dist_attn_config=cfg,
        )
        ql = dispatch(q, key).requires_grad_(True)
        kl = dispatch(k, key).requires_grad_(True)
        vl = dispatch(v, key).requires_grad_(True)
        out_l, meta_l = calc_attn(ql, kl, vl, key)
        out_full = undispatch(out_l, key)

        # global oracle
        mask = make_attn_mask(
            total, total, case["q_ranges"], case["k_ranges"], case["types"]
        )
        ref_o, ref_lse = ref_attn(q, k, v, mask)
        torch.testing.assert_close(out_full, ref_o, atol=1e-5, rtol=1e-4)

        # backward: loss = sum(out * dout) over GLOBAL rows
        dout_l = dispatch(dout, key)
        (out_l * dout_l).sum().backward()

        # reference grads
        qg = q.clone().requires_grad_(True)
        kg = k.clone().requires_grad_(True)
        vg = v.clone().requires_grad_(True)
        ro, _ = ref_attn(qg, kg, vg, mask)
        (ro * dout).sum().backward()
        from magi_attention.api import get_position_ids

        pos = get_position_ids(key)
        pad = key.pad_size
        dq_ref_pad = torch.cat([qg.grad, torch.zeros(pad, hq, d, dtype=qg.grad.dtype)])
        dk_ref_pad = torch.cat([kg.grad, torch.zeros(pad, hk, d, dtype=kg.grad.dtype)])
        dv_ref_pad = torch.cat([vg.grad, torch.zeros(pad, hk, d, dtype=vg.grad.dtype)])
        torch.testing.assert_close(ql.grad, dq_ref_pad[pos], atol=1e-5, rtol=1e-4)
        torch.testing.assert_close(kl.grad, dk_ref_pad[pos], atol=1e-5, rtol=1e-4)
        torch.testing.assert_close(vl.grad, dv_ref_pad[pos], atol=1e-5, rtol=1e-4)
    finally:
        dist.destroy_process_group()


@pytest.mark.parametrize("case_name", list(CASES.keys()))
@pytest.mark.parametrize("ws,degree", [(2, 1), (2, 2), (4, 2)])
def test_dist_attn_cpu(case_name, ws, degree):
    port = _free_port()
    mp.spawn(_worker, args=(ws, port, case_name, degree, None), nprocs=ws,
             join=True)


def _worker_new_mask(rank, ws, port, _a, _b, _c):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=ws)
    try:
        import magi_attention.functional.dist_attn as da
        from magi_attention.api import (
            calc_attn, dispatch, magi_attn_flex_key,
            make_flex_key_for_new_mask_after_dispatch, undispatch,
        )
        from magi_attention.common.ranges import AttnRanges
        from magi_attention.config import DispatchConfig, DistAttnConfig
        from tests.dist_backend import OracleBackend

        da.register_test_attn_backend(OracleBackend)
        total, hq, hk, d = 512, 4, 2, 32
        g = torch.Generator().manual_seed(23)
        q = torch.randn(total, hq, d, generator=g, dtype=torch.float64)
        k = torch.randn(total, hk, d, generator=g, dtype=torch.float64)
        v = torch.randn(total, hk, d, generator=g, dtype=torch.float64)
        cfg = DistAttnConfig(dispatch_config=DispatchConfig(chunk_size=64))
        # dispatch under mask A (causal), compute under mask B (doc mask)
        key_a = magi_attn_flex_key(
            AttnRanges.from_ranges([[0, total]]),
            AttnRanges.from_ranges([[0, total]]),
            [1], total, total, hq, hk, d,
            cp_group_or_mesh=dist.group.WORLD, dist_attn_config=cfg,
        )
        qB = [[0, 200], [200, 512]]
        kB = [[0, 200], [200, 512]]
        tB = [1, 0]
        key_b = make_flex_key_for_new_mask_after_dispatch(
            AttnRanges.from_ranges(qB), AttnRanges.from_ranges(kB), tB,
            total, total, key_a,
        )
        ql = dispatch(q, key_a)
        kl = dispatch(k, key_a)
        vl = dispatch(v, key_a)
        out_b, _ = calc_attn(ql, kl, vl, key_b)
        full_b = undispatch(out_b, key_b)
        mask_b = make_attn_mask(total, total, qB, kB, tB)
        ref_b, _ = ref_attn(q, k, v, mask_b)
        torch.testing.assert_close(full_b, ref_b, atol=1e-5, rtol=1e-4)
    finally:
        dist.destroy_process_group()


def test_new_mask_after_dispatch():
    port = _free_port()
    mp.spawn(_worker_new_mask, args=(2, port, None, None, None), nprocs=2,
             join=True)


def _worker_asym_reduce(rank, ws, port, _a, _b, _c):
    """Regression: with SequentialDispatchAlg + causal, rank 0
    hosts the earliest chunks so its REMOTE need is empty (stage_tokens == 0)
    while the other ranks compute partial dK/dV for rank 0's KV rows. The
    backward group_reduce is a collective: rank 0 must still join every stage
    to RECEIVE those contributions."""
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=ws)
    try:
        import magi_attention.functional.dist_attn as da
        from magi_attention.api import calc_attn, dispatch, magi_attn_flex_key
        from magi_attention.common.ranges import AttnRanges
        from magi_attention.config import (
            DispatchConfig,
            DistAttnConfig,
            OverlapConfig,
            SequentialDispatchAlg,
        )
        from tests.dist_backend import OracleBackend

        da.register_test_attn_backend(OracleBackend)
        total, hq, hk, d = 512, 4, 2, 32
        g = torch.Generator().manual_seed(31)
        q = torch.randn(total, hq, d, generator=g, dtype=torch.float64)
        k = torch.randn(total, hk, d, generator=g, dtype=torch.float64)
        v = torch.randn(total, hk, d, generator=g, dtype=torch.float64)
        dout = torch.randn(total, hq, d, generator=g, dtype=torch.float64)
        cfg = DistAttnConfig(
            dispatch_config=DispatchConfig(
                chunk_size=64, alg=SequentialDispatchAlg()
            ),
            overlap_config=OverlapConfig(degree=1, min_chunk_size=32),
        )
        key = magi_attn_flex_key(
            AttnRanges.from_ranges([[0, total]]),
            AttnRanges.from_ranges([[0, total]]),
            [1], total, total, hq, hk, d,
            cp_group_or_mesh=dist.group.WORLD, dist_attn_config=cfg,
        )
        ql = dispatch(q, key).requires_grad_(True)
        kl = dispatch(k, key).requires_grad_(True)
        vl = dispatch(v, key).requires_grad_(True)
        out_l, _ = calc_attn(ql, kl, vl, key)
        dout_l = dispatch(dout, key)
        (out_l * dout_l).sum().backward()

        mask = make_attn_mask(total, total, [[0, total]], [[0, total]], [1])
        qg = q.clone().requires_grad_(True)
        kg = k.clone().requires_grad_(True)
        vg = v.clone().requires_grad_(True)
        ro, _ = ref_attn(qg, kg, vg, mask)
        (ro * dout).sum().backward()
        from magi_attention.api import get_position_ids

        pos = get_position_ids(key)
        torch.testing.assert_close(ql.grad, qg.grad[pos], atol=1e-5, rtol=1e-4)
        torch.testing.assert_close(kl.grad, kg.grad[pos], atol=1e-5, rtol=1e-4)
        torch.testing.assert_close(vl.grad, vg.grad[pos], atol=1e-5, rtol=1e-4)
    finally:
        dist.destroy_process_group()


@pytest.mark.parametrize("ws", [2, 4])
def test_asymmetric_stage_reduce(ws):
    port = _free_port()
    mp.spawn(_worker_asym_reduce, args=(ws, port, None, None, None), nprocs=ws,
             join=True)
