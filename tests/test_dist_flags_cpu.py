"""Multi-process (gloo) parity tests for the r2 runtime flags:
QO-comm (MAGI_ATTENTION_QO_COMM), high-precision reduce
(MAGI_ATTENTION_{FORWARD,BACKWARD}_HIGH_PRECISION_REDUCE) and tail-reduce
hiding (MAGI_ATTENTION_BWD_HIDE_TAIL_REDUCE) — full API flow vs the global
oracle with the test attention backend (reference env/comm.py:72-120)."""
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
    "causal_dense": dict(
        total=512,
        q_ranges=[[0, 512]], k_ranges=[[0, 512]], types=[1],
    ),
    "varlen_mixed": dict(
        total=640,
        q_ranges=[[0, 200], [200, 512], [512, 640]],
        k_ranges=[[0, 200], [200, 512], [512, 640]],
        types=[0, 1, 3],
    ),
    "q_overlap": dict(
        total=640,
        q_ranges=[[0, 256], [0, 256], [256, 640], [256, 640]],
        k_ranges=[[0, 256], [256, 448], [256, 640], [0, 128]],
        types=[1, 0, 1, 0],
    ),
}


def _worker(rank, ws, port, case_name, degree, flags):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    for kk, vv in flags.items():
        os.environ[kk] = vv
    dist.init_process_group("gloo", rank=rank, world_size=ws)
    try:
        import magi_attention.functional.dist_attn as da
        from magi_attention.api import (
            calc_attn, dispatch, get_position_ids, magi_attn_flex_key,
            undispatch,
        )
        from magi_attention.common.ranges import AttnRanges
        from magi_attention.config import (
            DispatchConfig, DistAttnConfig, OverlapConfig,
        )
        from tests.dist_backend import OracleBackend

        da.register_test_attn_backend(OracleBackend)
        case = CASES[case_name]
        total = case["total"]
        hq, hk, d = 4, 2, 32
        g = torch.Generator().manual_seed(29)
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
            case["types"], total, total, hq, hk, d,
            cp_group_or_mesh=dist.group.WORLD, dist_attn_config=cfg,
        )
        ql = dispatch(q, key).requires_grad_(True)
        kl = dispatch(k, key).requires_grad_(True)
        vl = dispatch(v, key).requires_grad_(True)
        out_l, _ = calc_attn(ql, kl, vl, key)
        out_full = undispatch(out_l, key)
        mask = make_attn_mask(
            total, total, case["q_ranges"], case["k_ranges"], case["types"]
        )
        ref_o, _ = ref_attn(q, k, v, mask)
        fwd_hp = flags.get("MAGI_ATTENTION_FORWARD_HIGH_PRECISION_REDUCE") == "1"
        qo_lowp = (flags.get("MAGI_ATTENTION_QO_COMM") == "1" and not fwd_hp
                   and ws > 1)
        # QO + default (bf16) wire rounds the partial out on the wire
        ftol = dict(atol=2e-2, rtol=2e-2) if qo_lowp else             dict(atol=1e-5, rtol=1e-4)
        torch.testing.assert_close(out_full, ref_o, **ftol)

        dout_l = dispatch(dout, key)
        (out_l * dout_l).sum().backward()
        qg = q.clone().requires_grad_(True)
        kg = k.clone().requires_grad_(True)
        vg = v.clone().requires_grad_(True)
        ro, _ = ref_attn(qg, kg, vg, mask)
        (ro * dout).sum().backward()
        pos = get_position_ids(key)
        pad = key.pad_size
        dq_ref = torch.cat([qg.grad, torch.zeros(pad, hq, d, dtype=torch.float64)])
        dk_ref = torch.cat([kg.grad, torch.zeros(pad, hk, d, dtype=torch.float64)])
        dv_ref = torch.cat([vg.grad, torch.zeros(pad, hk, d, dtype=torch.float64)])
        # bf16-wire reduce (the default) loosens the grad tolerance; with the
        # HP flags (fp32 wire) the fp64-oracle tolerance tightens back
        hp = flags.get("MAGI_ATTENTION_BACKWARD_HIGH_PRECISION_REDUCE") == "1"
        qo = flags.get("MAGI_ATTENTION_QO_COMM") == "1"
        tol = dict(atol=1e-5, rtol=1e-4) if (hp or ws == 1) else \
            dict(atol=5e-2, rtol=5e-2)
        torch.testing.assert_close(ql.grad, dq_ref[pos], **tol)
        torch.testing.assert_close(kl.grad, dk_ref[pos], **tol)
        torch.testing.assert_close(vl.grad, dv_ref[pos], **tol)
        if qo and rank == 0:
            # the plan must actually BE a QO plan
            from magi_attention.api.magi_attn_interface import (
                dist_attn_runtime_dict_mgr,
            )
            assert dist_attn_runtime_dict_mgr[key].runtime.qo_meta is not None
    finally:
        dist.destroy_process_group()


FLAG_SETS = {
    "qo": {"MAGI_ATTENTION_QO_COMM": "1",
           "MAGI_ATTENTION_BACKWARD_HIGH_PRECISION_REDUCE": "1",
           "MAGI_ATTENTION_FORWARD_HIGH_PRECISION_REDUCE": "1"},
    "qo_lowp": {"MAGI_ATTENTION_QO_COMM": "1"},
    "hp_reduce": {"MAGI_ATTENTION_BACKWARD_HIGH_PRECISION_REDUCE": "1"},
    "lowp_reduce": {},  # default: bf16 dKV wire
    "hide_tail": {"MAGI_ATTENTION_BWD_HIDE_TAIL_REDUCE": "1",
                  "MAGI_ATTENTION_BACKWARD_HIGH_PRECISION_REDUCE": "1"},
}


@pytest.mark.parametrize("case_name", list(CASES.keys()))
@pytest.mark.parametrize("flagset", list(FLAG_SETS.keys()))
@pytest.mark.parametrize("ws", [2])
def test_dist_flags(case_name, flagset, ws):
    port = _free_port()
    mp.spawn(_worker, args=(ws, port, case_name, 2, FLAG_SETS[flagset]),
             nprocs=ws, join=True)


def test_qo_ws4():
    port = _free_port()
    mp.spawn(_worker, args=(4, port, "varlen_mixed", 2, FLAG_SETS["qo"]),
             nprocs=4, join=True)


def _worker_gatherv(rank, ws, port, _a, _b, _c):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=ws)
    try:
        from magi_attention.comm.primitive import all_gather_v, reduce_scatter_v

        sizes = [3, 5, 0, 7][:ws]
        g = torch.Generator().manual_seed(5 + rank)
        local = torch.randn(sizes[rank], 4, generator=g)
        full = all_gather_v(local, sizes, dist.group.WORLD)
        # reference: gather via object exchange
        obj = [None] * ws
        dist.all_gather_object(obj, local)
        torch.testing.assert_close(full, torch.cat(obj, dim=0))

        contrib = torch.randn(sum(sizes), 4,
                              generator=torch.Generator().manual_seed(50 + rank))
        mine = reduce_scatter_v(contrib.clone(), sizes, dist.group.WORLD)
        obj = [None] * ws
        dist.all_gather_object(obj, contrib)
        total = torch.stack(obj).sum(0)
        start = sum(sizes[:rank])
        torch.testing.assert_close(mine, total[start:start + sizes[rank]])
    finally:
        dist.destroy_process_group()


@pytest.mark.parametrize("ws", [2, 4])
def test_uneven_gather_reduce_scatter(ws):
    port = _free_port()
    mp.spawn(_worker_gatherv, args=(ws, port, None, None, None), nprocs=ws,
             join=True)


def _worker_functional(rank, ws, port, _a, _b, _c):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=ws)
    try:
        from magi_attention.comm.functional import (
            all_gather_fwd_scatter_bwd,
            scatter_fwd_all_gather_bwd,
        )

        group = dist.group.WORLD
        sizes = [3, 5, 2, 7][:ws]
        g = torch.Generator().manual_seed(9 + rank)
        local = torch.randn(sizes[rank], 4, generator=g).requires_grad_(True)

        # gather-fwd: output is the concatenation; backward scatters the
        # grad back to this rank's slice
        full = all_gather_fwd_scatter_bwd(local, group, dim=0,
                                          split_sizes=sizes)
        obj = [None] * ws
        dist.all_gather_object(obj, local.detach())
        torch.testing.assert_close(full.detach(), torch.cat(obj, dim=0))
        gout = torch.randn(full.shape, generator=torch.Generator().manual_seed(77))
        full.backward(gout)
        start = sum(sizes[:rank])
        torch.testing.assert_close(local.grad,
                                   gout[start:start + sizes[rank]])

        # scatter-fwd: output is this rank's slice of the input; backward
        # all-gathers the slice grads
        big = torch.randn(sum(sizes), 4,
                          generator=torch.Generator().manual_seed(31)
                          ).requires_grad_(True)
        mine = scatter_fwd_all_gather_bwd(big, group, dim=0, split_sizes=sizes)
        torch.testing.assert_close(mine.detach(),
                                   big.detach()[start:start + sizes[rank]])
        gmine = torch.randn(mine.shape,
                            generator=torch.Generator().manual_seed(100 + rank))
        mine.backward(gmine)
        obj = [None] * ws
        dist.all_gather_object(obj, gmine)
        torch.testing.assert_close(big.grad, torch.cat(obj, dim=0))

        # even-split default (split_sizes=None), non-zero dim
        x = torch.randn(4, 2 * ws, generator=g).requires_grad_(True)
        piece = scatter_fwd_all_gather_bwd(x, group, dim=1, split_sizes=None)
        torch.testing.assert_close(
            piece.detach(), x.detach()[:, 2 * rank : 2 * rank + 2]
        )
    finally:
        dist.destroy_process_group()


@pytest.mark.parametrize("ws", [2, 4])
def test_comm_functional_gather_scatter(ws):
    port = _free_port()
    mp.spawn(_worker_functional, args=(ws, port, None, None, None), nprocs=ws,
             join=True)
