"""Distributed sink / softcap / max_logits through calc_attn (reference
api:1041 signature; dist sink = replicated [s_sink, hq], dsink all-reduced SUM;
max_logits all-reduced MAX — forward_meta.py:28). gloo ws=2 with the oracle
backend vs the global fp64 oracle."""
import os
import socket

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from oracle import make_attn_mask, ref_attn_with_grads


def _free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return p


def _worker(rank, ws, port, use_softcap):
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
        from magi_attention.common.ranges import AttnRanges
        from magi_attention.config import (
            DispatchConfig,
            DistAttnConfig,
            OverlapConfig,
        )
        from tests.dist_backend import OracleBackend

        da.register_test_attn_backend(OracleBackend)
        total, hq, hk, d, s_sink = 512, 4, 2, 32, 2
        qr_l = [[0, 256], [256, 512]]
        kr_l = [[0, 256], [128, 512]]
        tt = [1, 0]
        softcap = 12.0 if use_softcap else 0.0
        g = torch.Generator().manual_seed(41)
        q = torch.randn(total, hq, d, generator=g, dtype=torch.float64)
        k = torch.randn(total, hk, d, generator=g, dtype=torch.float64)
        v = torch.randn(total, hk, d, generator=g, dtype=torch.float64)
        dout = torch.randn(total, hq, d, generator=g, dtype=torch.float64)
        sink = (torch.randn(s_sink, hq, generator=g) * 2).double()

        cfg = DistAttnConfig(
            dispatch_config=DispatchConfig(chunk_size=64),
            overlap_config=OverlapConfig(degree=2, min_chunk_size=32),
        )
        key = magi_attn_flex_key(
            AttnRanges.from_ranges(qr_l), AttnRanges.from_ranges(kr_l), tt,
            total, total, hq, hk, d,
            cp_group_or_mesh=dist.group.WORLD, dist_attn_config=cfg,
        )
        ql = dispatch(q, key).requires_grad_(True)
        kl = dispatch(k, key).requires_grad_(True)
        vl = dispatch(v, key).requires_grad_(True)
        sink_l = sink.clone().requires_grad_(True)
        out_l, meta = calc_attn(
            ql, kl, vl, key, sink=sink_l, softcap=softcap,
            return_max_logits=True,
        )
        out_full = undispatch(out_l, key)

        mask = make_attn_mask(total, total, qr_l, kr_l, tt)
        hi = ref_attn_with_grads(q, k, v, mask, dout, softcap=softcap,
                                 sink=sink, sink_layout="sh")
        torch.testing.assert_close(out_full, hi[0], atol=1e-5, rtol=1e-4)

        # max_logits: global per-head max of the scaled (capped) logits
        qf = q.permute(1, 0, 2)
        kf = k.repeat_interleave(hq // hk, dim=1).permute(1, 0, 2)
        s = qf @ kf.transpose(-1, -2) * d ** -0.5
        if softcap > 0:
            s = softcap * torch.tanh(s / softcap)
        s = torch.where(mask.unsqueeze(0), s,
                        torch.full_like(s, float("-inf")))
        ml_ref = s.amax(dim=(-1, -2)).float()
        torch.testing.assert_close(meta.max_logits, ml_ref, atol=1e-5,
                                   rtol=1e-5)

        dout_l = dispatch(dout, key)
        (out_l * dout_l).sum().backward()
        from magi_attention.api import get_position_ids

        pos = get_position_ids(key)
        pad = key.pad_size
        for got, ref, h in ((ql.grad, hi[2], hq), (kl.grad, hi[3], hk),
                            (vl.grad, hi[4], hk)):
            ref_pad = torch.cat([ref, torch.zeros(pad, h, d, dtype=ref.dtype)])
            torch.testing.assert_close(got, ref_pad[pos], atol=1e-5, rtol=1e-4)
        torch.testing.assert_close(sink_l.grad, hi[5], atol=1e-5, rtol=1e-4)
    finally:
        dist.destroy_process_group()


@pytest.mark.parametrize("use_softcap", [False, True])
def test_dist_sink_softcap_max_logits(use_softcap):
    port = _free_port()
    mp.spawn(_worker, args=(2, port, use_softcap), nprocs=2, join=True)
