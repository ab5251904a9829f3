"""GPU (single-device) tests of the distributed product path at cp=1:
key -> dispatch -> calc_attn (FFA HIP kernel) -> undispatch -> backward."""
import os

import pytest
import torch

pytestmark = pytest.mark.gpu

from oracle import make_attn_mask, ref_attn_with_grads  # noqa: E402
from tests.util import assert_close_to_ref  # noqa: E402

requires_gpu = pytest.mark.skipif(
    not torch.cuda.is_available(), reason="needs MI355X"
)


def _init_pg():
    import torch.distributed as dist

    if not dist.is_initialized():
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29533")
        dist.init_process_group("nccl", rank=0, world_size=1)
    return dist.group.WORLD


@requires_gpu
@pytest.mark.parametrize(
    "case",
    [
        ("varlen", 1536, [[0, 512], [512, 1280], [1280, 1536]], [1, 1, 0]),
        ("sliding", 1024, None, None),
    ],
    ids=["varlen", "sliding"],
)
def test_calc_attn_cp1_gpu(case):
    from magi_attention.api import (
        calc_attn,
        dispatch,
        magi_attn_flex_key,
        undispatch,
    )
    from magi_attention.common.range import AttnRange
    from magi_attention.common.ranges import AttnRanges
    from magi_attention.config import DispatchConfig, DistAttnConfig

    name, total, qranges, types = case
    group = _init_pg()
    if name == "sliding":
        from magi_attention.api import infer_attn_mask_from_sliding_window

        qr, kr, tt = infer_attn_mask_from_sliding_window(
            AttnRange(0, total), AttnRange(0, total), (255, 0)
        )
        qranges = [[r.start, r.end] for r in qr]
        kranges = [[r.start, r.end] for r in kr]
        types = [t.to_int_type() for t in tt]
    else:
        kranges = qranges

    hq, hk, d = 4, 2, 128
    key = magi_attn_flex_key(
        AttnRanges.from_ranges(qranges), AttnRanges.from_ranges(kranges),
        types, total, total, hq, hk, d,
        cp_group_or_mesh=group,
        dist_attn_config=DistAttnConfig(
            dispatch_config=DispatchConfig(chunk_size=256)
        ),
    )
    g = torch.Generator().manual_seed(3)
    q = (torch.randn(total, hq, d, generator=g) * 0.5).bfloat16().cuda()
    k = (torch.randn(total, hk, d, generator=g) * 0.5).bfloat16().cuda()
    v = (torch.randn(total, hk, d, generator=g) * 0.5).bfloat16().cuda()
    dout = (torch.randn(total, hq, d, generator=g) * 0.5).bfloat16().cuda()

    ql = dispatch(q, key).requires_grad_(True)
    kl = dispatch(k, key).requires_grad_(True)
    vl = dispatch(v, key).requires_grad_(True)
    out_l, meta_l = calc_attn(ql, kl, vl, key)
    lse_l = meta_l.lse
    out = undispatch(out_l, key)
    out_l.backward(dispatch(dout, key))
    torch.cuda.synchronize()

    mask = make_attn_mask(total, total, qranges, kranges, types)
    qc, kc, vc, doc = [t.cpu() for t in (q, k, v, dout)]
    o_hi, _, dq_hi, dk_hi, dv_hi = ref_attn_with_grads(qc, kc, vc, mask, doc)
    o_lo, _, dq_lo, dk_lo, dv_lo = ref_attn_with_grads(
        qc, kc, vc, mask, doc, high_precision=False, p_dtype=torch.bfloat16
    )
    assert_close_to_ref(out.cpu().float(), o_hi.float(), o_lo.float(), "cp1:out")
    from magi_attention.api import get_position_ids

    pos = get_position_ids(key).cpu()
    pad = key.pad_size

    def padded(t):
        return torch.cat([t, torch.zeros(pad, *t.shape[1:], dtype=t.dtype)])

    assert_close_to_ref(
        ql.grad.cpu().float(), padded(dq_hi).float()[pos],
        padded(dq_lo).float()[pos], "cp1:dq",
    )
    assert_close_to_ref(
        kl.grad.cpu().float(), padded(dk_hi).float()[pos],
        padded(dk_lo).float()[pos], "cp1:dk",
    )
    assert_close_to_ref(
        vl.grad.cpu().float(), padded(dv_hi).float()[pos],
        padded(dv_lo).float()[pos], "cp1:dv",
    )


@requires_gpu
def test_calc_attn_cp1_sink_max_logits():
    """Distributed runtime at cp=1 with sink + softcap + max_logits on the
    HIP kernel path (sink postprocess + dsink kernels under the runtime)."""
    from magi_attention.api import (
        calc_attn,
        dispatch,
        get_position_ids,
        magi_attn_flex_key,
        undispatch,
    )
    from magi_attention.common.ranges import AttnRanges
    from magi_attention.config import DispatchConfig, DistAttnConfig

    pg = _init_pg()
    total, hq, hk, d, s_sink = 1024, 4, 2, 128, 2
    qr_l = [[0, 512], [512, 1024]]
    kr_l = [[0, 512], [256, 1024]]
    tt = [1, 0]
    g = torch.Generator().manual_seed(91)
    q = (torch.randn(total, hq, d, generator=g) * 0.5).bfloat16().cuda()
    k = (torch.randn(total, hk, d, generator=g) * 0.5).bfloat16().cuda()
    v = (torch.randn(total, hk, d, generator=g) * 0.5).bfloat16().cuda()
    dout = (torch.randn(total, hq, d, generator=g) * 0.5).bfloat16().cuda()
    sink = (torch.randn(s_sink, hq, generator=g) * 2).float().cuda()

    cfg = DistAttnConfig(dispatch_config=DispatchConfig(chunk_size=256))
    key = magi_attn_flex_key(
        AttnRanges.from_ranges(qr_l), AttnRanges.from_ranges(kr_l), tt,
        total, total, hq, hk, d, cp_group_or_mesh=pg, dist_attn_config=cfg,
    )
    ql = dispatch(q, key).requires_grad_(True)
    kl = dispatch(k, key).requires_grad_(True)
    vl = dispatch(v, key).requires_grad_(True)
    sink_l = sink.clone().requires_grad_(True)
    out_l, meta = calc_attn(ql, kl, vl, key, sink=sink_l,
                            return_max_logits=True)
    out_full = undispatch(out_l, key)
    out_l.backward(dispatch(dout, key))
    torch.cuda.synchronize()

    mask = make_attn_mask(total, total, qr_l, kr_l, tt)
    qc, kc, vc, doc = [t.cpu() for t in (q, k, v, dout)]
    sc = sink.cpu()
    hi = ref_attn_with_grads(qc, kc, vc, mask, doc, sink=sc)
    lo = ref_attn_with_grads(qc, kc, vc, mask, doc, sink=sc,
                             high_precision=False, p_dtype=torch.bfloat16)
    assert_close_to_ref(out_full.cpu().float(), hi[0].float(), lo[0].float(),
                        "cp1sink:out")
    assert meta.max_logits is not None and meta.max_logits.shape == (hq,)

    pos = get_position_ids(key).cpu()
    pad = key.pad_size

    def padded(t):
        return torch.cat([t, torch.zeros(pad, *t.shape[1:], dtype=t.dtype)])

    for got, i, name, fl in [
        (ql.grad, 2, "dq", 3e-3), (kl.grad, 3, "dk", 3e-3),
        (vl.grad, 4, "dv", 3e-3),
    ]:
        assert_close_to_ref(got.cpu().float(), padded(hi[i]).float()[pos],
                            padded(lo[i]).float()[pos], f"cp1sink:{name}",
                            floor=fl)
    assert_close_to_ref(sink_l.grad.cpu(), hi[5].float(), lo[5].float(),
                        "cp1sink:dsink", floor=2e-2)


@requires_gpu
def test_calc_attn_cp1_auto_range_merge():
    """MAGI_ATTENTION_AUTO_RANGE_MERGE=1 under the runtime (cp=1): merged
    segment tables per FFA call, results match the unmerged path."""
    from magi_attention.api import calc_attn, dispatch, magi_attn_flex_key, undispatch
    from magi_attention.common.ranges import AttnRanges
    from magi_attention.config import DispatchConfig, DistAttnConfig

    pg = _init_pg()
    total, hq, hk, d = 1024, 4, 2, 128
    # block-sparse rows: repeated q ranges with scattered k blocks
    qr_l = [[0, 256], [0, 256], [256, 768], [256, 768], [768, 1024]]
    kr_l = [[0, 128], [512, 768], [0, 512], [768, 1024], [768, 1024]]
    tt = [0, 0, 0, 0, 1]
    g = torch.Generator().manual_seed(17)
    q = (torch.randn(total, hq, d, generator=g) * 0.5).bfloat16().cuda()
    k = (torch.randn(total, hk, d, generator=g) * 0.5).bfloat16().cuda()
    v = (torch.randn(total, hk, d, generator=g) * 0.5).bfloat16().cuda()
    dout = (torch.randn(total, hq, d, generator=g) * 0.5).bfloat16().cuda()
    cfg = DistAttnConfig(dispatch_config=DispatchConfig(chunk_size=256))

    def run():
        key = magi_attn_flex_key(
            AttnRanges.from_ranges(qr_l), AttnRanges.from_ranges(kr_l), tt,
            total, total, hq, hk, d, cp_group_or_mesh=pg,
            dist_attn_config=cfg,
        )
        ql = dispatch(q, key).requires_grad_(True)
        kl = dispatch(k, key).requires_grad_(True)
        vl = dispatch(v, key).requires_grad_(True)
        out_l, _ = calc_attn(ql, kl, vl, key)
        full = undispatch(out_l, key)
        out_l.backward(dispatch(dout, key))
        torch.cuda.synchronize()
        return full, ql.grad.clone(), kl.grad.clone(), vl.grad.clone()

    base = run()
    os.environ["MAGI_ATTENTION_AUTO_RANGE_MERGE"] = "1"
    try:
        merged = run()  # env snapshot differs -> fresh key/plan
    finally:
        del os.environ["MAGI_ATTENTION_AUTO_RANGE_MERGE"]
    for a, b, name in zip(base, merged, ["out", "dq", "dk", "dv"]):
        torch.testing.assert_close(a.float(), b.float(), atol=3e-2, rtol=3e-2)
