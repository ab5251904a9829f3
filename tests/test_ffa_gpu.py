"""GPU parity tests for the HIP FFA kernels (run on a real MI355X via gpurun).

Parity procedure: bf16 inputs -> HIP kernel vs the fp64 CPU oracle on the SAME
(bf16-rounded) inputs, budgeted by the bf16 oracle's own error (tests/util.py,
mirroring the reference's testing/precision.py calibration)."""
import ctypes

import pytest
import torch

pytestmark = pytest.mark.gpu

from oracle import make_attn_mask, ref_attn, ref_attn_with_grads  # noqa: E402
from tests.util import assert_close_to_ref, make_flex_case  # noqa: E402

requires_gpu = pytest.mark.skipif(
    not torch.cuda.is_available(), reason="needs MI355X"
)


@requires_gpu
def test_mfma_32x32x16_layout():
    """Empirically pin the MFMA fragment layout the kernels assume
    (A[i=l%32][k=(l/32)*8+e], B[k][j=l%32], D row=(r&3)+8*(r>>2)+4*(l>>5),
    col=l%32). Asymmetric operands catch transposes."""
    from magi_attention import _ffa_lib

    torch.manual_seed(0)
    A = torch.randn(32, 16).bfloat16().cuda()
    B = torch.randn(16, 32).bfloat16().cuda()
    D = torch.zeros(32, 32, dtype=torch.float32, device="cuda")
    lib = _ffa_lib.lib()
    rc = lib.magi_probe_mfma(
        ctypes.c_void_p(A.data_ptr()),
        ctypes.c_void_p(B.data_ptr()),
        ctypes.c_void_p(D.data_ptr()),
        _ffa_lib.current_stream_ptr(),
    )
    assert rc == 0
    torch.cuda.synchronize()
    ref = (A.float() @ B.float()).cpu()
    torch.testing.assert_close(D.cpu(), ref, atol=2e-2, rtol=2e-2)


CASES = [
    # name, tq, tk, hq, hk, d, q_ranges, k_ranges, types
    ("full_64", 64, 64, 2, 2, 64, [[0, 64]], [[0, 64]], [0]),
    ("causal_256", 256, 256, 4, 4, 128, [[0, 256]], [[0, 256]], [1]),
    ("causal_unaligned", 173, 211, 2, 1, 64, [[0, 173]], [[0, 211]], [1]),
    ("invcausal_160", 160, 96, 2, 2, 128, [[0, 160]], [[0, 96]], [2]),
    ("bicausal_192", 192, 224, 3, 1, 64, [[0, 192]], [[0, 224]], [3]),
    (
        "varlen_3doc",
        384, 384, 4, 2, 128,
        [[0, 128], [128, 320], [320, 384]],
        [[0, 128], [128, 320], [320, 384]],
        [1, 1, 1],
    ),
    (
        "overlap_q",  # overlapping q_ranges exercise the lock-merge epilogue
        256, 512, 2, 2, 64,
        [[0, 256], [64, 192], [0, 128]],
        [[0, 128], [128, 384], [384, 512]],
        [1, 0, 2],
    ),
    ("empty_rows", 128, 64, 1, 1, 64, [[0, 64]], [[0, 64]], [1]),
    ("tiny_5x2", 5, 2, 1, 1, 64, [[0, 5]], [[0, 2]], [1]),
    (
        "dense_2k",  # config-1 shape class
        2048, 2048, 8, 8, 64, [[0, 2048]], [[0, 2048]], [1],
    ),
    (
        "varlen_4k_gqa",  # config-2 shape class (scaled for CPU oracle)
        4096, 4096, 16, 4, 128,
        [[0, 1024], [1024, 2560], [2560, 3072], [3072, 4096]],
        [[0, 1024], [1024, 2560], [2560, 3072], [3072, 4096]],
        [1, 1, 1, 1],
    ),
]


@requires_gpu
@pytest.mark.parametrize("case", CASES, ids=[c[0] for c in CASES])
def test_ffa_fwd_parity(case):
    from magi_attention.functional import flex_flash_attn_func

    name, tq, tk, hq, hk, d, qr_l, kr_l, ty_l = case
    q, k, v, dout, qr, kr, tm = make_flex_case(tq, tk, hq, hk, d, qr_l, kr_l, ty_l)
    out, meta = flex_flash_attn_func(q, k, v, qr, kr, tm)
    torch.cuda.synchronize()

    mask = make_attn_mask(tq, tk, qr_l, kr_l, ty_l)
    qc, kc, vc = q.cpu(), k.cpu(), v.cpu()
    ref_o_hi, ref_lse = ref_attn(qc, kc, vc, mask, high_precision=True)
    ref_o_lo, _ = ref_attn(
        qc, kc, vc, mask, high_precision=False, p_dtype=torch.bfloat16
    )
    assert_close_to_ref(out.cpu().float(), ref_o_hi.float(), ref_o_lo.float(),
                        f"{name}:out")
    lse = meta.lse.cpu()
    finite = torch.isfinite(ref_lse)
    assert torch.equal(torch.isfinite(lse), finite), f"{name}: lse -inf pattern"
    torch.testing.assert_close(
        lse[finite], ref_lse[finite], atol=5e-3, rtol=1e-3
    )


@requires_gpu
@pytest.mark.parametrize(
    "case", [c for c in CASES if c[0] not in ("dense_2k",)], ids=[
        c[0] for c in CASES if c[0] not in ("dense_2k",)
    ]
)
def test_ffa_fwd_bwd_parity(case):
    from magi_attention.functional import flex_flash_attn_func

    name, tq, tk, hq, hk, d, qr_l, kr_l, ty_l = case
    q, k, v, dout, qr, kr, tm = make_flex_case(tq, tk, hq, hk, d, qr_l, kr_l, ty_l)
    q.requires_grad_(True)
    k.requires_grad_(True)
    v.requires_grad_(True)
    out, meta = flex_flash_attn_func(q, k, v, qr, kr, tm)
    out.backward(dout)
    torch.cuda.synchronize()

    mask = make_attn_mask(tq, tk, qr_l, kr_l, ty_l)
    qc, kc, vc, doc = [t.detach().cpu() for t in (q, k, v, dout)]
    o_hi, _, dq_hi, dk_hi, dv_hi = ref_attn_with_grads(qc, kc, vc, mask, doc)
    o_lo, _, dq_lo, dk_lo, dv_lo = ref_attn_with_grads(
        qc, kc, vc, mask, doc, high_precision=False, p_dtype=torch.bfloat16
    )
    assert_close_to_ref(q.grad.cpu().float(), dq_hi.float(), dq_lo.float(), f"{name}:dq")
    assert_close_to_ref(k.grad.cpu().float(), dk_hi.float(), dk_lo.float(), f"{name}:dk")
    assert_close_to_ref(v.grad.cpu().float(), dv_hi.float(), dv_lo.float(), f"{name}:dv")


@requires_gpu
def test_fwd_softcap():
    """Softcap parity is pinned by the kernel-formula restatement in the
    oracle (the reference's CPU oracle does not support softcap)."""
    from magi_attention.functional import flex_flash_attn_func

    tq = tk = 128
    q, k, v, dout, qr, kr, tm = make_flex_case(
        tq, tk, 2, 2, 64, [[0, tq]], [[0, tk]], [1]
    )
    out, meta = flex_flash_attn_func(q, k, v, qr, kr, tm, softcap=20.0)
    torch.cuda.synchronize()
    mask = make_attn_mask(tq, tk, [[0, tq]], [[0, tk]], [1])
    o_hi, _ = ref_attn(q.cpu(), k.cpu(), v.cpu(), mask, softcap=20.0)
    o_lo, _ = ref_attn(
        q.cpu(), k.cpu(), v.cpu(), mask,
        softcap=20.0, high_precision=False, p_dtype=torch.bfloat16,
    )
    assert_close_to_ref(out.cpu().float(), o_hi.float(), o_lo.float(), "softcap:out")


@requires_gpu
def test_bwd_softcap():
    """Softcap backward (tanh-derivative dscale path in both bwd passes)."""
    from magi_attention.functional import flex_flash_attn_func
    from oracle import ref_attn_with_grads

    tq = tk = 192
    q, k, v, dout, qr, kr, tm = make_flex_case(
        tq, tk, 4, 2, 128, [[0, tq]], [[0, tk]], [1], seed=9
    )
    q.requires_grad_(True)
    k.requires_grad_(True)
    v.requires_grad_(True)
    out, _ = flex_flash_attn_func(q, k, v, qr, kr, tm, softcap=15.0)
    out.backward(dout)
    torch.cuda.synchronize()
    mask = make_attn_mask(tq, tk, [[0, tq]], [[0, tk]], [1])
    qc, kc, vc, doc = [t.detach().cpu() for t in (q, k, v, dout)]
    hi = ref_attn_with_grads(qc, kc, vc, mask, doc, softcap=15.0)
    lo = ref_attn_with_grads(qc, kc, vc, mask, doc, softcap=15.0,
                             high_precision=False, p_dtype=torch.bfloat16)
    for g, ghi, glo, name in [
        (q.grad, hi[2], lo[2], "dq"),
        (k.grad, hi[3], lo[3], "dk"),
        (v.grad, hi[4], lo[4], "dv"),
    ]:
        assert_close_to_ref(g.cpu().float(), ghi.float(), glo.float(),
                            f"softcap:{name}")


@requires_gpu
def test_cross_launch_accumulation():
    """The CP runtime accumulates multiple kernel launches into one
    (out_acc, lse_acc) pair: splitting K across two calls must equal one call
    over the union (reference dist_attn.py fwd_out_lse_use_acc path)."""
    from magi_attention.functional.flex_flash_attn import _flex_flash_attn_forward

    tq, tk, hq, hk, d = 192, 256, 2, 2, 64
    q, k, v, dout, qr, kr, tm = make_flex_case(
        tq, tk, hq, hk, d, [[0, tq]], [[0, tk]], [0]
    )
    out = torch.zeros(tq, hq, d, dtype=torch.float32, device="cuda")
    lse = torch.full((tq, hq), float("-inf"), dtype=torch.float32, device="cuda")
    half = 128
    scale = d ** -0.5
    for ks, ke in ((0, half), (half, tk)):
        _flex_flash_attn_forward(
            q=q, k=k, v=v, sink=None, sink_layout="sh", out=out, lse=lse,
            q_ranges=torch.tensor([[0, tq]], dtype=torch.int32, device="cuda"),
            k_ranges=torch.tensor([[ks, ke]], dtype=torch.int32, device="cuda"),
            attn_type_map=None, softmax_scale=scale, softcap=0.0, out_type=None,
            disable_fwd_atomic_reduction=False, deterministic=False, sm_margin=0,
        )
    torch.cuda.synchronize()
    mask = make_attn_mask(tq, tk, [[0, tq]], [[0, tk]], [0])
    o_hi, lse_hi = ref_attn(q.cpu(), k.cpu(), v.cpu(), mask)
    o_lo, _ = ref_attn(q.cpu(), k.cpu(), v.cpu(), mask, high_precision=False,
                       p_dtype=torch.bfloat16)
    assert_close_to_ref(out.cpu(), o_hi.float(), o_lo.float(), "acc:out")
    torch.testing.assert_close(lse.cpu(), lse_hi, atol=5e-3, rtol=1e-3)


@requires_gpu
def test_range_gather_reduce():
    from magi_attention.ops import range_gather, range_reduce

    g = torch.Generator().manual_seed(1)
    x = torch.randn(100, 4, 32, generator=g).bfloat16().cuda()
    ranges = torch.tensor([[5, 20], [40, 45], [60, 100]], dtype=torch.int32)
    starts = torch.tensor([0, 15, 20], dtype=torch.int32)
    total = 60
    out = range_gather(x, ranges.cuda(), starts.cuda(), total)
    torch.cuda.synchronize()
    ref = torch.cat([x[5:20], x[40:45], x[60:100]])
    assert torch.equal(out.cpu(), ref.cpu())

    # sum-reduce back (f32)
    y = out.float()
    acc = torch.zeros(100, 4, 32, device="cuda")
    # reduce: ranges index into SOURCE y, out_starts into acc
    src_ranges = torch.tensor([[0, 15], [15, 20], [20, 60]], dtype=torch.int32)
    dst_starts = torch.tensor([5, 40, 60], dtype=torch.int32)
    range_reduce(y, acc, src_ranges.cuda(), dst_starts.cuda(), op="sum")
    range_reduce(y, acc, src_ranges.cuda(), dst_starts.cuda(), op="sum")
    torch.cuda.synchronize()
    ref2 = torch.zeros(100, 4, 32)
    ref2[5:20] = 2 * y[0:15].cpu()
    ref2[40:45] = 2 * y[15:20].cpu()
    ref2[60:100] = 2 * y[20:60].cpu()
    torch.testing.assert_close(acc.cpu(), ref2)


@requires_gpu
def test_correct_out_lse_kernel():
    """Fused merge of two partial (out,lse) sets vs the oracle merge
    (reference functional/utils.py:371)."""
    from magi_attention.ops import correct_out_lse
    from oracle.ref_attn import merge_out_lse

    g = torch.Generator().manual_seed(5)
    t, h, d = 100, 4, 64
    o1 = torch.randn(t, h, d, generator=g).float()
    o2 = torch.randn(t, h, d, generator=g).float()
    l1 = torch.randn(t, h, generator=g).float() * 3
    l2 = torch.randn(t, h, generator=g).float() * 3
    # some -inf rows on both sides
    l1[:5] = float("-inf")
    l2[3:8] = float("-inf")
    o1[:5] = 0
    o2[3:8] = 0
    ref_o, ref_l = merge_out_lse([o1, o2], [l1, l2])

    o1c, l1c = o1.cuda(), l1.cuda()
    correct_out_lse(o1c, l1c, o2.cuda(), l2.cuda())
    torch.cuda.synchronize()
    torch.testing.assert_close(o1c.cpu().double(), ref_o, atol=1e-5, rtol=1e-5)
    fin = torch.isfinite(ref_l)
    torch.testing.assert_close(l1c.cpu()[fin].double(), ref_l[fin], atol=1e-5,
                               rtol=1e-5)
    assert torch.equal(torch.isfinite(l1c.cpu()), fin)


@requires_gpu
def test_range_reduce_lse_weighted():
    """lse-weighted row reduce (reference _range_reduce.py:239): merging
    partial (out,lse) rows into a destination must equal the oracle merge."""
    from magi_attention.ops import range_reduce
    from oracle.ref_attn import merge_out_lse

    g = torch.Generator().manual_seed(6)
    t, h, d = 64, 2, 32
    dst_o = torch.randn(t, h, d, generator=g).float()
    dst_l = torch.randn(t, h, generator=g).float()
    src_o = torch.randn(40, h, d, generator=g).float()
    src_l = torch.randn(40, h, generator=g).float()

    in_ranges = torch.tensor([[0, 25], [25, 40]], dtype=torch.int32)
    out_starts = torch.tensor([10, 45], dtype=torch.int32)

    ref_o = dst_o.clone().double()
    ref_l = dst_l.clone().double()
    for (a, b), o in zip(in_ranges.tolist(), out_starts.tolist()):
        mo, ml = merge_out_lse(
            [ref_o[o:o + b - a], src_o[a:b]], [ref_l[o:o + b - a].float(), src_l[a:b]]
        )
        ref_o[o:o + b - a] = mo
        ref_l[o:o + b - a] = ml

    do, dl = dst_o.cuda(), dst_l.cuda()
    range_reduce(src_o.cuda(), do, in_ranges.cuda(), out_starts.cuda(),
                 op="lse", in_lse=src_l.cuda(), out_lse=dl)
    torch.cuda.synchronize()
    torch.testing.assert_close(do.cpu().double(), ref_o, atol=1e-5, rtol=1e-5)
    torch.testing.assert_close(dl.cpu().double(), ref_l, atol=1e-5, rtol=1e-5)


@requires_gpu
@pytest.mark.parametrize("layout", ["sh", "ssh"])
def test_sink_fwd_bwd(layout):
    """Attention sinks (reference flash_fwd_postprocess_kernel.h:39 +
    bwd-preprocess dsink): value-less logits in the softmax denominator,
    folded once after the merge; rows with no allowed key get lse=lse_sink,
    out=0. Covers both layouts and empty rows (q rows outside every range)."""
    from magi_attention.functional import flex_flash_attn_func
    from oracle import ref_attn_with_grads

    tq = tk = 256
    hq, hk, d = 4, 2, 128
    s_sink = 3
    # rows 192..256 are in NO q_range -> empty rows, sink only
    qr_l, kr_l, tt = [[0, 128], [128, 192]], [[0, 192], [64, 256]], [1, 0]
    q, k, v, dout, qr, kr, tm = make_flex_case(
        tq, tk, hq, hk, d, qr_l, kr_l, tt, seed=77
    )
    g = torch.Generator().manual_seed(78)
    shape = (s_sink, hq) if layout == "sh" else (tq, s_sink, hq)
    sink = (torch.randn(*shape, generator=g) * 2.0).float().cuda()

    q.requires_grad_(True)
    k.requires_grad_(True)
    v.requires_grad_(True)
    sink.requires_grad_(True)
    out, meta = flex_flash_attn_func(
        q, k, v, qr, kr, tm, sink=sink, sink_layout=layout
    )
    out.backward(dout)
    torch.cuda.synchronize()

    mask = make_attn_mask(tq, tk, qr_l, kr_l, tt)
    qc, kc, vc, doc = [t.detach().cpu() for t in (q, k, v, dout)]
    sc = sink.detach().cpu()
    hi = ref_attn_with_grads(qc, kc, vc, mask, doc, sink=sc, sink_layout=layout)
    lo = ref_attn_with_grads(qc, kc, vc, mask, doc, sink=sc, sink_layout=layout,
                             high_precision=False, p_dtype=torch.bfloat16)
    assert_close_to_ref(out.detach().cpu().float(), hi[0].float(),
                        lo[0].float(), "sink:out")
    assert_close_to_ref(meta.lse.cpu(), hi[1], lo[1], "sink:lse")
    for gv, ghi, glo, name in [
        (q.grad, hi[2], lo[2], "dq"),
        (k.grad, hi[3], lo[3], "dk"),
        (v.grad, hi[4], lo[4], "dv"),
        (sink.grad, hi[5], lo[5], "dsink"),
    ]:
        # dsink ("sh") is a sum over all tq rows: cancellation inflates the
        # relative error beyond the per-element calibration — widen the floor
        floor = 2e-2 if name == "dsink" and layout == "sh" else 3e-3
        assert_close_to_ref(gv.cpu().float(), ghi.float(), glo.float(),
                            f"sink:{name}", floor=floor)


@requires_gpu
def test_max_logits():
    """return_max_logits (reference flex_flash_attn.py:397-408,
    forward_meta.py:28): per-head max of the scaled logits over all allowed
    (row, key) pairs."""
    from magi_attention.functional import flex_flash_attn_func

    tq = tk = 320
    hq, hk, d = 4, 2, 128
    q, k, v, dout, qr, kr, tm = make_flex_case(
        tq, tk, hq, hk, d, [[0, 192], [192, 320]], [[0, 256], [64, 320]],
        [1, 3], seed=5
    )
    out, meta = flex_flash_attn_func(q, k, v, qr, kr, tm,
                                     return_max_logits=True)
    torch.cuda.synchronize()
    assert meta.max_logits is not None and meta.max_logits.shape == (hq,)

    mask = make_attn_mask(tq, tk, [[0, 192], [192, 320]],
                          [[0, 256], [64, 320]], [1, 3])
    qc = q.detach().cpu().double().permute(1, 0, 2)
    kc = k.detach().cpu().double().repeat_interleave(hq // hk, dim=1)
    s = torch.matmul(qc, kc.permute(1, 0, 2).transpose(-1, -2)) * d ** -0.5
    s = torch.where(mask.unsqueeze(0), s, torch.full_like(s, float("-inf")))
    ref = s.amax(dim=(-1, -2)).float()
    torch.testing.assert_close(meta.max_logits.cpu(), ref, atol=2e-2,
                               rtol=2e-2)


@requires_gpu
def test_auto_range_merge():
    """auto_range_merge (reference flex_flash_attn.py:79 merge_ranges +
    mainloop qk_map): repeated q ranges iterate their k segments IN-KERNEL
    with the online softmax carried across, instead of per-pair launches with
    lock-merge. Must match both the oracle and the merge=False path."""
    from magi_attention.functional import flex_flash_attn_func
    from oracle import ref_attn_with_grads

    tq = tk = 1024
    hq, hk, d = 4, 2, 128
    # block-sparse: q [0,256) attends 3 scattered k blocks; q [256,640) two;
    # q [640,1024) causal tail; plus an OVERLAPPING extra q range
    qr_l = [[0, 256], [0, 256], [0, 256], [256, 640], [256, 640],
            [640, 1024], [128, 384]]
    kr_l = [[0, 128], [384, 512], [640, 768], [0, 256], [512, 896],
            [640, 1024], [896, 1024]]
    tt = [0, 0, 1, 0, 0, 1, 0]
    q, k, v, dout, qr, kr, tm = make_flex_case(
        tq, tk, hq, hk, d, qr_l, kr_l, tt, seed=13
    )

    def run(arm):
        qq = q.detach().clone().requires_grad_(True)
        kk = k.detach().clone().requires_grad_(True)
        vv = v.detach().clone().requires_grad_(True)
        out, meta = flex_flash_attn_func(qq, kk, vv, qr, kr, tm,
                                         auto_range_merge=arm)
        out.backward(dout)
        torch.cuda.synchronize()
        return out.detach(), meta.lse, qq.grad, kk.grad, vv.grad

    o_m, lse_m, dq_m, dk_m, dv_m = run(True)
    o_u, lse_u, dq_u, dk_u, dv_u = run(False)

    mask = make_attn_mask(tq, tk, qr_l, kr_l, tt)
    qc, kc, vc, doc = [t.detach().cpu() for t in (q, k, v, dout)]
    hi = ref_attn_with_grads(qc, kc, vc, mask, doc)
    lo = ref_attn_with_grads(qc, kc, vc, mask, doc, high_precision=False,
                             p_dtype=torch.bfloat16)
    for got, ghi, glo, name in [
        (o_m, hi[0], lo[0], "arm:out"), (lse_m.cpu(), hi[1], lo[1], "arm:lse"),
        (dq_m, hi[2], lo[2], "arm:dq"), (dk_m, hi[3], lo[3], "arm:dk"),
        (dv_m, hi[4], lo[4], "arm:dv"),
    ]:
        assert_close_to_ref(got.cpu().float(), ghi.float(), glo.float(), name)
    # merged and unmerged paths agree within bf16 reduction noise
    torch.testing.assert_close(o_m.float(), o_u.float(), atol=3e-2, rtol=3e-2)
    torch.testing.assert_close(lse_m, lse_u, atol=1e-4, rtol=1e-4)


@requires_gpu
@pytest.mark.parametrize("hd", [48, 96])
def test_odd_head_dims(hd):
    """Head dims other than 64/128 run zero-padded in the next bucket
    (reference flash_api.cpp:322-334 bucketing) — exact, incl. gradients."""
    from magi_attention.functional import flex_flash_attn_func
    from oracle import ref_attn_with_grads

    tq = tk = 256
    q, k, v, dout, qr, kr, tm = make_flex_case(
        tq, tk, 4, 2, hd, [[0, tq]], [[0, tk]], [1], seed=3
    )
    q.requires_grad_(True)
    k.requires_grad_(True)
    v.requires_grad_(True)
    out, meta = flex_flash_attn_func(q, k, v, qr, kr, tm)
    assert out.shape[-1] == hd
    out.backward(dout)
    torch.cuda.synchronize()
    mask = make_attn_mask(tq, tk, [[0, tq]], [[0, tk]], [1])
    qc, kc, vc, doc = [t.detach().cpu() for t in (q, k, v, dout)]
    hi = ref_attn_with_grads(qc, kc, vc, mask, doc)
    lo = ref_attn_with_grads(qc, kc, vc, mask, doc, high_precision=False,
                             p_dtype=torch.bfloat16)
    for got, ghi, glo, name in [
        (out.detach(), hi[0], lo[0], "out"), (q.grad, hi[2], lo[2], "dq"),
        (k.grad, hi[3], lo[3], "dk"), (v.grad, hi[4], lo[4], "dv"),
    ]:
        assert_close_to_ref(got.cpu().float(), ghi.float(), glo.float(),
                            f"hd{hd}:{name}")


@requires_gpu
def test_native_range_utils():
    """Native argsort/reorder/unique kernels vs the torch reference
    (reference extensions/sort_and_reorder_ranges.cu,
    unique_consecutive_pairs.cu) — the magi_attn_ext boundary ops."""
    import magi_attention.magi_attn_ext as ext

    g = torch.Generator().manual_seed(9)
    for n in (1, 7, 255, 1024, 4097):
        starts = torch.randint(0, 500, (n,), generator=g, dtype=torch.int32)
        ends = starts + torch.randint(1, 100, (n,), generator=g,
                                      dtype=torch.int32)
        ranges = torch.stack([starts, ends], 1).cuda()
        idx = ext.argsort_ranges(ranges)
        key = ranges[:, 0].long() * (1 << 31) + ranges[:, 1].long()
        ref = torch.argsort(key.cpu(), stable=True).to(torch.int32)
        assert torch.equal(idx.cpu(), ref), n

        qro, kro, tmo = ext.reorder_ranges_and_attn_type_maps(
            ranges, ranges + 1, torch.arange(n, dtype=torch.int32).cuda(),
            idx,
        )
        assert torch.equal(qro.cpu(), ranges.cpu()[ref.long()])
        assert torch.equal(tmo.cpu(), ref)

        srt = ranges[idx.long()]
        uniq, inv, cnt = ext.unique_consecutive_pairs(srt)
        ru, rinv = torch.unique_consecutive(srt.cpu(), dim=0,
                                            return_inverse=True)
        assert int(cnt.item()) == ru.shape[0], n
        assert torch.equal(uniq.cpu(), ru.to(torch.int32))
        assert torch.equal(inv.cpu(), rinv.to(torch.int32))


@requires_gpu
def test_head_dim_192_bucket():
    """D=192 bucket (MLA head dims; reference flash_api.cpp:322 buckets
    <=64/<=128/<=192, tile_size.h:42): exact D=192 plus a padded d=160 case,
    fwd+bwd vs the fp64 oracle."""
    from oracle import make_attn_mask, ref_attn_with_grads
    from magi_attention.functional import flex_flash_attn_func

    for d in (192, 160):
        tq = tk = 1024
        hq, hk = 4, 2
        g = torch.Generator().manual_seed(33 + d)
        q = (torch.randn(tq, hq, d, generator=g) * 0.5).bfloat16().cuda().requires_grad_(True)
        k = (torch.randn(tk, hk, d, generator=g) * 0.5).bfloat16().cuda().requires_grad_(True)
        v = (torch.randn(tk, hk, d, generator=g) * 0.5).bfloat16().cuda().requires_grad_(True)
        dout = (torch.randn(tq, hq, d, generator=g) * 0.5).bfloat16().cuda()
        qr_l = [[0, 512], [512, 1024]]
        ty_l = [1, 0]
        qr = torch.tensor(qr_l, dtype=torch.int32, device="cuda")
        tm = torch.tensor(ty_l, dtype=torch.int32, device="cuda")
        out, meta = flex_flash_attn_func(q, k, v, qr, qr.clone(), tm)
        out.backward(dout)
        torch.cuda.synchronize()

        mask = make_attn_mask(tq, tk, qr_l, qr_l, ty_l)
        qc, kc, vc, doc = [t.detach().cpu() for t in (q, k, v, dout)]
        o_hi, _, dq_hi, dk_hi, dv_hi = ref_attn_with_grads(qc, kc, vc, mask, doc)
        o_lo, _, dq_lo, dk_lo, dv_lo = ref_attn_with_grads(
            qc, kc, vc, mask, doc, high_precision=False,
            p_dtype=torch.bfloat16)
        assert_close_to_ref(out.detach().cpu().float(), o_hi.float(),
                            o_lo.float(), f"d{d}:out")
        assert_close_to_ref(q.grad.cpu().float(), dq_hi.float(),
                            dq_lo.float(), f"d{d}:dq", ratio=4.5)
        assert_close_to_ref(k.grad.cpu().float(), dk_hi.float(),
                            dk_lo.float(), f"d{d}:dk", ratio=4.5)
        assert_close_to_ref(v.grad.cpu().float(), dv_hi.float(),
                            dv_lo.float(), f"d{d}:dv", ratio=4.5)


@requires_gpu
def test_fwd_staging_ring_equivalence():
    """The 3-slot constant-distance staging ring must be bitwise-identical to
    the 2-slot full-drain ring (same MFMA order; only the barrier/prefetch
    schedule differs). Regression for the r2 fwd pipeline port."""
    import os

    from magi_attention.functional import flex_flash_attn_func

    torch.manual_seed(5)
    n, hq, d = 2048 + 192, 4, 128
    q = (torch.randn(n, hq, d) * 0.5).bfloat16().cuda()
    k = (torch.randn(n, hq, d) * 0.5).bfloat16().cuda()
    v = (torch.randn(n, hq, d) * 0.5).bfloat16().cuda()
    qr = torch.tensor([[0, n]], dtype=torch.int32, device="cuda")
    tm = torch.tensor([1], dtype=torch.int32, device="cuda")

    outs = {}
    try:
        for nbuf in ("2", "3"):
            os.environ["MAGI_FWD_NBUF"] = nbuf
            with torch.no_grad():
                o, meta = flex_flash_attn_func(q, k, v, qr, qr.clone(), tm,
                                               max_seqlen_q=n)
            torch.cuda.synchronize()
            outs[nbuf] = (o, meta.lse)
    finally:
        os.environ.pop("MAGI_FWD_NBUF", None)
    assert torch.equal(outs["2"][0], outs["3"][0])
    assert torch.equal(outs["2"][1], outs["3"][1])


@requires_gpu
def test_bwd_fused_variant_equivalence():
    """The gated fused-dkv cadence variants (MODE3 fat-wave, W6 64-row ring)
    must produce bit-identical dk/dv to the default v2 kernel (documented
    measurement points; profiles/r2_final_pmc.md)."""
    import os

    from magi_attention.functional import flex_flash_attn_func

    torch.manual_seed(6)
    n, hq, d = 2048 + 320, 2, 128
    qr = torch.tensor([[0, n]], dtype=torch.int32, device="cuda")
    tm = torch.tensor([1], dtype=torch.int32, device="cuda")

    def grads(env):
        for kk, vv in env.items():
            os.environ[kk] = vv
        try:
            torch.manual_seed(6)
            q = (torch.randn(n, hq, d) * 0.5).bfloat16().cuda().requires_grad_(True)
            k = (torch.randn(n, hq, d) * 0.5).bfloat16().cuda().requires_grad_(True)
            v = (torch.randn(n, hq, d) * 0.5).bfloat16().cuda().requires_grad_(True)
            o, _ = flex_flash_attn_func(q, k, v, qr, qr.clone(), tm,
                                        max_seqlen_q=n)
            torch.manual_seed(7)
            o.backward(torch.randn_like(o))
            torch.cuda.synchronize()
            return q.grad.clone(), k.grad.clone(), v.grad.clone()
        finally:
            for kk in env:
                os.environ.pop(kk, None)

    g0 = grads({})
    for tag, env in (("fat", {"MAGI_BWD_FAT": "1"}),
                     ("w6", {"MAGI_BWD_W6": "1"})):
        g = grads(env)
        for name, x, y in zip("dq dk dv".split(), g, g0):
            assert torch.equal(x, y), (tag, name)
