"""fp8 (e4m3) forward extension parity (MI355X-native; the reference has no
fp8 compute path — SURVEY.md §8c caveat). Budgets are calibrated by a low
oracle that quantises inputs AND P to fp8."""
import pytest
import torch

pytestmark = pytest.mark.gpu

from oracle import make_attn_mask, ref_attn  # noqa: E402
from tests.util import assert_close_to_ref  # noqa: E402

requires_gpu = pytest.mark.skipif(
    not torch.cuda.is_available(), reason="needs MI355X"
)


@requires_gpu
@pytest.mark.parametrize(
    "case",
    [
        ("full_256", 256, 256, 2, 2, 128, [[0, 256]], [[0, 256]], [0]),
        ("causal_512", 512, 512, 4, 4, 128, [[0, 512]], [[0, 512]], [1]),
        ("gqa_varlen", 1024, 1024, 8, 1, 128,
         [[0, 384], [384, 1024]], [[0, 384], [384, 1024]], [1, 1]),
        ("d64_causal", 256, 256, 2, 2, 64, [[0, 256]], [[0, 256]], [1]),
    ],
    ids=lambda c: c[0] if isinstance(c, tuple) else str(c),
)
def test_fp8_fwd_parity(case):
    from magi_attention.functional import flex_flash_attn_func

    name, tq, tk, hq, hk, d, qr_l, kr_l, ty_l = case
    g = torch.Generator().manual_seed(11)
    qf = (torch.randn(tq, hq, d, generator=g) * 0.5)
    kf = (torch.randn(tk, hk, d, generator=g) * 0.5)
    vf = (torch.randn(tk, hk, d, generator=g) * 0.5)
    q8 = qf.to(torch.float8_e4m3fn).cuda()
    k8 = kf.to(torch.float8_e4m3fn).cuda()
    v8 = vf.to(torch.float8_e4m3fn).cuda()
    qr = torch.tensor(qr_l, dtype=torch.int32, device="cuda")
    kr = torch.tensor(kr_l, dtype=torch.int32, device="cuda")
    tm = torch.tensor(ty_l, dtype=torch.int32, device="cuda")
    out, meta = flex_flash_attn_func(q8, k8, v8, qr, kr, tm)
    torch.cuda.synchronize()
    assert out.dtype == torch.bfloat16

    # oracle on the SAME fp8-quantised inputs
    mask = make_attn_mask(tq, tk, qr_l, kr_l, ty_l)
    qq = q8.cpu().float()
    kk = k8.cpu().float()
    vv = v8.cpu().float()
    o_hi, lse_hi = ref_attn(qq, kk, vv, mask, high_precision=True)
    o_lo, _ = ref_attn(qq, kk, vv, mask, high_precision=False,
                       p_dtype=torch.float8_e4m3fn)
    assert_close_to_ref(out.cpu().float(), o_hi.float(), o_lo.float(),
                        f"fp8:{name}:out", floor=1e-2)
    lse = meta.lse.cpu()
    fin = torch.isfinite(lse_hi)
    assert torch.equal(torch.isfinite(lse), fin)
    torch.testing.assert_close(lse[fin], lse_hi[fin], atol=3e-2, rtol=3e-3)


@requires_gpu
def test_fp8_fwd_bwd_autograd():
    """fp8 autograd: fwd on fp8 MFMAs, bwd on bf16 kernels over upcast
    operands (documented mixed-precision policy)."""
    from magi_attention.functional import flex_flash_attn_func

    tq = tk = 256
    hq = hk = 2
    d = 128
    g = torch.Generator().manual_seed(12)
    q = (torch.randn(tq, hq, d, generator=g) * 0.5).to(torch.float8_e4m3fn).cuda().requires_grad_(True)
    k = (torch.randn(tk, hk, d, generator=g) * 0.5).to(torch.float8_e4m3fn).cuda().requires_grad_(True)
    v = (torch.randn(tk, hk, d, generator=g) * 0.5).to(torch.float8_e4m3fn).cuda().requires_grad_(True)
    qr = torch.tensor([[0, tq]], dtype=torch.int32, device="cuda")
    kr = torch.tensor([[0, tk]], dtype=torch.int32, device="cuda")
    tm = torch.tensor([1], dtype=torch.int32, device="cuda")
    out, meta = flex_flash_attn_func(q, k, v, qr, kr, tm)
    out.sum().backward()
    torch.cuda.synchronize()
    assert q.grad is not None and q.grad.dtype == torch.float8_e4m3fn
    assert torch.isfinite(q.grad.float()).all()
