"""Targeted GPU debug for the fwd kernel (run via gpurun; not a pytest)."""
import ctypes
import sys

import torch

sys.path.insert(0, ".")
from magi_attention import _ffa_lib  # noqa: E402
from magi_attention.functional import flex_flash_attn_func  # noqa: E402
from oracle import make_attn_mask, ref_attn  # noqa: E402

lib = _ffa_lib.lib()


def probe_lane():
    inp = torch.arange(64, dtype=torch.int32).cuda()
    out = torch.zeros(192, dtype=torch.int32).cuda()
    rc = lib.magi_probe_lane(
        ctypes.c_void_p(inp.data_ptr()), ctypes.c_void_p(out.data_ptr()),
        _ffa_lib.current_stream_ptr(),
    )
    torch.cuda.synchronize()
    o = out.cpu().tolist()
    r0, r1, bp = o[:64], o[64:128], o[128:]
    # expected per my model: r0[l<32]=a[l]=l ; r0[l>=32]=b[l-32]=l-32+1000
    exp_r0 = [l if l < 32 else (l - 32 + 1000) for l in range(64)]
    exp_r1 = [(l + 32) if l < 32 else (l + 1000) for l in range(64)]
    exp_bp = [l & 31 for l in range(64)]
    print("permlane r0 ok:", r0 == exp_r0)
    print("permlane r1 ok:", r1 == exp_r1)
    print("bpermute ok:", bp == exp_bp)
    if r0 != exp_r0:
        print("  r0:", r0)
    if r1 != exp_r1:
        print("  r1:", r1)
    if bp != exp_bp:
        print("  bp:", bp)


def run_case(name, tq, tk, hq, hk, d, direct=False):
    g = torch.Generator().manual_seed(7)
    q = (torch.randn(tq, hq, d, generator=g) * 0.5).bfloat16().cuda()
    k = (torch.randn(tk, hk, d, generator=g) * 0.5).bfloat16().cuda()
    v = (torch.randn(tk, hk, d, generator=g) * 0.5).bfloat16().cuda()
    qr = torch.tensor([[0, tq]], dtype=torch.int32, device="cuda")
    kr = torch.tensor([[0, tk]], dtype=torch.int32, device="cuda")
    out, meta = flex_flash_attn_func(
        q, k, v, qr, kr, None, disable_fwd_atomic_reduction=direct
    )
    torch.cuda.synchronize()
    mask = make_attn_mask(tq, tk, [[0, tq]], [[0, tk]], [0])
    ref_o, ref_lse = ref_attn(q.cpu(), k.cpu(), v.cpu(), mask)
    err = (out.cpu().float() - ref_o.float()).abs()
    rel = err.norm() / ref_o.float().norm()
    lse_err = (meta.lse.cpu() - ref_lse).abs().max()
    print(f"{name}: relL2={rel:.3e} maxabs={err.max():.3e} lse_err={lse_err:.3e}")
    if rel > 1e-2:
        # error map: which (q, h) rows are bad
        bad = err.amax(dim=2)  # [tq, hq]
        rows = (bad > 5 * bad.median()).nonzero()
        print("  worst rows (q,h):", rows[:20].tolist())
        qrow = int(err.amax(dim=(1, 2)).argmax())
        print(f"  worst q row {qrow}: err by d:",
              [round(float(x), 4) for x in err[qrow, 0, ::8]])
        print("  lse row:", float(meta.lse.cpu()[qrow, 0]),
              "ref", float(ref_lse[qrow, 0]))


probe_lane()
run_case("t32_single_tile", 32, 32, 1, 1, 64)
run_case("t32_two_tiles", 32, 64, 1, 1, 64)
run_case("t32_k33_unaligned", 32, 33, 1, 1, 64)
run_case("t64_two_waves", 64, 64, 2, 2, 64)
run_case("t64_direct", 64, 64, 2, 2, 64, direct=True)
run_case("t128_d128", 128, 128, 2, 2, 128)
run_case("t256", 256, 256, 2, 2, 64)
