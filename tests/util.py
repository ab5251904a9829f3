"""Shared test helpers: tolerance scheme following the reference's calibrated
mismatch procedure (magi_attention/testing/precision.py:36-152): the kernel is
allowed the error the LOW-precision oracle itself makes against the fp64
oracle, times a safety ratio, plus a small floor."""
from __future__ import annotations

import torch

MISMATCH_RATIO = 3.0
FLOOR = 3e-3


def assert_close_to_ref(
    actual: torch.Tensor,
    ref_hi: torch.Tensor,
    ref_lo: torch.Tensor,
    what: str = "tensor",
    ratio: float = MISMATCH_RATIO,
    floor: float = FLOOR,
):
    """actual vs fp64-oracle, budgeted by the bf16-oracle's own error."""
    actual = actual.double()
    ref_hi = ref_hi.double()
    ref_lo = ref_lo.double()
    denom = ref_hi.norm().clamp_min(1e-8)
    err_kernel = (actual - ref_hi).norm() / denom
    err_lo = (ref_lo - ref_hi).norm() / denom
    budget = max(ratio * err_lo.item(), floor)
    print(f"    [{what}] err={err_kernel.item():.3e} budget={budget:.3e}")
    assert err_kernel.item() <= budget, (
        f"{what}: rel-L2 error {err_kernel.item():.3e} exceeds budget "
        f"{budget:.3e} (low-precision oracle error {err_lo.item():.3e})"
    )


def make_flex_case(
    tq, tk, hq, hk, d, q_ranges, k_ranges, types, seed=42, device="cuda",
    dtype=torch.bfloat16,
):
    g = torch.Generator(device="cpu").manual_seed(seed)
    q = (torch.randn(tq, hq, d, generator=g) * 0.5).to(dtype).to(device)
    k = (torch.randn(tk, hk, d, generator=g) * 0.5).to(dtype).to(device)
    v = (torch.randn(tk, hk, d, generator=g) * 0.5).to(dtype).to(device)
    dout = (torch.randn(tq, hq, d, generator=g) * 0.5).to(dtype).to(device)
    qr = torch.tensor(q_ranges, dtype=torch.int32, device=device).reshape(-1, 2)
    kr = torch.tensor(k_ranges, dtype=torch.int32, device=device).reshape(-1, 2)
    tm = torch.tensor(types, dtype=torch.int32, device=device)
    return q, k, v, dout, qr, kr, tm
