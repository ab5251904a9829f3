"""Calibrated-mismatch comparison helpers (reference surface:
testing/precision.py — assert_close with a mismatch-ratio budget,
extract_mismatch_threshold for calibrating that budget from a low-precision
reference, calc_inf_norm). The same procedure DESIGN.md §5b describes for
the kernel parity suites."""
from __future__ import annotations

import re

import torch
import torch.distributed as dist

from ..functional.utils import safe_subtract
from ..utils import max_fp_dtype

EPSILON = 1e-8

MISMATCH_THRES_RATIO: float = 2.0
MAX_MISMATCH_THRES: float = 0.75
NORM_RTOL_RATIO: float = 2.0

# published single-end bandwidth / rate constants the reference's perf
# assertions use, plus this target's own numbers
IB_BANDWIDTH = 50e9
H100_TFLOPS_16 = 989.5e12
H100_MATMUL_MFU = 0.7
H100_NVLINK_BANDWIDTH = 450e9
H100_NVLINK_A2A_BWU = 0.6
H800_TFLOPS_16 = 989.5e12
H800_NVLINK_BANDWIDTH = 200e9
H800_NVLINK_A2A_BWU = 0.6
MI355X_TFLOPS_16 = 2500e12  # dense bf16 MFMA peak (no sparsity)
MI355X_XGMI_BANDWIDTH = 153e9  # per link, 7 links per GPU


def extract_mismatch_info(error_msg: str) -> tuple[int, int, float]:
    m = re.search(r"Mismatched elements: (\d+) / (\d+)", error_msg)
    if not m:
        raise ValueError(f"Could not find mismatch elements in {error_msg=}")
    mismatched, total = int(m.group(1)), int(m.group(2))
    return mismatched, total, mismatched / total


@torch.no_grad
def extract_mismatch_threshold(
    actual: torch.Tensor,
    expected: torch.Tensor,
    atol: float,
    rtol: float,
    mismatch_thres_ratio: float = 1.0,
    min_mismatch_thres: float = 0.0,
    max_mismatch_thres: float = 1.0,
) -> float:
    """Measure the mismatch ratio a LOW-PRECISION reference itself produces
    against the ground truth, scale it, clamp it — the budget the kernel
    under test is then allowed."""
    ratio = 0.0
    try:
        torch.testing.assert_close(actual, expected, atol=atol, rtol=rtol)
    except AssertionError as e:
        _, _, ratio = extract_mismatch_info(str(e))
    return min(max(ratio * mismatch_thres_ratio, min_mismatch_thres),
               max_mismatch_thres)


@torch.no_grad
def assert_close(
    a: torch.Tensor,
    b: torch.Tensor,
    atol: float = 1e-5,
    rtol: float = 1e-5,
    mismatch_threshold: float = 0,
    test_case: str = "",
    print_rank: int = 0,
) -> None:
    """torch.testing.assert_close with an allowed mismatch RATIO: the check
    passes when at most mismatch_threshold of elements fall outside
    (atol, rtol)."""
    assert 0 <= mismatch_threshold <= 1, (
        f"{mismatch_threshold=} must be between 0 and 1"
    )
    if dist.is_initialized():
        printing = print_rank == -1 or dist.get_rank() == print_rank
    else:
        printing = True

    try:
        torch.testing.assert_close(a, b, atol=atol, rtol=rtol)
        if printing:
            print(f"[{test_case}]: has no mismatch")
    except AssertionError as e:
        error_msg = str(e)
        mismatched, total, ratio = extract_mismatch_info(error_msg)
        info = (
            f"[{test_case}]: mismatch_ratio = {mismatched} / {total} "
            f"= {ratio * 100:.4f} % | "
            f"mismatch_threshold={mismatch_threshold * 100:.2f} %"
        )
        if ratio <= mismatch_threshold:
            if printing:
                print(info)
            return
        raise type(e)(
            f"\n>>>>>>>  Torch Error Message: \n\n{error_msg}\n\n"
            f">>>>>>>  Mismatch Detailed Info: \n\n{info}\n\n"
        ) from e


@torch.no_grad
def calc_inf_norm(a: torch.Tensor, b: torch.Tensor) -> float:
    dtype = max_fp_dtype(a.dtype, b.dtype, torch.float32)
    return safe_subtract(a.to(dtype), b.to(dtype)).norm(p=float("inf")).item()
