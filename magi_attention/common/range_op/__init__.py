# Range row ops — host wrappers over the HIP kernels in csrc/range_ops.hip.
# Reference surface: magi_attention/common/range_op/_range_gather.py:126,
# _range_reduce.py:360, _range_fill.py:65 (Triton there; hand-written HIP here).
from __future__ import annotations

from typing import Optional

import torch

from ... import _ffa_lib as _L  # noqa: TID252


def _row_elems(t: torch.Tensor) -> int:
    n = 1
    for s in t.shape[1:]:
        n *= s
    return n


def _elem_size(t: torch.Tensor) -> int:
    return t.element_size()


def _default_starts(ranges: torch.Tensor):
    sizes = (ranges[:, 1] - ranges[:, 0]).to(torch.int64)
    cu = torch.cumsum(sizes, 0) - sizes
    return cu.to(torch.int32), int(sizes.sum().item())


def range_gather(
    input: torch.Tensor,
    ranges: torch.Tensor,
    out_starts: Optional[torch.Tensor] = None,
    total_size: Optional[int] = None,
    output: Optional[torch.Tensor] = None,
    dim: int = 0,
) -> torch.Tensor:
    """output[out_starts[i] + j] = input[ranges[i,0] + j]. Packs rows of
    `input` selected by `ranges` into a dense output (dim=0 only)."""
    assert dim == 0
    input = input.contiguous()
    if out_starts is None or total_size is None:
        out_starts, total_size = _default_starts(ranges)
    if output is None:
        output = torch.empty(
            (total_size, *input.shape[1:]), dtype=input.dtype, device=input.device
        )
    if ranges.numel() == 0:
        return output
    args = _L.MagiRangeOpArgs(
        input=_L.ptr(input), output=_L.ptr(output),
        in_ranges=_L.ptr(ranges.contiguous()),
        out_starts=_L.ptr(out_starts.contiguous()),
        in_lse=_L.ptr(None), out_lse=_L.ptr(None),
        n_ranges=ranges.shape[0], row_elems=_row_elems(input),
        total_rows=total_size, elem_size=_elem_size(input),
        n_heads=0, reduce_op=0, stream=_L.current_stream_ptr(),
    )
    _L.check(_L.lib().magi_range_gather(args), "magi_range_gather")
    return output


def range_scatter(
    input: torch.Tensor,
    output: torch.Tensor,
    ranges: torch.Tensor,
    in_starts: torch.Tensor,
) -> torch.Tensor:
    """output[ranges[i,0] + j] = input[in_starts[i] + j] (inverse of gather).
    Implemented with the same kernel by swapping the index roles: build
    in_ranges over the packed input and out_starts over the output."""
    sizes = ranges[:, 1] - ranges[:, 0]
    in_ranges = torch.stack([in_starts, in_starts + sizes], dim=1).to(torch.int32)
    args = _L.MagiRangeOpArgs(
        input=_L.ptr(input.contiguous()), output=_L.ptr(output),
        in_ranges=_L.ptr(in_ranges.contiguous()),
        out_starts=_L.ptr(ranges[:, 0].contiguous()),
        in_lse=_L.ptr(None), out_lse=_L.ptr(None),
        n_ranges=ranges.shape[0], row_elems=_row_elems(input),
        total_rows=int(sizes.sum().item()) if ranges.numel() else 0,
        elem_size=_elem_size(input),
        n_heads=0, reduce_op=0, stream=_L.current_stream_ptr(),
    )
    _L.check(_L.lib().magi_range_gather(args), "magi_range_scatter")
    return output


def range_reduce(
    input: torch.Tensor,
    output: torch.Tensor,
    in_ranges: torch.Tensor,
    out_starts: torch.Tensor,
    op: str = "sum",
    in_lse: Optional[torch.Tensor] = None,
    out_lse: Optional[torch.Tensor] = None,
    total_rows: Optional[int] = None,
) -> torch.Tensor:
    """output rows [out_starts[i] ...] (+)= input rows [in_ranges[i] ...].
    op: "sum" (f32 add) or "lse" (online-softmax weighted merge of
    (out,lse) rows; reference _range_reduce.py:239)."""
    if in_ranges.numel() == 0:
        return output
    if total_rows is None:
        total_rows = int((in_ranges[:, 1] - in_ranges[:, 0]).sum().item())
    opc = {"sum": 1, "lse": 2}[op]
    if opc == 1:
        assert input.dtype == torch.float32 and output.dtype == torch.float32
    n_heads = input.shape[1] if input.dim() == 3 else 0
    args = _L.MagiRangeOpArgs(
        input=_L.ptr(input.contiguous()), output=_L.ptr(output),
        in_ranges=_L.ptr(in_ranges.contiguous()),
        out_starts=_L.ptr(out_starts.contiguous()),
        in_lse=_L.ptr(in_lse), out_lse=_L.ptr(out_lse),
        n_ranges=in_ranges.shape[0], row_elems=_row_elems(input),
        total_rows=total_rows, elem_size=_elem_size(input),
        n_heads=n_heads, reduce_op=opc, stream=_L.current_stream_ptr(),
    )
    _L.check(_L.lib().magi_range_reduce(args), "magi_range_reduce")
    return output


def correct_out_lse(
    out1: torch.Tensor,
    lse1: torch.Tensor,
    out2: torch.Tensor,
    lse2: torch.Tensor,
) -> tuple[torch.Tensor, torch.Tensor]:
    """In-place fused merge of partial (out2,lse2) into (out1,lse1)
    (reference functional/utils.py:371 correct_out_lse_kernel)."""
    t, h, d = out1.shape
    assert out1.dtype == torch.float32 and out2.dtype == torch.float32
    args = _L.MagiCorrectArgs(
        out1=_L.ptr(out1), lse1=_L.ptr(lse1), out2=_L.ptr(out2.contiguous()),
        lse2=_L.ptr(lse2.contiguous()), total_rows=t, n_heads=h, d=d,
        stream=_L.current_stream_ptr(),
    )
    _L.check(_L.lib().magi_correct_out_lse(args), "magi_correct_out_lse")
    return out1, lse1
