# magi_attn_ext — native extension surface
# (reference csrc/extensions/magi_attn_ext.cpp:125-621 pybind module; the
# reference package imports it at init, magi_attention/__init__.py:27-43).
#
# MI355X rebuild: KernelBarrier and events are backed by the HIP C-ABI library
# (csrc/ext_utils.hip via ctypes); the small range-sorting helpers
# (argsort_ranges / reorder / unique_consecutive_pairs, used by merge_ranges
# on the auto_range_merge path, flex_flash_attn.py:150-167) run as torch GPU
# ops over [N,2] int32 tensors — N is at most a few 10^4, so these are
# microsecond-scale; dedicated HIP kernels are a later-round item.
from __future__ import annotations

import time
from typing import Dict, List, Tuple

import torch

from . import _ffa_lib
from .common.enum import AttnMaskType  # noqa: F401  (C++ type alias)
from .common.range import AttnRange, RangeError  # noqa: F401  (C++ ranges alias)
from .common.ranges import (  # noqa: F401
    AttnRanges,
    check_valid_cu_seqlens,
    is_valid_cu_seqlens,
)
from .common.rectangle import AttnRectangle  # noqa: F401
from .common.rectangles import AttnRectangles  # noqa: F401


class KernelBarrier:
    """GPU spin barrier ordering comm kernels after compute without host sync
    (reference extensions/kernel_barrier.cu:103; used by dist_attn.py:3054)."""

    def __init__(self, device=None):
        self._counter = torch.zeros(1, dtype=torch.int32,
                                    device=device or "cuda")
        self._target = 0

    def produce(self, stream=None) -> None:
        self._target += 1
        ptr = _ffa_lib.ptr(self._counter)
        s = (_ffa_lib.ctypes.c_void_p(stream)
             if stream is not None else _ffa_lib.current_stream_ptr())
        _ffa_lib.check(
            _ffa_lib.lib().magi_kernel_barrier_produce(ptr, s), "barrier produce"
        )

    def synchronize(self, stream=None) -> None:
        ptr = _ffa_lib.ptr(self._counter)
        s = (_ffa_lib.ctypes.c_void_p(stream)
             if stream is not None else _ffa_lib.current_stream_ptr())
        _ffa_lib.check(
            _ffa_lib.lib().magi_kernel_barrier_synchronize(
                ptr, self._target, s),
            "barrier synchronize",
        )

    def reset(self) -> None:
        self._counter.zero_()
        self._target = 0

    def get_value(self) -> int:
        return int(self._counter.item())


_SORT_CAP = 8192  # LDS capacity of the native single-WG sort


def argsort_ranges(ranges: torch.Tensor) -> torch.Tensor:
    """Indices sorting [N,2] int32 ranges lexicographically by (start, end)
    (reference extensions/sort_and_reorder_ranges.cu). Native HIP bitonic
    sort (csrc/ext_utils.hip magi_argsort_ranges) for CUDA tensors with
    N <= 8192 (the planner regime); torch fallback above / on CPU."""
    n = ranges.shape[0]
    if ranges.is_cuda and 0 < n <= _SORT_CAP:
        out = torch.empty(n, dtype=torch.int32, device=ranges.device)
        rc = _ffa_lib.lib().magi_argsort_ranges(
            _ffa_lib.ptr(ranges.contiguous()), _ffa_lib.ptr(out), n,
            _ffa_lib.current_stream_ptr())
        if rc == 0:
            return out
    key = ranges[:, 0].long() * (1 << 31) + ranges[:, 1].long()
    return torch.argsort(key, stable=True).to(torch.int32)


def reorder_ranges_and_attn_type_maps(
    q_ranges: torch.Tensor,
    k_ranges: torch.Tensor,
    attn_type_map: torch.Tensor,
    order: torch.Tensor,
) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    n = order.shape[0]
    if q_ranges.is_cuda and n > 0:
        qro = torch.empty_like(q_ranges)
        kro = torch.empty_like(k_ranges)
        tmo = torch.empty_like(attn_type_map)
        rc = _ffa_lib.lib().magi_reorder_ranges(
            _ffa_lib.ptr(q_ranges.contiguous()),
            _ffa_lib.ptr(k_ranges.contiguous()),
            _ffa_lib.ptr(attn_type_map.contiguous()),
            _ffa_lib.ptr(order.contiguous()),
            _ffa_lib.ptr(qro), _ffa_lib.ptr(kro), _ffa_lib.ptr(tmo), n,
            _ffa_lib.current_stream_ptr())
        if rc == 0:
            return qro, kro, tmo
    idx = order.long()
    return (
        q_ranges.index_select(0, idx),
        k_ranges.index_select(0, idx),
        attn_type_map.index_select(0, idx),
    )


def unique_consecutive_pairs(
    ranges: torch.Tensor,
) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    """Unique consecutive (start,end) pairs of sorted ranges + inverse map +
    count (reference extensions/unique_consecutive_pairs.cu; feeds
    merge_ranges for auto_range_merge). Native HIP single-WG scan for CUDA
    tensors with N <= 8192; torch fallback above / on CPU."""
    n = ranges.shape[0]
    if ranges.is_cuda and 0 < n <= _SORT_CAP:
        uniq = torch.empty_like(ranges)
        inverse = torch.empty(n, dtype=torch.int32, device=ranges.device)
        count = torch.zeros(1, dtype=torch.int32, device=ranges.device)
        rc = _ffa_lib.lib().magi_unique_pairs(
            _ffa_lib.ptr(ranges.contiguous()), _ffa_lib.ptr(uniq),
            _ffa_lib.ptr(inverse), _ffa_lib.ptr(count), n,
            _ffa_lib.current_stream_ptr())
        if rc == 0:
            cnt = int(count.item())
            return uniq[:cnt], inverse, count
    uniq, inverse = torch.unique_consecutive(
        ranges, dim=0, return_inverse=True
    )
    count = torch.tensor([uniq.shape[0]], dtype=torch.int32,
                         device=ranges.device)
    return uniq.to(torch.int32), inverse.to(torch.int32), count


# ---- named event timing (reference extensions/profile_utils.cu) ----
_events: Dict[str, List] = {}


def start_event(name: str) -> None:
    if torch.cuda.is_available():
        ev = torch.cuda.Event(enable_timing=True)
        ev.record()
        _events[name] = [ev, None]
    else:
        _events[name] = [time.perf_counter(), None]


def stop_event(name: str) -> None:
    if name not in _events:
        return
    if torch.cuda.is_available():
        ev = torch.cuda.Event(enable_timing=True)
        ev.record()
        _events[name][1] = ev
    else:
        _events[name][1] = time.perf_counter()


def elapsed_ms(name: str) -> float:
    a, b = _events.get(name, (None, None))
    if a is None or b is None:
        return 0.0
    if isinstance(a, float):
        return (b - a) * 1e3
    torch.cuda.synchronize()
    return a.elapsed_time(b)


def elapsed_ms_event(name: str) -> float:
    """Reference magi_attn_ext.elapsed_ms_event: ms between a named event's
    start/stop records."""
    return elapsed_ms(name)


def produce(kernel_barrier: "KernelBarrier | None") -> None:
    """Reference magi_attn_ext.produce: launch a producer kernel that
    increments the barrier count by 1 (no-op for None)."""
    if kernel_barrier is not None:
        kernel_barrier.produce()


def expand_attn_ranges(ranges: AttnRanges, stride: int, num_heads_group: int) -> AttnRanges:
    """Reference magi_attn_ext.expand_attn_ranges (the DynamicAttnSolver's
    range expansion): each range [s, e) expands to the num_heads_group
    per-head-group row blocks [h*stride + s, h*stride + e)."""
    out = AttnRanges()
    for h in range(int(num_heads_group)):
        off = h * int(stride)
        for r in ranges:
            out.append(AttnRange(r.start + off, r.end + off))
    return out


def destroy_event(name: str) -> None:
    _events.pop(name, None)
