"""Profiling-range annotations (reference surface: utils/nvtx.py —
add_nvtx_event, instrument_nvtx, switch_profile, nvtx_range_push/pop).

On ROCm, ``torch.cuda.nvtx`` maps to rocTX markers, so these ranges show up
in rocprofv3's marker trace domain. NOTE (MI355X pool rule): collect marker
traces in their own rocprofv3 run — never combined with ``--pmc``.

This rebuild does not use a tracing compiler, so the reference's
torch.library custom-op registration (there only to survive torch.compile
graph capture) is not replicated; the functions are plain calls.
"""
from __future__ import annotations

from functools import wraps
from typing import Any, Callable, TypeVar, cast

import torch

F = TypeVar("F", bound=Callable[..., Any])

_EMIT_NVTX_CTX = None


def nvtx_range_push(event_name: str) -> None:
    torch.cuda.nvtx.range_push(event_name)


def nvtx_range_pop() -> None:
    torch.cuda.nvtx.range_pop()


class add_nvtx_event:
    """Context manager wrapping a code block in a named rocTX/NVTX range."""

    def __init__(self, event_name: str):
        self.enter_name = event_name

    def __enter__(self):
        torch.cuda.nvtx.range_push(self.enter_name)
        return self

    def __exit__(self, *excinfo):
        torch.cuda.nvtx.range_pop()


def instrument_nvtx(func: F) -> F:
    """Decorator recording a range for the duration of each call."""

    @wraps(func)
    def wrapped_fn(*args, **kwargs):
        with add_nvtx_event(func.__qualname__):
            return func(*args, **kwargs)

    return cast(F, wrapped_fn)


def switch_profile(
    iter_id: int,
    start: int,
    end: int,
    profile_ranks: list[int],
    event_name: str | None = None,
    record_shape: bool = True,
) -> None:
    """Turn the profiler on at iteration `start` and off at `end` on the
    given ranks, with one named range per iteration in between (reference
    utils/nvtx.py:110 switch_profile)."""
    if not torch.distributed.is_initialized():
        assert profile_ranks == [0], (
            "profile_ranks can only contain rank0 if torch.distributed "
            "is not initialized"
        )
    elif torch.distributed.get_rank() not in profile_ranks:
        return

    global _EMIT_NVTX_CTX
    if event_name is None:
        event_name = f"iter{iter_id}"

    if iter_id == start:
        if record_shape:
            ctx = torch.autograd.profiler.emit_nvtx(record_shapes=True)
            ctx.__enter__()
            _EMIT_NVTX_CTX = ctx
        torch.cuda.cudart().cudaProfilerStart()  # hipProfilerStart on ROCm
        torch.cuda.nvtx.range_push(event_name)
    elif iter_id == end:
        torch.cuda.nvtx.range_pop()
        torch.cuda.cudart().cudaProfilerStop()
        if record_shape and _EMIT_NVTX_CTX is not None:
            _EMIT_NVTX_CTX.__exit__(None, None, None)
            _EMIT_NVTX_CTX = None
    elif start < iter_id < end:
        torch.cuda.nvtx.range_pop()
        torch.cuda.nvtx.range_push(event_name)
