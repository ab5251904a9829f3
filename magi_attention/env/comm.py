"""Reference-named view of the comm env flags (reference surface:
env/comm.py)."""
from __future__ import annotations

from . import (  # noqa: F401
    _get,
    ffa_backward_sm_margin,
    ffa_forward_sm_margin,
    is_backward_high_precision_reduce_enable,
    is_forward_high_precision_reduce_enable,
    is_hierarchical_comm_enable,
    is_native_grpcoll_enable,
    is_qo_comm_enable,
)


def ffa_fwd_sm_margin_save_for_comm() -> int:
    return ffa_forward_sm_margin()


def ffa_bwd_sm_margin_save_for_comm() -> int:
    return ffa_backward_sm_margin()


def is_fwd_high_precision_reduce_enable() -> bool:
    return is_forward_high_precision_reduce_enable()


def is_bwd_high_precision_reduce_enable() -> bool:
    return is_backward_high_precision_reduce_enable()


def dsink_all_reduce_op() -> str:
    """MAGI_ATTENTION_DSINK_ALL_REDUCE_OP: 'sum' (default) or 'avg' for the
    cross-rank dsink all-reduce."""
    op = _get("MAGI_ATTENTION_DSINK_ALL_REDUCE_OP", "sum")
    assert op in ("sum", "avg"), f"invalid dsink all-reduce op {op!r}"
    return op
