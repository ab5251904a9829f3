"""Functional surface (reference magi_attention/functional/__init__.py)."""
from .dispatch import dispatch_func, undispatch_func  # noqa: F401
from .dist_attn import dist_attn_func  # noqa: F401
from .flex_flash_attn import (  # noqa: F401
    flex_flash_attn_func,
    merge_ranges,
)
from .roll import roll_p2p as roll_func  # noqa: F401
from .roll import roll_simple_p2p as roll_simple_func  # noqa: F401
from .utils import (  # noqa: F401
    correct_attn_lse,
    correct_attn_lse_with_sink,
    correct_attn_out,
    correct_attn_out_lse,
    correct_attn_out_lse_with_sink,
    correct_attn_out_with_sink,
)


def ffa_fa4_func(*args, **kwargs):
    """Reference FA4 (Blackwell) kernel backend — NVIDIA-only; the MI355X
    engine has a single native FFA backend."""
    raise NotImplementedError(
        "The FA4 backend targets NVIDIA sm100; this engine's native gfx950 "
        "FFA kernels are the only backend."
    )


__all__ = [
    "dispatch_func",
    "undispatch_func",
    "dist_attn_func",
    "flex_flash_attn_func",
    "merge_ranges",
    "roll_func",
    "roll_simple_func",
    "correct_attn_lse",
    "correct_attn_lse_with_sink",
    "correct_attn_out",
    "correct_attn_out_lse",
    "correct_attn_out_lse_with_sink",
    "correct_attn_out_with_sink",
    "ffa_fa4_func",
]
