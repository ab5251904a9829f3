# Common types for the MI355X-native magi_attention rebuild.
from ..env import is_cpp_backend_enable  # noqa: F401
from . import enum, jit, range_op  # noqa: F401
from .enum import AttnMaskType  # noqa: F401
from .forward_meta import AttnForwardMeta  # noqa: F401
from .mask import AttnMask  # noqa: F401
from .range import AttnRange, RangeError  # noqa: F401
from .ranges import AttnRanges  # noqa: F401
from .rectangle import AttnRectangle  # noqa: F401
from .rectangles import AttnRectangles  # noqa: F401

# The reference swaps its Python range/geometry types for a pybind C++
# backend under MAGI_ATTENTION_CPP_BACKEND (common/__init__.py:36-68). The
# MI355X rebuild keeps the Python types and accelerates the heavy range OPS
# natively instead (magi_attn_ext -> libmagi_ffa.so), so this is always
# False here; the name is exported for surface parity.
USE_CPP_BACKEND = False

__all__ = [
    "enum",
    "jit",
    "AttnMask",
    "AttnMaskType",
    "AttnForwardMeta",
    "AttnRange",
    "RangeError",
    "AttnRanges",
    "AttnRectangle",
    "AttnRectangles",
    "range_op",
    "USE_CPP_BACKEND",
    "is_cpp_backend_enable",
]
