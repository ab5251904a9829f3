# Common types for the MI355X-native magi_attention rebuild.
from . import enum  # noqa: F401
from .enum import AttnMaskType  # noqa: F401
from .forward_meta import AttnForwardMeta  # noqa: F401
from .range import AttnRange, RangeError  # noqa: F401
from .ranges import AttnRanges  # noqa: F401

__all__ = [
    "enum",
    "AttnMaskType",
    "AttnForwardMeta",
    "AttnRange",
    "RangeError",
    "AttnRanges",
]
