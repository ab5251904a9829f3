# Convenience alias module for the range/merge HIP ops.
from .common.range_op import (  # noqa: F401
    correct_out_lse,
    range_gather,
    range_reduce,
    range_scatter,
)
