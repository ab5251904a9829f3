from . import metaclass, nvtx, sparse_utils  # noqa: F401
from ._utils import *  # noqa: F401,F403
from ._utils import __all__ as _utils_all
from .debug import debugpy_listen  # noqa: F401
from .sparse_utils import (
    build_index_attn_indices,
    get_sdpa_mask_from_index_attn_indices,
)

__all__ = [
    "nvtx",
    "debugpy_listen",
    "sparse_utils",
    "metaclass",
    "build_index_attn_indices",
    "get_sdpa_mask_from_index_attn_indices",
    *_utils_all,
]
