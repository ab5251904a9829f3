from .sparse_utils import (
    build_index_attn_indices,
    get_sdpa_mask_from_index_attn_indices,
)

__all__ = [
    "build_index_attn_indices",
    "get_sdpa_mask_from_index_attn_indices",
]
