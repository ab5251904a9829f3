"""Index-attention helper utilities (reference surface:
magi_attention/utils/sparse_utils.py:534 build_index_attn_indices,
:576 get_sdpa_mask_from_index_attn_indices).

The index_attn contract (see functional/flex_flash_attn.py and
csrc/ffa_index.hip): Q/K/V are packed in (b, s, h) row order, KV heads folded
into the K row dimension, so the GLOBAL K row id for (batch b, token t,
kv-head h) is (b * S_kv + t) * NHK + h. indices[(b*S_q + t), j] lists the
global rows that q token row attends (slot dim j = the kv head the ids belong
to); -1 entries are contiguous tail padding.
"""
from __future__ import annotations

import torch


def build_index_attn_indices(
    B: int,
    NHK: int,
    S_q: int,
    S_kv: int,
    topk,
    max_topk: int,
    device: str | torch.device = "cuda",
    k_block_size: int = 1,
) -> torch.Tensor:
    """Random per-token topk index lists in the global-row-id encoding.

    topk: int, or list[int] of length B for per-batch topk. Returns
    (B * S_q, NHK, max_topk) int32, -1-padded, each list sorted ascending
    (sampled without replacement from the batch's own S_kv tokens).
    """
    assert k_block_size == 1, "only token-level KV (k_block_size=1) is supported"
    topk_per_batch = [topk] * B if isinstance(topk, int) else list(topk)
    assert len(topk_per_batch) == B
    assert all(0 < t <= S_kv and t <= max_topk for t in topk_per_batch)

    total_q = B * S_q
    out = torch.full((total_q, NHK, max_topk), -1, dtype=torch.int32, device=device)
    h_off = torch.arange(NHK, device=device).view(1, 1, NHK, 1)
    for b in range(B):
        tk = topk_per_batch[b]
        # per-(token, head) sample without replacement: first tk of a random
        # permutation, reported in ascending order
        scores = torch.rand(S_q, NHK, S_kv, device=device)
        sel = scores.argsort(dim=-1)[..., :tk].sort(dim=-1).values  # local ids
        gids = (b * S_kv + sel.unsqueeze(0)) * NHK + h_off  # (1, S_q, NHK, tk)
        out[b * S_q : (b + 1) * S_q, :, :tk] = gids.squeeze(0).int()
    return out


def get_sdpa_mask_from_index_attn_indices(
    index_attn_indices: torch.Tensor,
    B: int,
    NHQ: int,
    NHK: int,
    S_q: int,
    S_kv: int,
    device: str | torch.device = "cuda",
    k_block_size: int = 1,
) -> torch.Tensor:
    """Dense boolean SDPA mask [B, NHQ, S_q, S_kv] from index lists.

    An id g in slot (row i, head j) allows q token i (all its GQA query
    heads under kv head j) to attend kv column g // NHK - b * S_kv of batch
    b = i // S_q. Invalid (-1) entries contribute nothing.
    """
    assert k_block_size == 1, "only token-level KV (k_block_size=1) is supported"
    total_q = B * S_q
    assert index_attn_indices.shape[0] == total_q
    assert index_attn_indices.shape[1] == NHK
    gqa = NHQ // NHK

    idx = index_attn_indices.to(device=device, dtype=torch.int64)
    row = (
        torch.arange(total_q, device=device)
        .view(total_q, 1, 1)
        .expand_as(idx)
    )
    slot = (
        torch.arange(NHK, device=device).view(1, NHK, 1).expand_as(idx)
    )
    valid = idx >= 0
    r, j, g = row[valid], slot[valid], idx[valid]
    b = r // S_q
    t = r % S_q
    col = g // NHK - b * S_kv
    assert bool((col >= 0).all()) and bool((col < S_kv).all()), (
        "index_attn id decodes outside its batch's KV range"
    )

    mask = torch.zeros(B, NHK, S_q, S_kv, dtype=torch.bool, device=device)
    mask[b, j, t, col] = True
    if gqa > 1:
        mask = (
            mask.unsqueeze(2)
            .expand(B, NHK, gqa, S_q, S_kv)
            .reshape(B, NHQ, S_q, S_kv)
        )
    return mask
