"""CPU tests for the index-attention helpers (utils/sparse_utils.py),
checked against a brute-force restatement of the reference semantics
(reference utils/sparse_utils.py:534-634)."""
import torch

from magi_attention.utils import (
    build_index_attn_indices,
    get_sdpa_mask_from_index_attn_indices,
)


def _mask_bruteforce(indices, B, NHQ, NHK, S_q, S_kv):
    gqa = NHQ // NHK
    m = torch.zeros(B, NHQ, S_q, S_kv, dtype=torch.bool)
    for i in range(B * S_q):
        b, t = i // S_q, i % S_q
        for j in range(NHK):
            for g in indices[i, j].tolist():
                if g < 0:
                    continue
                col = g // NHK - b * S_kv
                for g2 in range(gqa):
                    m[b, j * gqa + g2, t, col] = True
    return m


def test_build_indices_properties():
    torch.manual_seed(0)
    B, NHK, S_q, S_kv, tk, mt = 2, 3, 4, 16, 5, 8
    idx = build_index_attn_indices(B, NHK, S_q, S_kv, tk, mt, device="cpu")
    assert idx.shape == (B * S_q, NHK, mt) and idx.dtype == torch.int32
    # padding contiguous at tail, exactly mt - tk entries
    assert (idx[:, :, tk:] == -1).all() and (idx[:, :, :tk] >= 0).all()
    for i in range(B * S_q):
        b = i // S_q
        for j in range(NHK):
            row = idx[i, j, :tk].long()
            assert (row % NHK == j).all()          # head encoding
            loc = row // NHK - b * S_kv
            assert (0 <= loc).all() and (loc < S_kv).all()  # batch-local
            assert (loc[1:] > loc[:-1]).all()      # sorted, no repeats


def test_build_indices_per_batch_topk():
    torch.manual_seed(1)
    idx = build_index_attn_indices(3, 2, 4, 32, [3, 7, 1], 8, device="cpu")
    for b, tk in enumerate([3, 7, 1]):
        blk = idx[b * 4 : (b + 1) * 4]
        assert (blk[:, :, :tk] >= 0).all() and (blk[:, :, tk:] == -1).all()


def test_sdpa_mask_matches_bruteforce():
    torch.manual_seed(2)
    for B, NHQ, NHK, S_q, S_kv, tk, mt in [
        (1, 4, 1, 8, 16, 4, 8),
        (2, 6, 3, 4, 12, 5, 8),
        (2, 2, 2, 4, 8, [3, 8], 8),
    ]:
        idx = build_index_attn_indices(B, NHK, S_q, S_kv, tk, mt, device="cpu")
        got = get_sdpa_mask_from_index_attn_indices(
            idx, B, NHQ, NHK, S_q, S_kv, device="cpu"
        )
        want = _mask_bruteforce(idx, B, NHQ, NHK, S_q, S_kv)
        assert torch.equal(got, want)


def test_index_attn_api_validation():
    import pytest

    from magi_attention.functional import flex_flash_attn_func

    q = torch.randn(8, 4, 64, dtype=torch.bfloat16)
    k = torch.randn(16, 1, 64, dtype=torch.bfloat16)
    v = torch.randn_like(k)
    idx = torch.zeros(8, 1, 64, dtype=torch.int32)
    rng = torch.tensor([[0, 8]], dtype=torch.int32)
    # both or neither sparse inputs rejected
    with pytest.raises(AssertionError):
        flex_flash_attn_func(q, k, v, rng, rng, index_attn_indices=idx)
    with pytest.raises(AssertionError):
        flex_flash_attn_func(q, k, v, None, None)
    # max_topk must be a multiple of 64
    with pytest.raises(AssertionError):
        flex_flash_attn_func(
            q, k, v, index_attn_indices=torch.zeros(8, 1, 32, dtype=torch.int32)
        )
    # forward-only
    qg = q.clone().requires_grad_(True)
    with pytest.raises(AssertionError):
        flex_flash_attn_func(qg, k, v, index_attn_indices=idx)
