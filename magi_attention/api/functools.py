# API helper functions (reference magi_attention/api/functools.py:27-335 —
# signatures kept).
from __future__ import annotations

from typing import List, Tuple

import torch

from ..common.enum import AttnMaskType
from ..common.range import AttnRange
from ..common.ranges import AttnRanges
from ..meta.geometry import normalize


def compute_pad_size(total_seqlen_q: int, cp_size: int, chunk_size: int) -> int:
    """Tokens to pad so total is divisible by chunk_size*cp_size
    (reference functools.py:27)."""
    block = chunk_size * cp_size
    rem = total_seqlen_q % block
    return 0 if rem == 0 else block - rem


def squash_batch_dim(x: torch.Tensor) -> torch.Tensor:
    """[b, s, ...] -> [b*s, ...] (reference functools.py:54)."""
    return x.reshape(-1, *x.shape[2:])


def infer_varlen_mask_from_batch(
    batch_size: int, seq_len: int
) -> tuple[AttnRanges, AttnRanges]:
    """Per-sample full ranges for a squashed [b, s] batch
    (reference functools.py:68)."""
    rr = AttnRanges.from_ranges(
        [(i * seq_len, (i + 1) * seq_len) for i in range(batch_size)]
    )
    return rr, rr.clone()


def pad_at_dim(x: torch.Tensor, dim: int, pad_size: int,
               value: float = 0.0) -> torch.Tensor:
    if pad_size == 0:
        return x
    shape = list(x.shape)
    shape[dim] = pad_size
    return torch.cat([x, x.new_full(shape, value)], dim=dim)


def unpad_at_dim(x: torch.Tensor, dim: int, orig_size: int) -> torch.Tensor:
    return x.narrow(dim, 0, orig_size)


def apply_padding(x: torch.Tensor, pad_size: int) -> torch.Tensor:
    return pad_at_dim(x, 0, pad_size)


def _band_to_triples(
    qs: int, qe: int, ks: int, ke: int, window_size: Tuple[int, int]
) -> tuple[AttnRanges, AttnRanges, List[AttnMaskType]]:
    """Sliding-window region as aligned typed slices via the band-geometry
    engine (role of reference functools.py:180; bottom-right aligned window)."""
    sk = ke - ks
    # Reference functools.py:218-224: when q_range is longer than k_range,
    # only the LAST sk q rows participate (bottom-right alignment); earlier
    # rows attend nothing and get no triple. The window-clamp shortcut below
    # is only exact after this trim.
    if qe - qs > sk:
        qs = qe - sk
    left, right = window_size
    D = ke - qe  # bottom-right diagonal offset in global coords
    lo = None if (left == -1 or left >= sk - 1) else D - left
    up = None if (right == -1 or right >= sk - 1) else D + right
    slices = normalize(qs, qe, ks, ke, lo, up)
    qr, kr, tt = AttnRanges(), AttnRanges(), []
    for sl in slices:
        qr.append(AttnRange(sl.qs, sl.qe))
        kr.append(AttnRange(sl.ks, sl.ke))
        tt.append(AttnMaskType.from_int_type(sl.t))
    return qr, kr, tt


def infer_attn_mask_from_sliding_window(
    q_range: AttnRange,
    k_range: AttnRange,
    window_size: Tuple[int, int],
) -> tuple[AttnRanges, AttnRanges, List[AttnMaskType]]:
    """Reference functools.py:180: one sliding-window mask region ->
    (q_ranges, k_ranges, mask types)."""
    assert len(window_size) == 2
    return _band_to_triples(
        q_range.start, q_range.end, k_range.start, k_range.end, window_size
    )


def infer_attn_mask_from_cu_seqlens(
    cu_seqlens_q: torch.Tensor,
    cu_seqlens_k: torch.Tensor,
    causal: bool = False,
    window_size: Tuple[int, int] = (-1, -1),
    global_window_size: int = 0,
) -> tuple[AttnRanges, AttnRanges, List[AttnMaskType], int, int]:
    """Reference functools.py:335: varlen doc masks (optionally causal or
    sliding-window) from cu_seqlens."""
    assert global_window_size == 0, "global_window_size lands in a later round"
    cq = cu_seqlens_q.tolist()
    ck = cu_seqlens_k.tolist()
    total_q, total_k = int(cq[-1]), int(ck[-1])
    if window_size == (-1, -1):
        q_ranges = AttnRanges.from_ranges(list(zip(cq[:-1], cq[1:])))
        k_ranges = AttnRanges.from_ranges(list(zip(ck[:-1], ck[1:])))
        t = AttnMaskType.CAUSAL if causal else AttnMaskType.FULL
        return q_ranges, k_ranges, [t] * len(q_ranges), total_q, total_k
    assert not causal, "causal must be False when window_size is set"
    q_ranges, k_ranges = AttnRanges(), AttnRanges()
    types: List[AttnMaskType] = []
    for (qa, qb), (ka, kb) in zip(zip(cq[:-1], cq[1:]), zip(ck[:-1], ck[1:])):
        qr, kr, tt = _band_to_triples(qa, qb, ka, kb, window_size)
        q_ranges.extend(qr)
        k_ranges.extend(kr)
        types.extend(tt)
    return q_ranges, k_ranges, types, total_q, total_k
