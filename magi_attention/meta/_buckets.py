"""Chunked-mask bucket factories (reference surface:
meta/_make_dispatch_meta.py:251 make_global_bucket_from_qk_ranges and :377
make_bucket_per_rank_from_qk_ranges): slice the global (q, k, type) mask
per fixed-size token chunk into the AttnBucket IR, then group chunks by the
dispatch partition. Per-chunk clipping keeps each slice's own type: a
causal slice clipped at the bottom shifts its k end with the clipped q end
(bottom-right alignment), an inv-causal slice clipped at the top shifts its
k start, a bi-causal slice shifts both; full slices keep their k range."""
from __future__ import annotations

from ..common.enum import AttnMaskType
from ..common.range import AttnRange
from ..common.ranges import AttnRanges
from ..utils import argsort
from .container import AttnBucket, AttnChunk, AttnSlice


def _sorted_by_q(q_ranges, k_ranges, attn_mask_type):
    order = argsort(q_ranges, key=lambda r: (r.start, r.end))
    return (
        AttnRanges.from_ranges([q_ranges[i] for i in order]),
        AttnRanges.from_ranges([k_ranges[i] for i in order]),
        [attn_mask_type[i] for i in order],
    )


def _clip_slice_to_rows(qr, kr, mt, row_lo: int, row_hi: int):
    """(q_range, k_range, area) of the slice restricted to q rows
    [row_lo, row_hi), or None when the clip is empty."""
    sk = kr.seqlen
    if mt == AttnMaskType.FULL:
        qs, qe = max(qr.start, row_lo), min(qr.end, row_hi)
        if qs >= qe:
            return None
        return AttnRange(qs, qe), AttnRange(kr.start, kr.end), (qe - qs) * sk
    if mt == AttnMaskType.CAUSAL:
        # rows above qe - sk have no keys
        qs = max(qr.start, row_lo, qr.end - sk)
        qe = min(qr.end, row_hi)
        if qs >= qe:
            return None
        ke = kr.end - (qr.end - qe)
        base, height = ke - kr.start, qe - qs
        return AttnRange(qs, qe), AttnRange(kr.start, ke), (
            (2 * base - height + 1) * height // 2
        )
    if mt == AttnMaskType.INVCAUSAL:
        qs = max(qr.start, row_lo)
        qe = min(qr.end, row_hi, qr.start + sk)
        if qs >= qe:
            return None
        ks = kr.start + (qs - qr.start)
        base, height = kr.end - ks, qe - qs
        return AttnRange(qs, qe), AttnRange(ks, kr.end), (
            (2 * base - height + 1) * height // 2
        )
    if mt == AttnMaskType.BICAUSAL:
        qs, qe = max(qr.start, row_lo), min(qr.end, row_hi)
        band = sk - qr.seqlen + 1
        if qs >= qe or band <= 0:
            return None
        ks = kr.start + (qs - qr.start)
        ke = kr.end - (qr.end - qe)
        return AttnRange(qs, qe), AttnRange(ks, ke), band * (qe - qs)
    raise ValueError(f"Invalid mask type {mt}")


def make_global_bucket_from_qk_ranges(
    q_ranges: AttnRanges,
    k_ranges: AttnRanges,
    attn_mask_type: list[AttnMaskType],
    num_chunks: int,
    chunk_size: int,
    chunk_actual_sizes: list[int] | None = None,
    sort: bool = True,
) -> AttnBucket:
    if sort:
        q_ranges, k_ranges, attn_mask_type = _sorted_by_q(
            q_ranges, k_ranges, attn_mask_type
        )
    n = len(q_ranges)
    bucket = AttnBucket()
    first = 0  # first range that can still intersect the sweep
    for chunk_id in range(num_chunks):
        lo = chunk_id * chunk_size
        hi = (
            lo + chunk_actual_sizes[chunk_id]
            if chunk_actual_sizes is not None
            else lo + chunk_size
        )
        chunk = AttnChunk(chunk_id=chunk_id)
        while first < n and q_ranges[first].end <= lo and q_ranges[first].start < lo:
            first += 1
        i = first
        slice_id = 0
        while i < n and q_ranges[i].start < hi:
            clip = _clip_slice_to_rows(
                q_ranges[i], k_ranges[i], attn_mask_type[i], lo, hi
            )
            if clip is not None and clip[1].seqlen > 0 and clip[2] > 0:
                s = AttnSlice(
                    slice_id=slice_id,
                    mask_type=attn_mask_type[i],
                    q_range=clip[0],
                    k_range=clip[1],
                )
                s.area = clip[2]
                chunk.q_slices.append(s)
                chunk.sample_ids.append(i)
                slice_id += 1
            i += 1
        bucket.q_chunks.append(chunk)
    return bucket


def make_bucket_per_rank_from_qk_ranges(
    q_ranges: AttnRanges,
    k_ranges: AttnRanges,
    attn_mask_type: list[AttnMaskType],
    dispatch_meta,
    sort: bool = True,
) -> list[AttnBucket]:
    global_bucket = make_global_bucket_from_qk_ranges(
        q_ranges=q_ranges,
        k_ranges=k_ranges,
        attn_mask_type=attn_mask_type,
        num_chunks=dispatch_meta.num_chunks,
        chunk_size=dispatch_meta.chunk_size,
        chunk_actual_sizes=getattr(dispatch_meta, "chunk_actual_sizes", None),
        sort=sort,
    )
    return [
        AttnBucket(
            cp_rank=rank,
            q_chunks=[global_bucket.q_chunks[c] for c in partition],
        )
        for rank, partition in enumerate(dispatch_meta.partitions)
    ]
