"""GroundTruthDispatcher (reference surface: testing/gt_dispatcher.py:27):
the naive, dense-mask-backed computation of per-chunk attention areas that
dispatch tests compare the real (arithmetic) planner against — every chunk's
slices come from materializing the full AttnMask and re-inferring sub-mask
tuples, with areas counted cell by cell."""
from __future__ import annotations

from typing import List

from torch import nn

from ..common import AttnMask, AttnRanges
from ..common.enum import AttnMaskType
from ..common.range import AttnRange
from ..config import DispatchAlg
from ..meta.container import AttnBucket, AttnChunk, AttnSlice


class GroundTruthDispatcher(nn.Module):
    def __init__(self, alg: DispatchAlg) -> None:
        super().__init__()
        self.alg = alg
        self._self_attn_mask: AttnMask = None  # type: ignore[assignment]
        self._cross_attn_mask: AttnMask = None  # type: ignore[assignment]
        self._chunk_masks: list[AttnMask] = []

    def _compute_self_attn_areas(
        self,
        q_ranges: AttnRanges,
        k_ranges: AttnRanges,
        attn_mask_type: List[AttnMaskType],
        chunk_size: int | None = None,
    ) -> AttnBucket:
        """Dense-mask ground truth of make_global_bucket_from_qk_ranges:
        materialize the whole mask, slice per chunk, re-infer each chunk's
        (q_range, k_range, type) tuples and count areas exactly."""
        ts = q_ranges.end
        if chunk_size is None:
            chunk_size = ts
        num_chunks = ts // chunk_size
        full_k = AttnRange(start=0, end=ts)

        self._self_attn_mask = AttnMask.from_ranges(
            q_ranges=q_ranges,
            k_ranges=k_ranges,
            attn_mask_type=attn_mask_type,
            total_seqlen_q=ts,
            total_seqlen_k=ts,  # self-attn: k space = q space
        )

        bucket = AttnBucket()
        self._chunk_masks = []
        for chunk_id in range(num_chunks):
            lo = chunk_id * chunk_size
            chunk_mask = self._self_attn_mask.make_sub_mask(
                q_range=AttnRange(lo, lo + chunk_size), k_range=full_k
            )
            self._chunk_masks.append(chunk_mask)
            chunk = AttnChunk(chunk_id=chunk_id)
            for slice_id, (qr, kr, mt) in enumerate(chunk_mask.tuples()):
                s = AttnSlice(
                    slice_id=slice_id,
                    q_range=qr.offset(lo),
                    k_range=kr,
                    mask_type=mt,
                )
                s.area = chunk_mask.calc_sub_area(q_range=qr, k_range=kr)
                chunk.q_slices.append(s)
            bucket.q_chunks.append(chunk)
        return bucket
