"""CPU tests for the chunked-mask bucket IR (meta/container +
meta/_buckets factories, reference _make_dispatch_meta.py:251/:377):
per-chunk slices must re-rasterize to exactly the chunk's rows of the
global mask, and bucket areas must match brute-force counts."""
import numpy as np
import torch

from magi_attention.common import AttnMaskType, AttnRanges
from magi_attention.meta import (
    DispatchMeta,
    make_bucket_per_rank_from_qk_ranges,
    make_global_bucket_from_qk_ranges,
)
from oracle import make_attn_mask


def _rand_mask(rng, n, n_ranges):
    cuts = sorted(rng.choice(np.arange(1, n), size=n_ranges - 1, replace=False))
    bounds = [0] + [int(c) for c in cuts] + [n]
    qrs, krs, tts = [], [], []
    for a, b in zip(bounds, bounds[1:]):
        ks = int(rng.integers(0, n - 1))
        ke = int(rng.integers(ks + 1, n + 1))
        t = int(rng.integers(0, 4))
        if t == 3 and (b - a) > (ke - ks):
            t = 1  # degenerate bi-causal -> causal
        qrs.append((a, b))
        krs.append((ks, ke))
        tts.append(t)
    return qrs, krs, tts


def test_global_bucket_rasters_to_mask():
    rng = np.random.default_rng(5)
    for _ in range(25):
        n = int(rng.integers(8, 33) // 4 * 4)
        chunk = int(rng.choice([2, 4, 8]))
        if n % chunk:
            continue
        qrs, krs, tts = _rand_mask(rng, n, int(rng.integers(2, 5)))
        dense = make_attn_mask(n, n, qrs, krs, tts)
        bucket = make_global_bucket_from_qk_ranges(
            AttnRanges.from_ranges(qrs),
            AttnRanges.from_ranges(krs),
            [AttnMaskType.from_int_type(t) for t in tts],
            num_chunks=n // chunk,
            chunk_size=chunk,
        )
        assert len(bucket.q_chunks) == n // chunk
        got = torch.zeros(n, n, dtype=torch.bool)
        for c in bucket.q_chunks:
            lo, hi = c.chunk_id * chunk, (c.chunk_id + 1) * chunk
            for s in c.q_slices:
                assert lo <= s.q_range.start and s.q_range.end <= hi
                sub = make_attn_mask(
                    n, n,
                    [[s.q_range.start, s.q_range.end]],
                    [[s.k_range.start, s.k_range.end]],
                    [s.mask_type.to_int_type()],
                )
                assert not (got & sub).any(), "slices overlap"
                got |= sub
                assert s.area == int(sub.sum())
            assert c.area == int(dense[lo:hi].sum())
        assert torch.equal(got, dense)
        assert bucket.area == int(dense.sum())


def test_bucket_per_rank_grouping():
    rng = np.random.default_rng(6)
    n, chunk, cp = 32, 4, 4
    qrs, krs, tts = _rand_mask(rng, n, 3)
    dense = make_attn_mask(n, n, qrs, krs, tts)
    partitions = [[0, 5], [1, 4], [2, 7], [3, 6]]
    meta = DispatchMeta(
        cp_size=cp, cp_rank=0, chunk_size=chunk, total_seqlen=n,
        num_chunks=n // chunk, partitions=partitions,
    )
    buckets = make_bucket_per_rank_from_qk_ranges(
        AttnRanges.from_ranges(qrs),
        AttnRanges.from_ranges(krs),
        [AttnMaskType.from_int_type(t) for t in tts],
        meta,
    )
    assert len(buckets) == cp
    for rank, b in enumerate(buckets):
        assert b.cp_rank == rank
        want = sum(
            int(dense[c * chunk : (c + 1) * chunk].sum())
            for c in partitions[rank]
        )
        assert b.area == want
        assert b.areas == [
            int(dense[c * chunk : (c + 1) * chunk].sum())
            for c in partitions[rank]
        ]
    assert sum(b.area for b in buckets) == int(dense.sum())


def test_chunk_iou_and_slices():
    from magi_attention.meta.container import AttnChunk, AttnSlice
    from magi_attention.common.range import AttnRange

    s1 = AttnSlice(mask_type=AttnMaskType.FULL,
                   q_range=AttnRange(0, 2), k_range=AttnRange(0, 8))
    s2 = AttnSlice(mask_type=AttnMaskType.FULL,
                   q_range=AttnRange(2, 4), k_range=AttnRange(4, 12))
    assert s1.area == 16 and s2.area == 16
    assert abs(s1.iou_with(s2) - 4 / 12) < 1e-9
    c = AttnChunk(chunk_id=0, q_slices=[s1, s2])
    assert c.area == 32
    assert abs(c.iou - 4 / 12) < 1e-9
    c2 = AttnChunk(chunk_id=1, q_slices=[s2])
    assert abs(c.iou_with(c2) - 8 / 12) < 1e-9


def test_gt_dispatcher_fuzz_vs_factory():
    """Randomized agreement between the dense-mask GroundTruthDispatcher and
    the arithmetic bucket factory (FULL/CAUSAL blocks — AttnMask.from_ranges'
    domain), per-chunk."""
    from magi_attention.config import DispatchAlg
    from magi_attention.testing import GroundTruthDispatcher

    rng = np.random.default_rng(12)
    for _ in range(15):
        chunk = int(rng.choice([2, 4, 8]))
        n_chunks = int(rng.integers(2, 7))
        n = chunk * n_chunks
        cuts = sorted(set(int(c) for c in rng.integers(1, n, size=2)))
        bounds = [0] + cuts + [n]
        qrs, krs, tts = [], [], []
        for a, b in zip(bounds, bounds[1:]):
            if a == b:
                continue
            ks = int(rng.integers(0, n))
            ke = int(rng.integers(ks + 1, n + 1))
            qrs.append((a, b))
            krs.append((ks, ke))
            tts.append(AttnMaskType.CAUSAL if rng.integers(0, 2) else
                       AttnMaskType.FULL)
        gt = GroundTruthDispatcher(alg=DispatchAlg())
        b_gt = gt._compute_self_attn_areas(
            AttnRanges.from_ranges(qrs), AttnRanges.from_ranges(krs), tts,
            chunk_size=chunk,
        )
        b_ar = make_global_bucket_from_qk_ranges(
            AttnRanges.from_ranges(qrs), AttnRanges.from_ranges(krs), tts,
            num_chunks=n_chunks, chunk_size=chunk,
        )
        assert b_gt.areas == b_ar.areas, (qrs, krs, tts, chunk)
        assert b_gt.area == b_ar.area
