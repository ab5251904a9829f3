"""CPU tests for common.AttnRectangle / AttnRectangles (reference surface
common/rectangle.py:28, rectangles.py:31): brute-force rasterization is the
oracle for band membership, area, cuts and the to_qk_range_mask_type
decomposition."""
import numpy as np
import torch

from magi_attention.common import (
    AttnMaskType,
    AttnRange,
    AttnRectangle,
    AttnRectangles,
)
from oracle import make_attn_mask


def _raster(rect, tq, tk):
    """Dense bool [tq, tk] of the rectangle's (q, k) membership."""
    m = torch.zeros(tq, tk, dtype=torch.bool)
    q = torch.arange(tq).unsqueeze(1)
    k = torch.arange(tk).unsqueeze(0)
    m = (
        (q >= rect.q_range.start) & (q < rect.q_range.end)
        & (k >= rect.k_range.start) & (k < rect.k_range.end)
        & (k - q >= rect.d_range.start) & (k - q <= rect.d_range.end)
    )
    return m


def _rand_rect(rng, tq, tk):
    """Random valid rectangle (or None if the band is empty)."""
    qs = int(rng.integers(0, tq - 1)); qe = int(rng.integers(qs + 1, tq + 1))
    ks = int(rng.integers(0, tk - 1)); ke = int(rng.integers(ks + 1, tk + 1))
    dlo = int(rng.integers(ks - qe, ke - qs))
    dhi = int(rng.integers(dlo, ke - qs))
    try:
        return AttnRectangle(
            AttnRange(qs, qe), AttnRange(ks, ke), AttnRange(dlo, dhi)
        )
    except ValueError:
        return None


def test_mask_type_seeding_matches_oracle():
    """from (q_range, k_range, mask_type): the rectangle's raster must equal
    the FFA mask-type dense mask of the same slice."""
    rng = np.random.default_rng(0)
    for _ in range(40):
        tq, tk = int(rng.integers(2, 16)), int(rng.integers(2, 16))
        qs = int(rng.integers(0, tq - 1)); qe = int(rng.integers(qs + 1, tq + 1))
        ks = int(rng.integers(0, tk - 1)); ke = int(rng.integers(ks + 1, tk + 1))
        for t in range(4):
            if t == 3 and (qe - qs) > (ke - ks):
                continue  # degenerate bi-causal (empty rows)
            want = make_attn_mask(tq, tk, [[qs, qe]], [[ks, ke]], [t])
            try:
                rect = AttnRectangle(
                    AttnRange(qs, qe), AttnRange(ks, ke), mask_type=t
                )
            except ValueError:
                assert not want.any(), (qs, qe, ks, ke, t)
                continue
            got = _raster(rect, tq, tk)
            assert torch.equal(got, want), (qs, qe, ks, ke, t)
            assert rect.area() == int(want.sum())


def test_area_matches_bruteforce():
    rng = np.random.default_rng(1)
    for _ in range(100):
        tq, tk = int(rng.integers(2, 20)), int(rng.integers(2, 20))
        rect = _rand_rect(rng, tq, tk)
        if rect is None:
            continue
        assert rect.area() == int(_raster(rect, tq, tk).sum().item())


def test_cuts_partition_area_and_raster():
    rng = np.random.default_rng(2)
    for _ in range(60):
        tq, tk = int(rng.integers(3, 18)), int(rng.integers(3, 18))
        rect = _rand_rect(rng, tq, tk)
        if rect is None:
            continue
        base = _raster(rect, tq, tk)
        for cutter, lo, hi in (
            (rect.cut_q, 0, tq), (rect.cut_k, 0, tk)
        ):
            pos = int(rng.integers(lo, hi + 1))
            a, b = cutter(pos)
            got = torch.zeros_like(base)
            for part in (a, b):
                if part is not None:
                    got |= _raster(part, tq, tk)
            assert torch.equal(got, base), (rect, pos)
            area = sum(p.area() for p in (a, b) if p is not None)
            assert area == rect.area()


def test_decomposition_matches_raster():
    """to_qk_range_mask_type: the FFA triples must re-rasterize to exactly
    the rectangle's band, with disjoint q rows."""
    rng = np.random.default_rng(3)
    checked = 0
    for _ in range(120):
        tq, tk = int(rng.integers(2, 22)), int(rng.integers(2, 22))
        rect = _rand_rect(rng, tq, tk)
        if rect is None:
            continue
        triples = rect.to_qk_range_mask_type()
        got = torch.zeros(tq, tk, dtype=torch.bool)
        q_seen = set()
        for qr, kr, t in triples:
            rows = set(range(qr.start, qr.end))
            assert not (rows & q_seen), "q rows overlap across triples"
            q_seen |= rows
            got |= make_attn_mask(
                tq, tk, [[qr.start, qr.end]], [[kr.start, kr.end]], [t]
            )
        assert torch.equal(got, _raster(rect, tq, tk)), rect
        checked += 1
    assert checked > 60


def test_classification():
    r = AttnRectangle(AttnRange(0, 4), AttnRange(0, 4), mask_type=1)
    assert r.is_causal() and not r.is_full()
    assert r.to_qk_range_mask_type() == [(AttnRange(0, 4), AttnRange(0, 4), 1)]
    r = AttnRectangle(AttnRange(0, 4), AttnRange(0, 8), mask_type=0)
    assert r.is_full()
    r = AttnRectangle(AttnRange(0, 4), AttnRange(0, 6), mask_type=3)
    assert r.is_bi_causal()
    r = AttnRectangle(AttnRange(2, 6), AttnRange(0, 8), mask_type=2)
    assert r.is_inv_causal()


def test_rectangles_container():
    rects = AttnRectangles.from_ranges(
        [(0, 4), (4, 8), (5, 5)],
        [(0, 4), (2, 8), (0, 8)],
        [1, 0, 0],
    )
    assert rects.size == 2  # empty q range dropped
    assert rects.area() == 10 + 4 * 6
    assert rects.total_seqlen_qo() == 8
    assert rects.get_kv_ranges_union().total_seqlen == 8
    left, right = rects.cut_q(4)
    assert left.area() + right.area() == rects.area()
    assert left.size == 1 and right.size == 1
    seg = rects.get_rects_within_q_segment(2, 6)
    tq, tk = 8, 8
    want = torch.zeros(tq, tk, dtype=torch.bool)
    for r in rects:
        want |= _raster(r, tq, tk)
    want[:2] = False
    want[6:] = False
    got = torch.zeros(tq, tk, dtype=torch.bool)
    for r in seg:
        got |= _raster(r, tq, tk)
    assert torch.equal(got, want)
    # degenerate bi-causal dropped
    assert AttnRectangles.from_ranges([(0, 5)], [(0, 3)], [3]).is_empty()


def test_rectangles_indexing_and_eq():
    rects = AttnRectangles.from_ranges(
        [(0, 4), (4, 8)], [(0, 4), (0, 8)], [1, 1]
    )
    assert len(rects) == 2 and not rects.is_empty()
    sub = rects[0:1]
    assert isinstance(sub, AttnRectangles) and sub.size == 1
    assert sub[0] == rects[0]
    clone = AttnRectangles.from_ranges(
        [(0, 4), (4, 8)], [(0, 4), (0, 8)], [1, 1]
    )
    assert clone == rects and hash(clone) == hash(rects)


def test_reference_golden_cases():
    """Concrete expectations ported from the reference's own
    tests/test_common/test_rectangle.py (init shrinks, mask-type d-band
    seeding, boundary intersections, validity flips)."""
    r = AttnRectangle(AttnRange(0, 10), AttnRange(0, 20), AttnRange(-5, 5))
    assert (r.q_range.start, r.q_range.end) == (0, 10)
    assert (r.k_range.start, r.k_range.end) == (0, 15)  # k shrunk by d_end
    assert (r.d_range.start, r.d_range.end) == (-5, 5)
    assert r.intersection_q_id_on_left_boundary() == 5
    assert r.intersection_q_id_on_right_boundary() == 9

    r = AttnRectangle(AttnRange(0, 10), AttnRange(0, 20), AttnRange(-100, 100))
    assert (r.d_range.start, r.d_range.end) == (-9, 19)

    r = AttnRectangle(AttnRange(0, 10), AttnRange(0, 20),
                      mask_type=AttnMaskType.CAUSAL)
    assert r.d_range.end == 10  # k_end - q_end
    r = AttnRectangle(AttnRange(0, 10), AttnRange(0, 20),
                      mask_type=AttnMaskType.BICAUSAL)
    assert (r.d_range.start, r.d_range.end) == (0, 10)
    r = AttnRectangle(AttnRange(0, 10), AttnRange(0, 20),
                      mask_type=AttnMaskType.INVCAUSAL)
    assert r.d_range.start == 0

    # unchecked setters allow a temporarily-invalid state; is_valid flags it
    bad = AttnRectangle(AttnRange(0, 10), AttnRange(0, 20), AttnRange(-5, 5))
    bad.q_range.start = 10
    bad.q_range.end = 0
    assert not bad.is_valid() and bad.get_valid_or_none() is None

    # segment clips (reference expectations)
    r = AttnRectangle(AttnRange(0, 20), AttnRange(0, 20), AttnRange(-5, 5))
    seg = r.get_rect_within_q_segment(5, 15)
    assert (seg.q_range.start, seg.q_range.end) == (5, 15)
    assert r.get_rect_within_q_segment(25, 35) is None
    seg = r.get_rect_within_k_segment(5, 15)
    assert (seg.k_range.start, seg.k_range.end) == (5, 15)
    assert r.get_rect_within_k_segment(25, 35) is None


def test_decomposition_reference_band_regime():
    """The reference's to_qk_range_mask_type contract on its own sampling
    regime (d band spanning both corner diagonals): at most 3 parts, and
    the parts RECONSTRUCTED from (q, k, mask_type) alone reproduce the
    area (round-trips the mask-type constructor)."""
    rng = np.random.default_rng(9)
    for _ in range(100):
        qs = int(rng.integers(0, 99)); qe = int(rng.integers(qs + 1, 101))
        ks = int(rng.integers(0, 99)); ke = int(rng.integers(ks + 1, 101))
        d_min = ks - (qe - 1)
        lo_c, hi_c = sorted((ks - qs, ke - qe))
        d_max = ke - 1 - qs
        ds = int(rng.integers(d_min, lo_c + 1))
        de = int(rng.integers(max(hi_c, ds), d_max + 1))
        rect = AttnRectangle(AttnRange(qs, qe), AttnRange(ks, ke),
                             AttnRange(ds, de))
        parts = rect.to_qk_range_mask_type()
        assert len(parts) <= 3, rect
        rebuilt = sum(
            AttnRectangle(qr, kr, mask_type=t).area() for qr, kr, t in parts
        )
        assert rebuilt == rect.area(), rect
