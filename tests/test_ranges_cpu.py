"""Fuzz tests for the AttnRange/AttnRanges set algebra against a
brute-force token-set model (reference surface common/range.py,
ranges.py — these primitives drive the planner's transfer tables)."""
import numpy as np

from magi_attention.common import AttnRange, AttnRanges


def _rand_ranges(rng, n_max=40, k_max=5):
    rr = AttnRanges()
    for _ in range(int(rng.integers(0, k_max + 1))):
        a = int(rng.integers(0, n_max))
        b = int(rng.integers(a, n_max + 1))
        rr.append(AttnRange(a, b))
    return rr


def _tokens(rr):
    s = set()
    for r in rr:
        s |= set(range(r.start, r.end))
    return s


def test_merge_and_sizes_fuzz():
    rng = np.random.default_rng(21)
    for _ in range(200):
        rr = _rand_ranges(rng)
        toks = _tokens(rr)
        m = rr.merge()
        assert _tokens(m) == toks
        assert m.is_sorted() and m.is_non_overlap() and m.is_merged()
        assert m.total_seqlen == len(toks)
        assert rr.union_size() == len(toks)
        dup = sum(r.seqlen for r in rr) - len(toks)
        assert rr.intersect_size() == dup


def test_hole_and_overlap_fuzz():
    rng = np.random.default_rng(22)
    for _ in range(200):
        a = _rand_ranges(rng)
        b = _rand_ranges(rng)
        ta, tb = _tokens(a), _tokens(b)
        holes = a.find_hole_ranges(b)
        assert _tokens(holes) == ta - tb
        overlap = a.find_overlap_ranges(b)
        assert _tokens(overlap) == ta & tb
        assert a.intersect_size_with(b) == len(ta & tb)
        assert a.union_size_with(b) == len(ta | tb)


def test_make_range_local_fuzz():
    """Local coords = position within the concatenation of the merged
    ranges; every sub-range of a member maps consistently and round-trips
    through the concatenated token order."""
    rng = np.random.default_rng(23)
    for _ in range(100):
        rr = _rand_ranges(rng, k_max=4)
        base = rr.merge()
        if base.is_empty():
            continue
        order = sorted(_tokens(rr))
        for r in base:
            s = int(rng.integers(r.start, r.end))
            e = int(rng.integers(s + 1, r.end + 1))
            loc = base.make_range_local(AttnRange(s, e), is_self_merged=True)
            assert order[loc.start] == s
            assert order[loc.end - 1] == e - 1
            assert loc.seqlen == e - s


def test_range_set_ops_fuzz():
    rng = np.random.default_rng(24)
    for _ in range(300):
        sa_ = int(rng.integers(0, 20)); a = AttnRange(sa_, int(rng.integers(sa_, 21)))
        sb_ = int(rng.integers(0, 20)); b = AttnRange(sb_, int(rng.integers(sb_, 21)))
        sa = set(range(a.start, a.end))
        sb = set(range(b.start, b.end))
        assert a.intersect_size(b) == len(sa & sb)
        assert a.union_size(b) == len(sa | sb)
        assert a.is_overlap_with(b) == bool(sa & sb)
        if sa:
            assert a.is_subrange_of(b) == (sa <= sb)
        diff = set()
        for d in a.diff_by(b):
            diff |= set(range(d.start, d.end))
        assert diff == sa - sb
        uni = set()
        for u in a.union(b):
            uni |= set(range(u.start, u.end))
        assert uni == sa | sb
