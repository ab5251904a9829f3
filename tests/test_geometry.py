"""Mask-geometry engine vs dense masks (the planner's correctness core)."""
import random

import pytest
import torch

from magi_attention.meta.geometry import (
    MaskSlice, area_in_rows, k_window, q_window, slice_from_raw, to_dense,
)
from oracle import make_attn_mask


def rand_slice(rng, tq, tk):
    qs = rng.randrange(0, tq - 1)
    qe = rng.randrange(qs + 1, tq + 1)
    ks = rng.randrange(0, tk - 1)
    ke = rng.randrange(ks + 1, tk + 1)
    t = rng.randrange(4)
    return qs, qe, ks, ke, t


def test_slice_from_raw_matches_reference_mask_semantics():
    rng = random.Random(0)
    tq = tk = 64
    for _ in range(200):
        qs, qe, ks, ke, t = rand_slice(rng, tq, tk)
        ours = to_dense(slice_from_raw(qs, qe, ks, ke, t), tq, tk)
        ref = make_attn_mask(tq, tk, [[qs, qe]], [[ks, ke]], [t])
        assert torch.equal(ours, ref), (qs, qe, ks, ke, t)


def test_q_and_k_window_preserve_mask():
    rng = random.Random(1)
    tq = tk = 96
    for _ in range(300):
        qs, qe, ks, ke, t = rand_slice(rng, tq, tk)
        base = slice_from_raw(qs, qe, ks, ke, t)
        dense = to_dense(base, tq, tk)
        # random q window
        a = rng.randrange(0, tq)
        b = rng.randrange(a + 1, tq + 1)
        sub = []
        for sl in base:
            sub.extend(q_window(sl, a, b))
        exp = dense.clone()
        exp[:a] = False
        exp[b:] = False
        assert torch.equal(to_dense(sub, tq, tk), exp), ("q", qs, qe, ks, ke, t, a, b)
        # random k window
        a = rng.randrange(0, tk)
        b = rng.randrange(a + 1, tk + 1)
        sub = []
        for sl in base:
            sub.extend(k_window(sl, a, b))
        exp = dense.clone()
        exp[:, :a] = False
        exp[:, b:] = False
        assert torch.equal(to_dense(sub, tq, tk), exp), ("k", qs, qe, ks, ke, t, a, b)


def test_area_matches_dense():
    rng = random.Random(2)
    tq = tk = 80
    for _ in range(200):
        qs, qe, ks, ke, t = rand_slice(rng, tq, tk)
        base = slice_from_raw(qs, qe, ks, ke, t)
        dense = to_dense(base, tq, tk)
        assert sum(s.area() for s in base) == int(dense.sum())
        a = rng.randrange(0, tq)
        b = rng.randrange(a + 1, tq + 1)
        got = sum(area_in_rows(sl, a, b) for sl in base)
        assert got == int(dense[a:b].sum()), (qs, qe, ks, ke, t, a, b)


def test_no_double_coverage_after_windowing():
    """windows must PARTITION the mask (no overlapping sub-slices), since the
    kernel merges overlapping slices additively."""
    rng = random.Random(3)
    tq = tk = 64
    for _ in range(200):
        qs, qe, ks, ke, t = rand_slice(rng, tq, tk)
        base = slice_from_raw(qs, qe, ks, ke, t)
        cuts = sorted(rng.sample(range(1, tk), 3))
        pieces = []
        for sl in base:
            prev = 0
            for c in cuts + [tk]:
                pieces.extend(k_window(sl, prev, c))
                prev = c
        cnt = torch.zeros(tq, tk, dtype=torch.int32)
        for sl in pieces:
            cnt += to_dense([sl], tq, tk).int()
        assert int(cnt.max()) <= 1
        assert torch.equal(cnt.bool(), to_dense(base, tq, tk))
