"""CPU tests for common.AttnMask (reference surface common/mask.py:29),
pinned on the reference's own golden matrix
(tests/test_common/test_attn_mask.py:29 test_mask_factory_constructors)
plus round-trip property checks against the oracle dense-mask builder."""
import numpy as np
import pytest
import torch

from magi_attention.common import AttnMask, AttnMaskType, AttnRanges
from magi_attention.common.range import AttnRange

Q_RANGES = [(0, 3), (3, 5), (5, 8), (8, 9), (9, 14), (14, 16)]
K_RANGES = [(0, 4), (2, 4), (3, 7), (4, 12), (6, 9), (1, 13)]
TYPES = [
    AttnMaskType.CAUSAL, AttnMaskType.CAUSAL, AttnMaskType.FULL,
    AttnMaskType.FULL, AttnMaskType.CAUSAL, AttnMaskType.CAUSAL,
]
# the reference's golden matrix for the ranges above (16x16 self-attn)
GOLDEN = [
    [1, 1, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0],
    [1, 1, 1, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0],
    [1, 1, 1, 1, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0],
    [0, 0, 1, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0],
    [0, 0, 1, 1, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0],
    [0, 0, 0, 1, 1, 1, 1, 0, 0, 0, 0, 0, 0, 0, 0, 0],
    [0, 0, 0, 1, 1, 1, 1, 0, 0, 0, 0, 0, 0, 0, 0, 0],
    [0, 0, 0, 1, 1, 1, 1, 0, 0, 0, 0, 0, 0, 0, 0, 0],
    [0, 0, 0, 0, 1, 1, 1, 1, 1, 1, 1, 1, 0, 0, 0, 0],
    [0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0],
    [0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0],
    [0, 0, 0, 0, 0, 0, 1, 0, 0, 0, 0, 0, 0, 0, 0, 0],
    [0, 0, 0, 0, 0, 0, 1, 1, 0, 0, 0, 0, 0, 0, 0, 0],
    [0, 0, 0, 0, 0, 0, 1, 1, 1, 0, 0, 0, 0, 0, 0, 0],
    [0, 1, 1, 1, 1, 1, 1, 1, 1, 1, 1, 1, 0, 0, 0, 0],
    [0, 1, 1, 1, 1, 1, 1, 1, 1, 1, 1, 1, 1, 0, 0, 0],
]


def _build():
    return AttnMask.from_ranges(
        q_ranges=AttnRanges.from_ranges(Q_RANGES),
        k_ranges=AttnRanges.from_ranges(K_RANGES),
        attn_mask_type=TYPES,
        total_seqlen_q=16,
        total_seqlen_k=16,
    )


def test_from_ranges_matches_reference_golden():
    m = _build()
    assert np.equal(m.mask_flag_array, np.array(GOLDEN)).all()
    assert torch.equal(
        m.mask_tensor[..., AttnMask.mask_flag_dim_idx],
        torch.tensor(GOLDEN, dtype=torch.int32),
    )


def test_from_mask_inverts_to_same_tuples():
    m2 = AttnMask.from_mask(torch.tensor(GOLDEN, dtype=torch.int32))
    assert m2.q_ranges == AttnRanges.from_ranges(Q_RANGES)
    assert m2.k_ranges == AttnRanges.from_ranges(K_RANGES)
    assert m2.attn_mask_type == TYPES
    assert m2 == AttnMask.from_mask(torch.tensor(GOLDEN, dtype=torch.int32))


def test_direct_init_forbidden():
    with pytest.raises(RuntimeError):
        AttnMask(
            mask_tensor=torch.zeros(4, 4, 1, dtype=torch.int32),
            q_ranges=AttnRanges.from_ranges([(0, 4)]),
            k_ranges=AttnRanges.from_ranges([(0, 4)]),
            attn_mask_type=[AttnMaskType.FULL],
            total_seqlen_q=4,
            total_seqlen_k=4,
        )


def test_sub_mask_and_area():
    m = _build()
    assert m.area == int(np.array(GOLDEN).sum())
    qr, kr = AttnRange(0, 9), AttnRange(0, 8)
    sub = m.make_sub_mask(qr, kr)
    want = np.array(GOLDEN)[0:9, 0:8]
    assert np.equal(sub.mask_flag_array, want).all()
    assert m.calc_sub_area(qr, kr) == int(want.sum())
    # sub-mask re-inference round-trips to the same dense matrix
    re = AttnMask.from_ranges(
        sub.q_ranges, sub.k_ranges, sub.attn_mask_type, 9, 8
    )
    assert np.equal(re.mask_flag_array, want).all()


def test_classification_properties():
    full = AttnMask.from_ranges(
        AttnRanges.from_ranges([(0, 8)]), AttnRanges.from_ranges([(0, 8)]),
        [AttnMaskType.FULL],
    )
    assert full.is_pure_full() and full.is_square() and not full.is_empty()
    causal = AttnMask.from_ranges(
        AttnRanges.from_ranges([(0, 8)]), AttnRanges.from_ranges([(0, 8)]),
        [AttnMaskType.CAUSAL],
    )
    assert causal.is_pure_causal() and not causal.is_pure_full()
    varlen = AttnMask.from_ranges(
        AttnRanges.from_ranges([(0, 4), (4, 8)]),
        AttnRanges.from_ranges([(0, 4), (4, 8)]),
        [AttnMaskType.CAUSAL] * 2,
    )
    assert varlen.is_varlen_causal() and not varlen.is_varlen_full()
    assert AttnMask.make_causal_mask(3, 5)[0].tolist() == [1, 1, 1, 0, 0]
    assert AttnMask.make_causal_mask(3, 5, align="top-left")[0].tolist() == [
        1, 0, 0, 0, 0,
    ]


def test_from_ranges_agrees_with_oracle_dense_mask():
    """Cross-check AttnMask.from_ranges against the (independently written)
    oracle dense-mask builder for FULL/CAUSAL blocks on random inputs with
    non-overlapping q rows."""
    from oracle import make_attn_mask

    rng = np.random.default_rng(3)
    for _ in range(20):
        n = int(rng.integers(4, 24))
        cuts = sorted(rng.choice(np.arange(1, n), size=2, replace=False))
        qrs = [(0, int(cuts[0])), (int(cuts[0]), int(cuts[1])), (int(cuts[1]), n)]
        krs = []
        tts = []
        for _q in qrs:
            a = int(rng.integers(0, n))
            b = int(rng.integers(a + 1, n + 1))
            krs.append((a, b))
            tts.append(int(rng.integers(0, 2)))  # 0 full / 1 causal
        m = AttnMask.from_ranges(
            AttnRanges.from_ranges(qrs), AttnRanges.from_ranges(krs),
            [AttnMaskType.FULL if t == 0 else AttnMaskType.CAUSAL for t in tts],
            n, n,
        )
        want = make_attn_mask(n, n, qrs, krs, tts)
        assert torch.equal(
            torch.from_numpy(m.mask_flag_array).bool(), want
        ), (qrs, krs, tts)


def test_from_mask_roundtrip_random():
    """from_ranges -> dense -> from_mask -> dense must reproduce the same
    matrix (the inferred tuples may be a different, canonical partition)."""
    rng = np.random.default_rng(7)
    for _ in range(20):
        n = int(rng.integers(3, 20))
        k = int(rng.integers(1, 4))
        starts = sorted(rng.choice(np.arange(0, n), size=k, replace=False))
        qrs, krs, tts = [], [], []
        prev = 0
        for s in list(starts[1:]) + [n]:
            a = int(rng.integers(0, n))
            b = int(rng.integers(a, n)) + 1
            qrs.append((prev, s))
            krs.append((a, b))
            tts.append(
                AttnMaskType.FULL if rng.integers(0, 2) == 0 else AttnMaskType.CAUSAL
            )
            prev = s
        if qrs[0][0] == qrs[0][1]:
            continue
        m = AttnMask.from_ranges(
            AttnRanges.from_ranges(qrs), AttnRanges.from_ranges(krs), tts, n, n
        )
        dense = m.mask_tensor[..., 0]
        try:
            m2 = AttnMask.from_mask(dense)
        except AssertionError:
            continue  # overlapping causal overwrite can break row contiguity
        assert np.equal(m2.mask_flag_array, m.mask_flag_array).all()


def test_sub_mask_reference_golden():
    """The reference's own test_make_sub_mask_with_calc_sub_area expectations
    (tests/test_common/test_attn_mask.py:132): exact sub-areas AND the exact
    canonical tuples from_mask infers on each sub-window."""
    m = AttnMask.from_ranges(
        AttnRanges.from_ranges([(0, 6), (6, 9), (9, 12), (12, 16)]),
        AttnRanges.from_ranges([(0, 4), (4, 12), (12, 15), (1, 13)]),
        [AttnMaskType.CAUSAL, AttnMaskType.FULL,
         AttnMaskType.CAUSAL, AttnMaskType.CAUSAL],
        16, 16,
    )
    assert m.area == 82

    s1 = m.make_sub_mask(AttnRange(4, 13), AttnRange(1, 13))
    assert m.calc_sub_area(AttnRange(4, 13), AttnRange(1, 13)) == 41
    assert s1.area == 41
    assert s1.q_ranges == AttnRanges.from_ranges([[0, 2], [2, 5], [5, 8], [8, 9]])
    assert s1.k_ranges == AttnRanges.from_ranges([[0, 3], [3, 11], [11, 12], [0, 9]])
    assert s1.attn_mask_type == [
        AttnMaskType.CAUSAL, AttnMaskType.FULL, AttnMaskType.FULL,
        AttnMaskType.FULL,
    ]

    s2 = m.make_sub_mask(AttnRange(0, 14), AttnRange(0, 7))
    assert m.calc_sub_area(AttnRange(0, 14), AttnRange(0, 7)) == 31
    assert s2.area == 31
    assert s2.q_ranges == AttnRanges.from_ranges([[0, 6], [6, 9], [9, 12], [12, 14]])
    assert s2.k_ranges == AttnRanges.from_ranges([[0, 4], [4, 7], [9, 9], [1, 7]])
    assert s2.attn_mask_type == [
        AttnMaskType.CAUSAL, AttnMaskType.FULL, AttnMaskType.CAUSAL,
        AttnMaskType.FULL,
    ]

    s3 = m.make_sub_mask(AttnRange(5, 16), AttnRange(3, 11))
    assert m.calc_sub_area(AttnRange(5, 16), AttnRange(3, 11)) == 53
    assert s3.area == 53
    assert s3.q_ranges == AttnRanges.from_ranges(
        [[0, 1], [1, 4], [4, 7], [7, 9], [9, 11]]
    )
    assert s3.k_ranges == AttnRanges.from_ranges(
        [[0, 1], [1, 8], [4, 4], [0, 8], [0, 8]]
    )
    assert s3.attn_mask_type == [
        AttnMaskType.FULL, AttnMaskType.FULL, AttnMaskType.CAUSAL,
        AttnMaskType.CAUSAL, AttnMaskType.FULL,
    ]

    assert m.calc_sub_area(AttnRange(4, 12), AttnRange(4, 12)) == 24
    assert m.make_sub_mask(AttnRange(4, 12), AttnRange(4, 12)).area == 24
