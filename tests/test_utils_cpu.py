"""CPU tests for magi_attention.utils helpers (reference utils/_utils.py
surface) against brute-force restatements."""
import numpy as np
import pytest
import torch

from magi_attention.utils import (
    argmax,
    argmin,
    argsort,
    ceil_div,
    cu_seqlens2seqlens,
    flatten_nested_list,
    get_factors,
    is_list_type_all,
    is_list_value_all,
    is_list_value_any,
    make_attn_mask_from_ffa_args,
    make_slice_mask_from_ffa_attn_type,
    max_fp_dtype,
    pad_and_pack_tensors,
    perm_idxs2unperm_idxs,
    seqlens2cu_seqlens,
    set_random_seed,
    str2seed,
    to_higher_fp_dtype,
    transpose_matrix,
    wrap_to_list,
)
from magi_attention.utils.metaclass import SingletonMeta


def test_small_helpers():
    assert ceil_div(7, 3) == 3 and ceil_div(6, 3) == 2
    assert seqlens2cu_seqlens([2, 3, 5]) == [0, 2, 5, 10]
    assert cu_seqlens2seqlens([0, 2, 5, 10]) == [2, 3, 5]
    assert flatten_nested_list([1, [2, (3, 4)], [[5]]]) == [1, 2, 3, 4, 5]
    perm = [2, 0, 3, 1]
    inv = perm_idxs2unperm_idxs(perm)
    assert [perm[i] for i in inv] == [0, 1, 2, 3]
    assert wrap_to_list(5, 3) == [5, 5, 5] and wrap_to_list((1, 2)) == [1, 2]
    assert get_factors(12) == [1, 2, 3, 4, 6, 12]
    assert transpose_matrix([[1, 2], [3, 4], [5, 6]]) == [[1, 3, 5], [2, 4, 6]]
    assert argmin([3, 1, 2]) == 1 and argmax([3, 1, 2]) == 0
    assert argsort([3, 1, 2]) == [1, 2, 0]


def test_list_predicates():
    assert is_list_value_all([1, 1, 1], 1)
    assert not is_list_value_all([1, 2], 1)
    assert is_list_value_all([7, 7], just_same=True)
    assert is_list_value_all([], allow_empty=True)
    assert is_list_value_any([0, 1, 0], 1)
    assert is_list_type_all([1, 2], int) and not is_list_type_all([1, "a"], int)
    assert is_list_type_all(["a", "b"], just_same=True)


def test_dtype_helpers():
    assert max_fp_dtype(torch.bfloat16, torch.float32) == torch.float32
    t = torch.zeros(2, dtype=torch.bfloat16)
    assert to_higher_fp_dtype(t, torch.float32).dtype == torch.float32
    assert to_higher_fp_dtype(t.float(), torch.bfloat16).dtype == torch.float32


def test_pad_and_pack():
    out = pad_and_pack_tensors(
        [torch.tensor([1.0, 2.0]), torch.tensor([3.0])], 3, padding_value=-1
    )
    assert out.tolist() == [[1, 2, -1], [3, -1, -1]]
    with pytest.raises(ValueError):
        pad_and_pack_tensors([torch.ones(5)], 3)
    with pytest.raises(ValueError):
        pad_and_pack_tensors([torch.ones(2, 2)], 4)


def test_slice_masks_match_oracle():
    from oracle import make_attn_mask

    rng = np.random.default_rng(11)
    for _ in range(10):
        sq, sk = int(rng.integers(1, 12)), int(rng.integers(1, 12))
        for t in range(4):
            got = make_slice_mask_from_ffa_attn_type(sq, sk, t, device="cpu")
            want = make_attn_mask(sq, sk, [[0, sq]], [[0, sk]], [t])
            assert torch.equal(got, want), (sq, sk, t)


def test_full_mask_from_ffa_args():
    from magi_attention.common import AttnRanges

    qrs = AttnRanges.from_ranges([(0, 4), (4, 10)])
    krs = AttnRanges.from_ranges([(0, 6), (2, 10)])
    m = make_attn_mask_from_ffa_args(qrs, krs, [1, 3], 10, 10, device="cpu")
    from oracle import make_attn_mask

    want = make_attn_mask(10, 10, [[0, 4], [4, 10]], [[0, 6], [2, 10]], [1, 3])
    assert torch.equal(m, want)


def test_seed_helpers():
    s = str2seed("magi")
    assert 0 <= s < 2**32 and s == str2seed("magi") != str2seed("attention")
    set_random_seed(123)
    a = torch.randn(4)
    set_random_seed(123)
    assert torch.equal(a, torch.randn(4))


def test_singleton_meta():
    class A(metaclass=SingletonMeta):
        def __init__(self):
            self.v = object()

    class B(metaclass=SingletonMeta):
        pass

    assert A() is A() and B() is B() and A() is not B()


def test_magi_attn_ext_surface_aliases():
    """The ext module exposes the reference pyi's type aliases + helpers
    (magi_attn_ext.pyi: AttnMaskType/AttnRectangle(s)/expand_attn_ranges/
    elapsed_ms_event/produce)."""
    import magi_attention.magi_attn_ext as ext
    from magi_attention.common import AttnRanges

    assert ext.AttnMaskType is not None and ext.AttnRectangles is not None
    rr = AttnRanges.from_ranges([(0, 3), (5, 8)])
    out = ext.expand_attn_ranges(rr, stride=10, num_heads_group=2)
    assert [(r.start, r.end) for r in out] == [
        (0, 3), (5, 8), (10, 13), (15, 18)
    ]
    ext.start_event("t")
    ext.stop_event("t")
    assert ext.elapsed_ms_event("t") >= 0.0
    ext.destroy_event("t")
    ext.produce(None)  # no-op contract
