"""CPU checks of the native boundary: the C-ABI library loads and exports
every symbol include/magi_ffa.h declares; the magi_attn_ext surface module
works (host-side parts)."""
import ctypes
import re
from pathlib import Path

import torch

ROOT = Path(__file__).resolve().parent.parent


def test_cabi_exports_all_declared_symbols():
    header = (ROOT / "include" / "magi_ffa.h").read_text()
    declared = re.findall(r"^int (magi_\w+)\(", header, re.M)
    assert len(declared) >= 8, declared
    lib = ctypes.CDLL(str(ROOT / "magi_attention" / "_libs" / "libmagi_ffa.so"))
    for sym in declared:
        assert getattr(lib, sym, None) is not None, f"missing symbol {sym}"


def test_ext_module_surface_cpu():
    from magi_attention import magi_attn_ext as ext

    r = torch.tensor([[5, 9], [0, 3], [5, 9], [2, 4]], dtype=torch.int32)
    order = ext.argsort_ranges(r)
    assert order.tolist() == [1, 3, 0, 2]
    q2, k2, t2 = ext.reorder_ranges_and_attn_type_maps(
        r, r + 1, torch.tensor([0, 1, 2, 3], dtype=torch.int32), order
    )
    assert q2[:, 0].tolist() == [0, 2, 5, 5]
    assert t2.tolist() == [1, 3, 0, 2]

    sorted_r = q2
    uniq, inv, cnt = ext.unique_consecutive_pairs(sorted_r)
    assert int(cnt[0]) == uniq.shape[0]
    assert torch.equal(uniq.long().index_select(0, inv.long()), sorted_r.long())

    assert ext.is_valid_cu_seqlens([0, 3, 7], 7)
    assert not ext.is_valid_cu_seqlens([1, 3], 3)

    ext.start_event("x")
    ext.stop_event("x")
    assert ext.elapsed_ms("x") >= 0.0
    ext.destroy_event("x")


def test_product_fails_loudly_without_extension(monkeypatch):
    """The compute path must never fall back silently when the HIP library is
    absent (tier rule: no CPU fallback on the product path)."""
    import magi_attention._ffa_lib as L

    monkeypatch.setattr(L, "_lib", None)
    monkeypatch.setattr(L, "_LIB_PATH", Path("/nonexistent/lib.so"))
    try:
        L.lib()
        assert False, "expected RuntimeError"
    except RuntimeError as e:
        assert "no fallback" in str(e).lower() or "native" in str(e).lower()


def test_cross_attn_raises_like_reference():
    """is_same_source=False: the reference raises NotImplementedError for
    every cross-attn case (_make_dispatch_meta.py:203-214) — so do we."""
    import pytest as _pytest

    from magi_attention.api import magi_attn_flex_key
    from magi_attention.common.ranges import AttnRanges

    for qp, kp, msg in [
        (True, False, "encoder-decoder"),
        (False, True, "multi-modal"),
        (True, True, "pure cross-attn"),
        (False, False, "trivial"),
    ]:
        with _pytest.raises(NotImplementedError, match=msg):
            magi_attn_flex_key(
                AttnRanges.from_ranges([[0, 128]]),
                AttnRanges.from_ranges([[0, 64]]),
                [0], 128, 64, 4, 2, 32,
                is_same_source=False, is_q_permutable=qp, is_k_permutable=kp,
            )


def test_correct_attn_helpers_match_oracle_merge():
    """functional.utils correct_attn_* vs the oracle's merge_out_lse."""
    import torch

    from magi_attention.functional import (
        correct_attn_lse,
        correct_attn_out_lse,
        correct_attn_out_lse_with_sink,
    )
    from oracle.ref_attn import merge_out_lse

    g = torch.Generator().manual_seed(3)
    o1 = torch.randn(32, 4, 16, generator=g)
    o2 = torch.randn(32, 4, 16, generator=g)
    l1 = torch.randn(32, 4, generator=g)
    l2 = torch.randn(32, 4, generator=g)
    l1[5] = float("-inf")  # empty-row case
    l2[7] = float("-inf")
    l1[9] = l2[9] = float("-inf")
    out, lse = correct_attn_out_lse(o1, l1, o2, l2)
    ref_o, ref_l = merge_out_lse([o1, o2], [l1, l2])
    torch.testing.assert_close(out.double(), ref_o, atol=1e-5, rtol=1e-5)
    torch.testing.assert_close(
        correct_attn_lse(l1, l2).double(), ref_l, atol=1e-5, rtol=1e-5
    )
    torch.testing.assert_close(lse.double(), ref_l, atol=1e-5, rtol=1e-5)

    # sink fold == appending the sink columns in the oracle
    from oracle import ref_attn

    q = torch.randn(32, 4, 16, generator=g).double()
    k = torch.randn(32, 4, 16, generator=g).double()
    v = torch.randn(32, 4, 16, generator=g).double()
    sink = torch.randn(2, 4, generator=g).float()
    mask = torch.ones(32, 32, dtype=torch.bool)
    o_ns, l_ns = ref_attn(q, k, v, mask)
    o_s, l_s = ref_attn(q, k, v, mask, sink=sink)
    o_f, l_f = correct_attn_out_lse_with_sink(o_ns.float(), l_ns, sink)
    torch.testing.assert_close(o_f.double(), o_s, atol=1e-5, rtol=1e-5)
    torch.testing.assert_close(l_f, l_s, atol=1e-5, rtol=1e-5)


def test_cabi_validation_codes_cpu():
    """The launchers' argument-validation paths return their documented
    negative codes BEFORE any GPU work — checkable in the CPU container
    (include/magi_ffa.h conventions; no kernel is launched)."""
    import ctypes

    import pytest

    from magi_attention import _ffa_lib
    from magi_attention._ffa_lib import MagiFfaFwdArgs, MagiFfaIndexArgs

    if not _ffa_lib.is_available():
        pytest.skip("native library not built")
    lib = _ffa_lib.lib()

    # index launcher
    a = MagiFfaIndexArgs(
        q=1, k=1, v=1, out=1, lse=1, indices_2d=1,
        total_q=8, total_k=8, max_topk=64, hq=4, hk=2, d=128,
        softmax_scale=1.0, softcap=0.0, out_is_fp32=0, stream=None,
    )
    assert lib.magi_ffa_fwd_index(ctypes.byref(a)) == -3  # hk must be 1
    a.hk = 1
    a.d = 96
    assert lib.magi_ffa_fwd_index(ctypes.byref(a)) == -2  # d bucket
    a.d = 128
    a.max_topk = 50
    assert lib.magi_ffa_fwd_index(ctypes.byref(a)) == -4  # 64 | max_topk
    a.max_topk = 64
    a.indices_2d = None
    assert lib.magi_ffa_fwd_index(ctypes.byref(a)) == -1  # null pointer

    # fwd launcher
    f = MagiFfaFwdArgs(
        q=1, k=1, v=1, out=1, lse=1, q_ranges=1, k_ranges=1,
        attn_type_map=None, locks=None, max_logits=None, qk_starts=None,
        n_ranges=1, total_q=8, total_k=8, hq=4, hk=2, d=96, max_seqlen_q=8,
        softmax_scale=1.0, softcap=0.0, out_is_fp32=1,
        disable_atomic_reduction=1, cu_margin=0, stream=None,
    )
    assert lib.magi_ffa_fwd(ctypes.byref(f)) == -2  # d bucket
    f.d = 128
    f.hq = 3  # not divisible by hk=2
    assert lib.magi_ffa_fwd(ctypes.byref(f)) == -3
    f.hq = 4
    f.disable_atomic_reduction = 0  # atomic merge without locks
    assert lib.magi_ffa_fwd(ctypes.byref(f)) == -4
    f.q = None
    assert lib.magi_ffa_fwd(ctypes.byref(f)) == -1
