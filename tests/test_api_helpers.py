"""Unit tests for api.functools helpers and the runtime-dict manager
(regressions for ADVICE r1 medium/low)."""
import torch

from magi_attention.api import infer_attn_mask_from_sliding_window
from magi_attention.common.range import AttnRange
from oracle import make_attn_mask


def _dense(tq, tk, qr, kr, tt):
    return make_attn_mask(
        tq, tk,
        [[r.start, r.end] for r in qr],
        [[r.start, r.end] for r in kr],
        [t.to_int_type() for t in tt],
    )


def _dense_window_ref(qs, qe, ks, ke, window):
    """Brute-force bottom-right-aligned sliding window with the reference's
    q-trim (functools.py:218-224): only the last (ke-ks) q rows participate."""
    tq, tk = qe, ke
    m = torch.zeros(tq, tk, dtype=torch.bool)
    sk = ke - ks
    qs_t = max(qs, qe - sk)
    left, right = window
    for qi in range(qs_t, qe):
        # bottom-right aligned diagonal: row qi pairs with key ke-(qe-qi)
        diag = ke - (qe - qi)
        lo = ks if left == -1 else max(ks, diag - left)
        hi = ke if right == -1 else min(ke, diag + right + 1)
        if lo < hi:
            m[qi, lo:hi] = True
    return m


def test_sliding_window_q_longer_than_k():
    # q_range longer than k_range must trim to the last
    # k_range.seqlen rows before the window shortcut applies.
    for window in [(-1, 3), (3, -1), (-1, -1), (2, 1), (0, 0)]:
        qr, kr, tt = infer_attn_mask_from_sliding_window(
            AttnRange(0, 10), AttnRange(0, 4), window
        )
        got = _dense(10, 4, qr, kr, tt)
        want = _dense_window_ref(0, 10, 0, 4, window)
        assert torch.equal(got, want), (window, got.int(), want.int())


def test_sliding_window_square_and_wide():
    for (qs, qe, ks, ke) in [(0, 8, 0, 8), (4, 12, 0, 16), (0, 6, 0, 10)]:
        for window in [(-1, 0), (0, -1), (2, 3), (-1, -1), (5, 0), (15, 15)]:
            qr, kr, tt = infer_attn_mask_from_sliding_window(
                AttnRange(qs, qe), AttnRange(ks, ke), window
            )
            got = _dense(qe, ke, qr, kr, tt)
            want = _dense_window_ref(qs, qe, ks, ke, window)
            assert torch.equal(got, want), (qs, qe, ks, ke, window)


def test_most_recent_key_per_group():
    # get_most_recent_key must honour its cp_group argument.
    from magi_attention.api.magi_attn_interface import DistAttnRuntimeDictManager
    from magi_attention.dist_attn_runtime_mgr import DistAttnRuntimeKey

    def key(tag, seq):
        return DistAttnRuntimeKey(
            q_ranges=((0, seq),), k_ranges=((0, seq),), attn_mask_type=(1,),
            total_seqlen_q=seq, total_seqlen_k=seq, pad_size=0, chunk_size=64,
            num_heads_q=1, num_heads_kv=1, head_dim=8,
            cp_group_tag=tag, config_tag="c", env_flags=(),
        )

    mgr = DistAttnRuntimeDictManager()
    ka = key((0, 1), 128)
    kb = key((2, 3), 256)
    mgr[ka] = "A"
    mgr[kb] = "B"
    # globally most recent = last touched
    assert mgr.get_most_recent_key() is kb
    _ = mgr[ka]
    assert mgr.get_most_recent_key() is ka

    # per-group filter via a fake group object routed through _group_tag
    class FakeGroup:
        pass

    import magi_attention.dist_attn_runtime_mgr as m

    orig = m._group_tag
    m._group_tag = lambda g, _o=orig: (2, 3) if isinstance(g, FakeGroup) else _o(g)
    try:
        assert mgr.get_most_recent_key(FakeGroup()) is kb
        mgr.clear(FakeGroup())
        assert mgr.get_most_recent_key(FakeGroup()) is None
        assert mgr.get_most_recent_key() is ka
    finally:
        m._group_tag = orig
