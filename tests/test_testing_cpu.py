"""CPU tests for the magi_attention.testing harness surface."""
import numpy as np
import pytest
import torch

from magi_attention.testing import (
    GroundTruthDispatcher,
    assert_close,
    parameterize,
    ref_attn_func,
)
from magi_attention.testing.flag_generator import FlagCombGenerator
from magi_attention.testing.precision import (
    extract_mismatch_info,
    extract_mismatch_threshold,
)
from magi_attention.testing.utils import (
    switch_envvar_context,
    switch_envvars,
)


def test_ref_attn_matches_oracle():
    """ref_attn_func (dense mask) vs the fp64 oracle over random flex
    masks, incl. GQA and empty rows."""
    from oracle import make_attn_mask, ref_attn

    rng = np.random.default_rng(4)
    for _ in range(8):
        tq, tk = int(rng.integers(4, 24)), int(rng.integers(4, 24))
        hq, hk = 4, int(rng.choice([1, 2, 4]))
        d = 16
        qrs = [[0, tq // 2], [tq // 2, tq]]
        krs = [[0, tk], [0, max(1, tk // 2)]]
        tts = [int(rng.integers(0, 2)), int(rng.integers(0, 2))]
        mask = make_attn_mask(tq, tk, qrs, krs, tts)
        q = torch.randn(tq, hq, d, dtype=torch.float64)
        k = torch.randn(tk, hk, d, dtype=torch.float64)
        v = torch.randn_like(k)
        out, meta = ref_attn_func(q, k, v, mask, return_lse=True)
        want_out, want_lse = ref_attn(q, k, v, mask, d ** -0.5)
        assert (out - want_out).abs().max().item() < 1e-12
        fin = want_lse.isfinite()
        # the oracle reports lse in fp32 — compare at fp32 resolution
        assert (meta.lse[fin] - want_lse[fin].double()).abs().max().item() < 1e-6
        assert bool((meta.lse.isfinite() == fin).all())


def test_ref_attn_sink():
    """Sink columns join the normalization only (reference sink semantics:
    out scaled by the sink-inclusive denominator, lse includes sinks)."""
    torch.manual_seed(0)
    tq, tk, h, d, ss = 6, 8, 2, 16, 2
    q = torch.randn(tq, h, d, dtype=torch.float64)
    k = torch.randn(tk, h, d, dtype=torch.float64)
    v = torch.randn_like(k)
    mask = torch.ones(tq, tk, dtype=torch.bool)
    sink = torch.randn(ss, h, dtype=torch.float64)
    out, meta = ref_attn_func(q, k, v, mask, sink=sink, return_lse=True)
    s = torch.einsum("qhd,khd->hqk", q, k) * d ** -0.5
    s_all = torch.cat([s, sink.t().unsqueeze(1).expand(h, tq, ss)], -1)
    p = torch.softmax(s_all, -1)[..., :tk]
    want = torch.einsum("hqk,khd->qhd", p, v)
    assert (out - want).abs().max().item() < 1e-12
    want_lse = torch.logsumexp(s_all, -1).t()
    assert (meta.lse - want_lse).abs().max().item() < 1e-12


def test_assert_close_mismatch_budget():
    a = torch.zeros(100)
    b = torch.zeros(100)
    b[:3] = 1.0  # 3% mismatch
    with pytest.raises(AssertionError):
        assert_close(a, b, test_case="strict")
    assert_close(a, b, mismatch_threshold=0.05, test_case="budgeted")
    with pytest.raises(AssertionError):
        assert_close(a, b, mismatch_threshold=0.01, test_case="too tight")
    m, t, r = extract_mismatch_info("Mismatched elements: 3 / 100 (3.0%)")
    assert (m, t, r) == (3, 100, 0.03)
    thr = extract_mismatch_threshold(b, a, atol=1e-5, rtol=1e-5,
                                     mismatch_thres_ratio=2.0)
    assert abs(thr - 0.06) < 1e-9


def test_parameterize_stacks_and_reports():
    seen = []

    @parameterize("x", [1, 2])
    @parameterize("y", ["a", "b"])
    def case(x, y):
        seen.append((x, y))
        if (x, y) == (2, "b"):
            raise ValueError("boom")

    with pytest.raises(ValueError) as ei:
        case()
    assert seen == [(1, "a"), (1, "b"), (2, "a"), (2, "b")]
    assert "x[1]=2" in str(ei.value).replace(" ", "").replace("\n", "") or "boom" in str(ei.value)


def test_flag_generator_heuristic():
    import itertools
    import random

    random.seed(0)
    gen = FlagCombGenerator(
        flags=["a", "b", "c"],
        options={"c": [0, 1, 2]},
        cycle_times=1,
    )
    combs = list(gen)
    # defaults first, then the all-non-default corner; the random fill draws
    # with replacement (reference behavior) but must COVER the whole space
    assert combs[0] == {"a": False, "b": False, "c": 0}
    assert combs[1] == {"a": True, "b": True, "c": 2}
    keys = {tuple(c.values()) for c in combs}
    assert keys == set(itertools.product([False, True], [False, True], [0, 1, 2]))
    assert gen.num_combs == 12

    gen2 = FlagCombGenerator(flags=["a", "b"], strategy="sequential",
                             cycle_times=1)
    assert [tuple(c.values()) for c in gen2] == [
        (False, False), (False, True), (True, False), (True, True)
    ]
    # get_next_valid_comb defers illegal draws
    gen3 = FlagCombGenerator(flags=["a", "b"], strategy="sequential",
                             cycle_times=1)
    c = gen3.get_next_valid_comb({}, lambda comb, cfg: comb["a"])
    assert c["a"] is True and len(gen3._deferred_combs) == 2


def test_env_switchers():
    import os

    with switch_envvar_context("MAGI_TEST_FLAG_X"):
        assert os.environ["MAGI_TEST_FLAG_X"] == "1"
    assert "MAGI_TEST_FLAG_X" not in os.environ
    back = switch_envvars(["MAGI_TEST_FLAG_X", "MAGI_TEST_FLAG_Y"],
                          enable_dict={"MAGI_TEST_FLAG_Y": False})
    assert os.environ["MAGI_TEST_FLAG_X"] == "1"
    assert os.environ["MAGI_TEST_FLAG_Y"] == "0"
    back()
    assert "MAGI_TEST_FLAG_X" not in os.environ


def test_gt_dispatcher_matches_bucket_factory():
    """The dense-mask ground truth must agree with the arithmetic bucket
    factory on chunk areas."""
    from magi_attention.common import AttnRanges
    from magi_attention.common.enum import AttnMaskType
    from magi_attention.config import DispatchAlg
    from magi_attention.meta import make_global_bucket_from_qk_ranges

    qrs = AttnRanges.from_ranges([(0, 8), (8, 16)])
    krs = AttnRanges.from_ranges([(0, 8), (4, 16)])
    tts = [AttnMaskType.CAUSAL, AttnMaskType.CAUSAL]
    gt = GroundTruthDispatcher(alg=DispatchAlg())
    bucket_gt = gt._compute_self_attn_areas(qrs, krs, tts, chunk_size=4)
    bucket = make_global_bucket_from_qk_ranges(
        qrs, krs, tts, num_chunks=4, chunk_size=4
    )
    assert bucket_gt.areas == bucket.areas
    assert bucket_gt.area == bucket.area


def test_ref_attn_func_matches_reference_golden_vectors():
    """testing.ref_attn_func vs the committed outputs of the REFERENCE's own
    ref_attn_func (tests/golden/golden_attn.pt, generated in-container from
    /root/reference by generate_from_reference.py) — the same pin the oracle
    carries, closing the loop on the harness's reference implementation."""
    from pathlib import Path

    golden = torch.load(
        Path(__file__).parent / "golden" / "golden_attn.pt", weights_only=False
    )
    checked = 0
    for case in golden:
        if case.get("softcap", 0.0):
            continue  # ref_attn_func rejects softcap (reference parity)
        q, k, v = case["q"].double(), case["k"].double(), case["v"].double()
        mask = case["mask"]
        out, meta = ref_attn_func(
            q, k, v, mask,
            softmax_scale=case["softmax_scale"],
            high_precision=True, return_lse=True,
        )
        ref_out, ref_lse = case["out"].double(), case["lse"]
        # golden tensors are stored at the reference run's fp32/bf16 output
        # precision — compare at that resolution
        err = (out - ref_out).abs().max().item()
        tol = 5e-3 if case["out"].dtype == torch.bfloat16 else 1e-6
        assert err < tol, (case["name"], err)
        fin = ref_lse.isfinite()
        lse_err = (meta.lse[fin].float() - ref_lse[fin]).abs().max().item()
        assert lse_err < 1e-5, (case["name"], lse_err)
        assert bool((meta.lse.isfinite() == fin).all()), case["name"]
        checked += 1
    assert checked >= 10
