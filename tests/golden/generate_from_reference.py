#!/usr/bin/env python3
"""Generate golden parity vectors from the REFERENCE's own pure-PyTorch oracle
(`magi_attention/testing/ref_attn.py:638 ref_attn_func`), run on CPU in the
build container where /root/reference is mounted.

This script is committed for provenance; the fixtures it writes
(tests/golden/golden_attn.pt) are committed too, because /root/reference does
NOT exist on the GPU box. Run (container only):

    python tests/golden/generate_from_reference.py

It stubs three modules the reference hard-imports (debugpy, expecttest,
magi_attention.magi_attn_ext) exactly as documented in BASELINE.md.
"""
import sys
import types
from pathlib import Path

import torch

REF = "/root/reference"


def _install_stubs():
    # fake debugpy (imported unconditionally by magi_attention/utils/debug.py:17)
    dbg = types.ModuleType("debugpy")
    dbg.listen = lambda *a, **k: None
    dbg.wait_for_client = lambda *a, **k: None
    dbg.breakpoint = lambda *a, **k: None
    sys.modules.setdefault("debugpy", dbg)
    # fake expecttest (pulled in via torch.testing._internal)
    exp = types.ModuleType("expecttest")
    exp.TestCase = object
    exp.assert_expected_inline = lambda *a, **k: None
    sys.modules.setdefault("expecttest", exp)
    sys.path.insert(0, REF)
    # fake magi_attention.magi_attn_ext (hard-imported by _flex_flash_attn_jit.py:31)
    import importlib

    spec = importlib.util.find_spec("magi_attention")
    assert spec is not None, "reference not found at /root/reference"
    ext = types.ModuleType("magi_attention.magi_attn_ext")

    def _make_dummy(name):
        return type(
            name,
            (),
            {
                "__init__": lambda self, *a, **k: None,
                "__getattr__": lambda self, n: (lambda *a, **k: None),
            },
        )

    def _getattr(name):
        if name.startswith("__"):
            raise AttributeError(name)
        return _make_dummy(name)

    ext.__file__ = "<stub>"
    ext.__getattr__ = _getattr  # type: ignore[attr-defined]
    sys.modules["magi_attention.magi_attn_ext"] = ext


def _cases():
    g = torch.Generator().manual_seed(42)

    def mk(tq, tk, hq, hk, d, q_ranges, k_ranges, types, name, scale=None, softcap=0.0):
        q = (torch.randn(tq, hq, d, generator=g, dtype=torch.float32) * 0.5)
        k = (torch.randn(tk, hk, d, generator=g, dtype=torch.float32) * 0.5)
        v = (torch.randn(tk, hk, d, generator=g, dtype=torch.float32) * 0.5)
        do = (torch.randn(tq, hq, d, generator=g, dtype=torch.float32) * 0.5)
        return dict(
            name=name, q=q, k=k, v=v, dout=do,
            q_ranges=torch.tensor(q_ranges, dtype=torch.int32),
            k_ranges=torch.tensor(k_ranges, dtype=torch.int32),
            attn_types=torch.tensor(types, dtype=torch.int32),
            softmax_scale=scale, softcap=softcap,
        )

    cases = [
        # simple full
        mk(16, 16, 2, 2, 32, [[0, 16]], [[0, 16]], [0], "full_16"),
        # causal square
        mk(32, 32, 2, 1, 64, [[0, 32]], [[0, 32]], [1], "causal_32_gqa"),
        # causal rectangular both directions (bottom-right alignment check)
        mk(5, 2, 1, 1, 16, [[0, 5]], [[0, 2]], [1], "causal_5x2"),
        mk(2, 5, 1, 1, 16, [[0, 2]], [[0, 5]], [1], "causal_2x5"),
        # inv-causal rectangular
        mk(5, 2, 1, 1, 16, [[0, 5]], [[0, 2]], [2], "invcausal_5x2"),
        mk(2, 5, 1, 1, 16, [[0, 2]], [[0, 5]], [2], "invcausal_2x5"),
        # bi-causal
        mk(2, 5, 1, 1, 16, [[0, 2]], [[0, 5]], [3], "bicausal_2x5"),
        mk(5, 5, 2, 2, 32, [[0, 5]], [[0, 5]], [3], "bicausal_5x5"),
        # varlen doc mask: 3 causal docs
        mk(
            48, 48, 4, 2, 64,
            [[0, 16], [16, 40], [40, 48]],
            [[0, 16], [16, 40], [40, 48]],
            [1, 1, 1],
            "varlen_causal_3docs",
        ),
        # overlapping q_ranges (atomic-merge path) + mixed types
        mk(
            32, 48, 2, 2, 32,
            [[0, 32], [8, 24], [0, 16]],
            [[0, 16], [16, 40], [40, 48]],
            [1, 0, 2],
            "overlap_q_mixed",
        ),
        # empty rows (some q rows with no allowed k)
        mk(16, 8, 1, 1, 32, [[0, 8]], [[0, 8]], [1], "empty_rows_tail"),
        # custom scale (reference oracle does not support softcap; softcap parity
        # is pinned by the kernel-formula restatement in oracle/ref_attn.py)
        mk(24, 24, 2, 2, 32, [[0, 24]], [[0, 24]], [1], "custom_scale", scale=0.2),
    ]
    return cases


def main():
    _install_stubs()
    from magi_attention.common.ranges import AttnRanges
    from magi_attention.testing.ref_attn import ref_attn_func  # noqa: reference import
    from magi_attention.utils._utils import make_attn_mask_from_ffa_args

    out_path = Path(__file__).parent / "golden_attn.pt"
    fixtures = []
    for c in _cases():
        mask = make_attn_mask_from_ffa_args(
            q_ranges=AttnRanges.from_ranges(c["q_ranges"].tolist()),
            k_ranges=AttnRanges.from_ranges(c["k_ranges"].tolist()),
            attn_type_map=c["attn_types"].tolist(),
            total_seqlen_q=c["q"].shape[0],
            total_seqlen_k=c["k"].shape[0],
            device="cpu",
        )
        q = c["q"].clone().requires_grad_(True)
        k = c["k"].clone().requires_grad_(True)
        v = c["v"].clone().requires_grad_(True)
        o, meta = ref_attn_func(
            q, k, v, mask,
            softmax_scale=c["softmax_scale"],
            softcap=c["softcap"],
            backend="torch",
            high_precision=True,
            return_lse=True,
        )
        lse = meta.lse if hasattr(meta, "lse") else meta
        o.backward(c["dout"])
        fixtures.append(
            dict(
                name=c["name"], q=c["q"], k=c["k"], v=c["v"], dout=c["dout"],
                q_ranges=c["q_ranges"], k_ranges=c["k_ranges"],
                attn_types=c["attn_types"], softmax_scale=c["softmax_scale"],
                softcap=c["softcap"], mask=mask,
                out=o.detach(), lse=lse.detach(),
                dq=q.grad, dk=k.grad, dv=v.grad,
            )
        )
        print(f"  {c['name']}: out {tuple(o.shape)} lse {tuple(lse.shape)}")
    torch.save(fixtures, out_path)
    print(f"wrote {out_path} ({out_path.stat().st_size/1e6:.2f} MB, {len(fixtures)} cases)")


if __name__ == "__main__":
    main()
