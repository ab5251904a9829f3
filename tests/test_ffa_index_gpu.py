"""GPU parity tests for the index-attention (token-gather) forward kernel
(csrc/ffa_index.hip) against an fp32 SDPA reference masked by
get_sdpa_mask_from_index_attn_indices.

Mirrors the reference's tests/test_attn/test_index_attn.py tiers: GQA ratios
128/64/32/16, cross-batch variable topk, S_q != S_kv, D=64/128, long-sequence
global-id arithmetic, all-padding rows."""
import pytest
import torch

pytestmark = pytest.mark.gpu

from magi_attention.utils import (  # noqa: E402
    build_index_attn_indices,
    get_sdpa_mask_from_index_attn_indices,
)

requires_gpu = pytest.mark.skipif(
    not torch.cuda.is_available(), reason="needs MI355X"
)


def _run_case(B, S_q, S_kv, NHQ, NHK, D, topk, max_topk, seed=0,
              softcap=0.0, indices=None):
    """Build (b,s,h)-packed q/k/v like the reference test
    (test_index_attn.py _run_sparse_attn_and_get_output), run the kernel,
    compare out+lse against masked fp32 SDPA."""
    from magi_attention.functional import flex_flash_attn_func

    torch.manual_seed(seed)
    dev = "cuda"
    ratio = NHQ // NHK
    q = torch.randn(B, S_q, NHQ, D, dtype=torch.bfloat16, device=dev)
    k = torch.randn(B, S_kv, NHK, D, dtype=torch.bfloat16, device=dev)
    v = torch.randn(B, S_kv, NHK, D, dtype=torch.bfloat16, device=dev)
    if indices is None:
        indices = build_index_attn_indices(
            B, NHK, S_q, S_kv, topk, max_topk, device=dev
        )

    # fold: q -> (b s h1) h2 d ; k/v -> (b s h) 1 d
    q_ffa = (
        q.view(B, S_q, NHK, ratio, D)
        .reshape(B * S_q * NHK, ratio, D)
        .contiguous()
    )
    k_ffa = k.reshape(B * S_kv * NHK, 1, D).contiguous()
    v_ffa = v.reshape(B * S_kv * NHK, 1, D).contiguous()

    with torch.no_grad():
        out, meta = flex_flash_attn_func(
            q_ffa, k_ffa, v_ffa,
            index_attn_indices=indices, softcap=softcap,
        )
    torch.cuda.synchronize()

    # fp32 SDPA reference with the dense mask
    mask = get_sdpa_mask_from_index_attn_indices(
        indices, B, NHQ, NHK, S_q, S_kv, device=dev
    )
    scale = D ** (-0.5)
    qf = q.permute(0, 2, 1, 3).float()  # B, NHQ, S_q, D
    kf = k.permute(0, 2, 1, 3).float()
    vf = v.permute(0, 2, 1, 3).float()
    kv_head = (
        torch.arange(NHQ, device=dev) // ratio
    )  # NHQ ordering is h1*ratio + h2 (fold order)
    kf = kf[:, kv_head]
    vf = vf[:, kv_head]
    s = torch.einsum("bhqd,bhkd->bhqk", qf, kf) * scale
    if softcap > 0:
        s = softcap * torch.tanh(s / softcap)
    s = s.masked_fill(~mask, float("-inf"))
    ref_lse = torch.logsumexp(s, dim=-1)  # B, NHQ, S_q (-inf for empty rows)
    p = torch.softmax(s, dim=-1).nan_to_num(0.0)
    ref_out = torch.einsum("bhqk,bhkd->bhqd", p, vf)

    got_out = (
        out.reshape(B, S_q, NHK, ratio, D)
        .permute(0, 2, 3, 1, 4)
        .reshape(B, NHQ, S_q, D)
        .float()
    )
    got_lse = (
        meta.lse.reshape(B, S_q, NHK, ratio)
        .permute(0, 2, 3, 1)
        .reshape(B, NHQ, S_q)
    )
    out_err = (got_out - ref_out).abs().max().item()
    finite = ref_lse.isfinite()
    lse_err = (
        (got_lse[finite] - ref_lse[finite]).abs().max().item()
        if finite.any() else 0.0
    )
    assert (got_lse.isfinite() == finite).all(), "lse -inf pattern mismatch"
    assert out_err < 2.5e-2, f"out err {out_err}"
    assert lse_err < 1e-2, f"lse err {lse_err}"
    # empty rows must keep zero out
    if (~finite).any():
        empty_out = got_out.permute(0, 2, 1, 3)[~finite.permute(0, 2, 1)]
        assert (empty_out == 0).all()


@requires_gpu
@pytest.mark.parametrize(
    "ratio,nhk", [(128, 1), (64, 1), (32, 2), (16, 2), (8, 1), (1, 4)]
)
def test_index_gqa_ratios(ratio, nhk):
    """Tier 1/3c/3d: DiT PackGQA shapes down to small-ratio MHA."""
    _run_case(B=2, S_q=33, S_kv=128, NHQ=ratio * nhk, NHK=nhk, D=128,
              topk=48, max_topk=64, seed=ratio)


@requires_gpu
def test_index_cross_batch_topk():
    """Tier 2a: per-batch different topk (padding paths per batch)."""
    _run_case(B=3, S_q=16, S_kv=256, NHQ=64, NHK=1, D=128,
              topk=[17, 192, 256], max_topk=256, seed=7)


@requires_gpu
def test_index_qkv_different_lengths():
    """Tier 2b: short unaligned Q, long KV."""
    _run_case(B=1, S_q=5, S_kv=512, NHQ=128, NHK=1, D=128,
              topk=100, max_topk=128, seed=11)


@requires_gpu
@pytest.mark.parametrize("d", [64, 128])
def test_index_head_dims(d):
    """Tier 3a."""
    _run_case(B=2, S_q=24, S_kv=192, NHQ=32, NHK=1, D=d,
              topk=64, max_topk=64, seed=d)


@requires_gpu
def test_index_softcap():
    _run_case(B=1, S_q=16, S_kv=128, NHQ=64, NHK=1, D=128,
              topk=64, max_topk=64, seed=3, softcap=30.0)


@requires_gpu
def test_index_long_sequence_ids():
    """Tier 3b: S_kv=65536 — global ids near 2^16*NHK exercise the 64-bit
    row arithmetic in the gather."""
    _run_case(B=1, S_q=4, S_kv=65536, NHQ=64, NHK=1, D=128,
              topk=128, max_topk=128, seed=13)


@requires_gpu
def test_index_all_padding_rows():
    """Rows whose whole list is -1 must return out=0, lse=-inf."""
    B, S_q, S_kv, NHQ, NHK, D = 1, 8, 64, 32, 1, 128
    idx = build_index_attn_indices(B, NHK, S_q, S_kv, 32, 64, device="cuda")
    idx[2] = -1
    idx[5] = -1
    _run_case(B, S_q, S_kv, NHQ, NHK, D, topk=None, max_topk=64,
              seed=17, indices=idx)


@requires_gpu
def test_index_odd_head_dim_padding():
    """d=96 routes through the zero-pad bucket path."""
    _run_case(B=1, S_q=16, S_kv=128, NHQ=64, NHK=1, D=96,
              topk=64, max_topk=64, seed=19)
