# ORACLE — TEST INFRASTRUCTURE ONLY (see oracle/__init__.py).
#
# CPU restatement of the reference's pure-PyTorch attention oracle:
#   - mask semantics follow the reference docstring
#     magi_attention/functional/flex_flash_attn.py:1247-1341 (4 mask types,
#     causal = bottom-right aligned, inv-causal = top-left aligned) and
#     magi_attention/csrc/flexible_flash_attention/mask.h:33-133.
#   - attention math follows magi_attention/testing/ref_attn.py:331-636
#     (explicit-softmax "torch" backend, fp64 high-precision ground truth,
#     returns natural-log LSE; empty rows -> lse=-inf, out=0).
#   - the partial-(out,lse) merge follows magi_attention/functional/utils.py:286-358
#     (lse = max + softplus(min - max); out = sum_i exp(lse_i - lse) * out_i).
#
# Parity pin: validated against golden vectors generated from the reference's
# own `ref_attn_func` running in the build container (tests/golden/, generated
# by tests/golden/generate_from_reference.py). Parity is therefore pinned to
# the reference by procedure AND by stored vectors.
from __future__ import annotations

import math

import torch

FULL, CAUSAL, INV_CAUSAL, BI_CAUSAL = 0, 1, 2, 3


def make_attn_mask(
    total_q: int,
    total_k: int,
    q_ranges,  # [n,2] int (list or tensor)
    k_ranges,  # [n,2] int
    attn_types,  # [n] int
    device=None,
) -> torch.Tensor:
    """Dense bool mask [total_q, total_k] from (q_range, k_range, attn_type)
    triples. Semantics: reference flex_flash_attn.py:1247-1341.

    For a slice with seqlen_q = sq, seqlen_k = sk and local coords (i, j):
      full:       always allowed
      causal:     j - i <= sk - sq          (bottom-right aligned)
      inv_causal: j >= i                    (top-left aligned)
      bi_causal:  both of the above
    Overlapping slices OR together.
    """
    q_ranges = torch.as_tensor(q_ranges, dtype=torch.long)
    k_ranges = torch.as_tensor(k_ranges, dtype=torch.long)
    attn_types = torch.as_tensor(attn_types, dtype=torch.long)
    mask = torch.zeros(total_q, total_k, dtype=torch.bool, device=device)
    for (qs, qe), (ks, ke), t in zip(
        q_ranges.tolist(), k_ranges.tolist(), attn_types.tolist()
    ):
        sq, sk = qe - qs, ke - ks
        if sq <= 0 or sk <= 0:
            continue
        i = torch.arange(sq, device=device).unsqueeze(1)  # [sq,1]
        j = torch.arange(sk, device=device).unsqueeze(0)  # [1,sk]
        if t == FULL:
            sub = torch.ones(sq, sk, dtype=torch.bool, device=device)
        elif t == CAUSAL:
            sub = (j - i) <= (sk - sq)
        elif t == INV_CAUSAL:
            sub = j >= i
        elif t == BI_CAUSAL:
            sub = ((j - i) <= (sk - sq)) & (j >= i)
        else:
            raise ValueError(f"unknown attn type {t}")
        mask[qs:qe, ks:ke] |= sub
    return mask


def ref_attn(
    q: torch.Tensor,  # [tq, hq, d]
    k: torch.Tensor,  # [tk, hk, d]
    v: torch.Tensor,  # [tk, hk, d]
    mask: torch.Tensor,  # [tq, tk] bool
    softmax_scale: float | None = None,
    softcap: float = 0.0,
    high_precision: bool = True,
    p_dtype: torch.dtype | None = None,
    sink: torch.Tensor | None = None,
    sink_layout: str = "sh",
) -> tuple[torch.Tensor, torch.Tensor]:
    """Explicit-softmax attention; returns (out [tq,hq,d] in q.dtype,
    lse [tq,hq] fp32, natural log). GQA: hq must be a multiple of hk.

    Math: s = (q @ k^T) * scale; optional softcap = softcap*tanh(s/softcap);
    masked entries -> -inf; lse = logsumexp(s); out = softmax(s) @ v.
    Rows with no allowed key: lse = -inf, out = 0
    (reference ref_attn.py:331-636 / fwd postprocess behaviour).
    """
    tq, hq, d = q.shape
    tk, hk, _ = k.shape
    assert hq % hk == 0
    g = hq // hk
    dt = torch.float64 if high_precision else torch.float32
    scale = softmax_scale if softmax_scale is not None else d ** (-0.5)

    qf = q.to(dt).permute(1, 0, 2)  # [hq, tq, d]
    kf = k.to(dt).repeat_interleave(g, dim=1).permute(1, 0, 2)  # [hq, tk, d]
    vf = v.to(dt).repeat_interleave(g, dim=1).permute(1, 0, 2)
    s = torch.matmul(qf, kf.transpose(-1, -2)) * scale  # [hq, tq, tk]
    if softcap > 0.0:
        s = softcap * torch.tanh(s / softcap)
    s = torch.where(mask.unsqueeze(0), s, torch.full_like(s, float("-inf")))
    if sink is not None:
        # attention-sink logits participate in the softmax denominator as
        # value-less extra columns — no softmax_scale, no softcap (reference
        # flash_fwd_postprocess_kernel.h: lse_sink folded after the mainloop)
        sf = sink.to(dt)
        if sink_layout == "sh":      # [s_sink, hq]
            extra = sf.permute(1, 0).unsqueeze(1).expand(hq, tq, sf.shape[0])
        elif sink_layout == "ssh":   # [tq, s_sink, hq]
            extra = sf.permute(2, 0, 1)
        else:
            raise ValueError(f"unsupported sink_layout {sink_layout}")
        s = torch.cat([s, extra], dim=-1)
    lse = torch.logsumexp(s, dim=-1)  # [hq, tq]; -inf for empty rows
    p = torch.exp(s - lse.unsqueeze(-1))
    p = torch.nan_to_num(p, nan=0.0)  # empty rows: -inf - -inf = nan -> 0
    if p_dtype is not None:
        # emulate a kernel that quantises P before the PV matmul (bf16 MFMA)
        p = p.to(p_dtype).to(p.dtype)
    if sink is not None:
        p = p[..., : kf.shape[1]]    # sink columns carry no value rows
    out = torch.matmul(p, vf)  # [hq, tq, d]
    return (
        out.permute(1, 0, 2).to(q.dtype),
        lse.permute(1, 0).to(torch.float32),
    )


def ref_attn_with_grads(
    q, k, v, mask, dout, softmax_scale=None, softcap: float = 0.0,
    high_precision: bool = True, p_dtype=None, sink=None, sink_layout="sh",
):
    """Forward + backward through autograd.
    Returns (out, lse, dq, dk, dv[, dsink if sink given])."""
    q_ = q.detach().clone().requires_grad_(True)
    k_ = k.detach().clone().requires_grad_(True)
    v_ = v.detach().clone().requires_grad_(True)
    s_ = sink.detach().clone().requires_grad_(True) if sink is not None else None
    out, lse = ref_attn(
        q_, k_, v_, mask, softmax_scale=softmax_scale, softcap=softcap,
        high_precision=high_precision, p_dtype=p_dtype,
        sink=s_, sink_layout=sink_layout,
    )
    out.backward(dout)
    if sink is not None:
        return out.detach(), lse.detach(), q_.grad, k_.grad, v_.grad, s_.grad
    return out.detach(), lse.detach(), q_.grad, k_.grad, v_.grad


def merge_out_lse(
    outs: list[torch.Tensor], lses: list[torch.Tensor]
) -> tuple[torch.Tensor, torch.Tensor]:
    """Online-softmax merge of partial (out, lse) pairs.
    Reference: magi_attention/functional/utils.py:286-358."""
    lse = lses[0].to(torch.float64)
    for l2 in lses[1:]:
        l2 = l2.to(torch.float64)
        mx = torch.maximum(lse, l2)
        mn = torch.minimum(lse, l2)
        delta = mn - mx
        delta = torch.where(torch.isnan(delta), torch.full_like(delta, -math.inf), delta)
        lse = mx + torch.nn.functional.softplus(delta)
    out = torch.zeros_like(outs[0], dtype=torch.float64)
    for o, l in zip(outs, lses):
        w = torch.exp(l.to(torch.float64) - lse)
        w = torch.nan_to_num(w, nan=0.0)
        out += w.unsqueeze(-1) * o.to(torch.float64)
    return out, lse
