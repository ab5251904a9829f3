"""Reference attention over a dense boolean mask (reference surface:
testing/ref_attn.py:553 ref_attn_func). Pure differentiable torch ops: with
high_precision=False the matmuls run at the INPUT precision — that is what
makes this usable as the low-precision reference of the calibrated-mismatch
procedure (precision.py extract_mismatch_threshold); high_precision=True
computes in fp64 as ground truth. The reference's online_softmax flag is a
memory optimization of ITS torch backend and is accepted as a no-op here."""
from __future__ import annotations

import torch

from ..common.forward_meta import AttnForwardMeta
from ..utils import max_fp_dtype


def ref_attn_func(
    q: torch.Tensor,
    k: torch.Tensor,
    v: torch.Tensor,
    mask: torch.Tensor,
    *,
    sink: torch.Tensor | None = None,
    softmax_scale: float | None = None,
    softcap: float = 0.0,
    layout: str = "thd",
    sink_layout: str = "sh",
    backend: str = "sdpa",
    high_precision: bool = False,
    return_lse: bool = False,
    return_max_logits: bool = False,
    online_softmax: bool = False,
) -> tuple[torch.Tensor, AttnForwardMeta]:
    """q [tq, hq, d], k/v [tk, hk, d] (GQA: hq % hk == 0), mask bool
    [tq, tk] (True = attend). Returns (out, AttnForwardMeta) with natural-log
    lse [tq, hq] and per-head max_logits [hq] when requested. Fully-masked
    rows produce out=0, lse=-inf (the kernels' convention)."""
    assert layout in ("thd",), f"Unsupported layout: {layout}"
    assert softcap == 0.0, "non-zero softcap is not supported by now"
    assert backend in ("sdpa", "torch"), f"Unsupported backend: {backend}"

    org_dtype = q.dtype
    lse_dtype = max_fp_dtype(org_dtype, torch.float32)
    if high_precision:
        q, k, v = q.double(), k.double(), v.double()
        lse_dtype = torch.float64

    tq, hq, d = q.shape
    tk, hk, _ = k.shape
    assert hq % hk == 0
    if hk != hq:
        k = k.repeat_interleave(hq // hk, dim=1)
        v = v.repeat_interleave(hq // hk, dim=1)
    scale = d ** (-0.5) if softmax_scale is None else softmax_scale

    s = torch.einsum("qhd,khd->hqk", q, k) * scale  # [hq, tq, tk]
    neg = torch.finfo(s.dtype).min
    s = s.masked_fill(~mask.unsqueeze(0), neg)

    max_logits = None
    if return_max_logits:
        ml = s.reshape(hq, -1).amax(dim=-1).to(max_fp_dtype(org_dtype, torch.float32))
        any_live = mask.any()
        max_logits = torch.where(
            any_live, ml, torch.full_like(ml, float("-inf"))
        )

    # sink columns join the softmax normalization only (no V rows)
    if sink is not None:
        sink = sink.to(s.dtype)
        if sink_layout == "sh":  # [s_sink, hq]
            cols = sink.t().unsqueeze(1).expand(hq, tq, sink.shape[0])
        elif sink_layout == "ssh":  # [tq, s_sink, hq]
            cols = sink.permute(2, 0, 1)
        else:
            raise ValueError(f"Unsupported sink_layout: {sink_layout}")
        s_all = torch.cat([s, cols], dim=-1)
    else:
        s_all = s

    live = mask.any(dim=-1) if sink is None else torch.ones(
        tq, dtype=torch.bool, device=q.device
    )
    m = s_all.amax(dim=-1, keepdim=True)
    e = torch.exp(s_all - m)
    e = torch.where(s_all <= neg / 2, torch.zeros_like(e), e)  # true -inf cells
    denom = e.sum(dim=-1, keepdim=True)
    p = e[..., :tk] / denom.clamp_min(torch.finfo(e.dtype).tiny)
    out = torch.einsum("hqk,khd->qhd", p, v)
    out = torch.where(live.view(tq, 1, 1), out, torch.zeros_like(out))
    out = out.to(org_dtype)

    lse = None
    if return_lse:
        lse_v = (m.squeeze(-1) + denom.squeeze(-1).log()).t().to(lse_dtype)
        lse = torch.where(
            live.view(tq, 1), lse_v, torch.full_like(lse_v, float("-inf"))
        )

    return out, AttnForwardMeta(lse=lse, max_logits=max_logits)
