# Online-attention correction helpers (reference functional/utils.py:286
# correct_attn_lse, :322 correct_attn_out, :467 correct_attn_out_lse, :561/:593
# sink variants) — the public merge math users compose partial (out, lse)
# pairs with. The in-engine merges run in the HIP kernels
# (csrc/range_ops.hip correct_out_lse / the fwd lock-merge epilogue); these
# torch forms keep the reference's API.
from __future__ import annotations

import torch
import torch.nn.functional as F


def _safe_sub(a: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
    # -inf - -inf -> nan; the merge semantics want -inf
    d = a - b
    return torch.where(torch.isnan(d), torch.full_like(d, float("-inf")), d)


def correct_attn_lse(
    lse1: torch.Tensor, lse2: torch.Tensor, inplace: bool = False
) -> torch.Tensor:
    """lse = log(exp(lse1) + exp(lse2)) = max + softplus(min - max)
    (reference functional/utils.py:286)."""
    assert lse1.dtype == lse2.dtype
    mn = torch.minimum(lse1, lse2).float()
    mx = torch.maximum(lse1, lse2).float()
    lse = mx + F.softplus(_safe_sub(mn, mx))
    return lse1.copy_(lse) if inplace else lse.to(lse1.dtype)


def correct_attn_out(
    out1: torch.Tensor,
    lse1: torch.Tensor,
    out2: torch.Tensor,
    lse2: torch.Tensor,
    lse: torch.Tensor,
    inplace: bool = False,
) -> torch.Tensor:
    """out = exp(lse1-lse)*out1 + exp(lse2-lse)*out2 (reference :322)."""
    w1 = torch.exp(_safe_sub(lse1.float(), lse.float())).nan_to_num(0.0)
    w2 = torch.exp(_safe_sub(lse2.float(), lse.float())).nan_to_num(0.0)
    out = w1.unsqueeze(-1) * out1.float() + w2.unsqueeze(-1) * out2.float()
    return out1.copy_(out) if inplace else out.to(out1.dtype)


def correct_attn_out_lse(
    out1: torch.Tensor,
    lse1: torch.Tensor,
    out2: torch.Tensor,
    lse2: torch.Tensor,
    inplace: bool = False,
) -> tuple[torch.Tensor, torch.Tensor]:
    """Merge two partial (out, lse) pairs (reference :467)."""
    lse = correct_attn_lse(lse1, lse2)
    out = correct_attn_out(out1, lse1, out2, lse2, lse)
    if inplace:
        return out1.copy_(out), lse1.copy_(lse.to(lse1.dtype))
    return out, lse


def _lse_sink(sink: torch.Tensor, sink_layout: str, lse: torch.Tensor):
    """lse_sink per (row?, head) from the sink logits (reference :262
    calc_lse_sink)."""
    if sink_layout == "sh":      # [s_sink, hq] -> [hq] broadcast over rows
        return torch.logsumexp(sink.float(), dim=0).expand_as(lse)
    if sink_layout == "ssh":     # [sq, s_sink, hq] -> [sq, hq]
        return torch.logsumexp(sink.float(), dim=1)
    raise ValueError(f"unsupported sink_layout {sink_layout}")


def correct_attn_lse_with_sink(
    lse: torch.Tensor,
    sink: torch.Tensor,
    sink_layout: str = "sh",
    inplace: bool = False,
) -> torch.Tensor:
    """lse' = log(exp(lse) + exp(lse_sink)) (reference :561)."""
    ls = _lse_sink(sink, sink_layout, lse)
    new = torch.logaddexp(lse.float(), ls)
    return lse.copy_(new) if inplace else new.to(lse.dtype)


def correct_attn_out_with_sink(
    out: torch.Tensor,
    lse: torch.Tensor,
    sink: torch.Tensor,
    sink_layout: str = "sh",
    inplace: bool = False,
) -> torch.Tensor:
    """out' = out * exp(lse - lse_with_sink) (reference :593)."""
    ls = _lse_sink(sink, sink_layout, lse)
    new_lse = torch.logaddexp(lse.float(), ls)
    w = torch.exp(_safe_sub(lse.float(), new_lse)).nan_to_num(0.0).unsqueeze(-1)
    return out.mul_(w.to(out.dtype)) if inplace else (out * w).to(out.dtype)


def correct_attn_out_lse_with_sink(
    out: torch.Tensor,
    lse: torch.Tensor,
    sink: torch.Tensor,
    sink_layout: str = "sh",
    inplace: bool = False,
) -> tuple[torch.Tensor, torch.Tensor]:
    """Fold sink logits into a final (out, lse) pair (reference :635; the
    in-engine form is csrc/range_ops.hip sink_postprocess_kernel)."""
    ls = _lse_sink(sink, sink_layout, lse)
    new_lse = torch.logaddexp(lse.float(), ls)
    w = torch.exp(_safe_sub(lse.float(), new_lse)).nan_to_num(0.0).unsqueeze(-1)
    if inplace:
        out.mul_(w.to(out.dtype))
        lse.copy_(new_lse)
        return out, lse
    return (out * w).to(out.dtype), new_lse.to(lse.dtype)


def safe_subtract(a: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
    """a - b with (-inf) - (-inf) = -inf instead of nan (reference :25)."""
    mask = (a == b) & (a == float("-inf"))
    return (a - b).masked_fill(mask, float("-inf"))


def safe_lse(x: torch.Tensor, dim: int = -1, keepdim: bool = False):
    """logsumexp whose all--inf rows give -inf with zero grads
    (reference :38)."""
    all_neg_inf = (x == float("-inf")).all(dim=dim, keepdim=keepdim)

    if x.requires_grad:
        x.register_hook(lambda g: g.nan_to_num(0.0))
    lse = torch.logsumexp(x, dim=dim, keepdim=keepdim)
    return lse.masked_fill(all_neg_inf, float("-inf"))


def safe_softmax(x: torch.Tensor, lse: torch.Tensor | None = None,
                 dim: int = -1) -> torch.Tensor:
    """softmax whose all--inf rows give all-zero rows with zero grads
    (reference :66)."""
    all_neg_inf = (x == float("-inf")).all(dim=dim, keepdim=True)
    if x.requires_grad:
        x.register_hook(lambda g: g.nan_to_num(0.0))
    if lse is not None:
        sm = torch.exp(safe_subtract(x, lse.unsqueeze(dim)))
    else:
        sm = F.softmax(x, dim=dim)
    return sm.masked_fill(all_neg_inf, 0.0)


def softmax_bwd(dout: torch.Tensor, out: torch.Tensor) -> torch.Tensor:
    """Standard softmax backward (reference :107)."""
    diag_out = torch.diag_embed(out)
    outer_out = torch.einsum("...ij, ...ik -> ...ijk", out, out)
    return torch.einsum("...ij, ...ijk -> ...ik", dout, diag_out - outer_out)


def calc_lse_sink(sink: torch.Tensor, seqlen_q: int,
                  sink_layout: str = "sh") -> torch.Tensor:
    """[seqlen_q, hq] log-sum-exp of the sink logits (reference :238)."""
    if sink_layout == "sh":
        return safe_lse(sink, dim=0, keepdim=True).repeat(seqlen_q, 1)
    if sink_layout == "ssh":
        return safe_lse(sink, dim=1)
    raise ValueError(f"unsupported sink_layout {sink_layout}")


def calc_lse_rescale_weight(lse_to_rescale: torch.Tensor,
                            rescaled_lse: torch.Tensor) -> torch.Tensor:
    """exp(old_lse - new_lse) as [sq, hq, 1] (reference :~250)."""
    return (
        torch.exp(safe_subtract(lse_to_rescale.float(), rescaled_lse.float()))
        .nan_to_num(0.0)
        .unsqueeze(-1)
    )


def sink_bwd(
    sink: torch.Tensor,
    lse: torch.Tensor,
    o: torch.Tensor,
    do: torch.Tensor,
    sink_layout: str = "sh",
    dsink: torch.Tensor | None = None,
) -> torch.Tensor:
    """dsink_j = -exp(sink_j - lse_row) * rowsum(dO*O) (reference sink_bwd;
    the in-engine form is csrc/range_ops.hip dsink_*_kernel)."""
    dpsum = (do.float() * o.float()).sum(-1)          # [sq, hq]
    live = torch.isfinite(lse.float())
    if sink_layout == "sh":
        # p[s, h, t] = exp(sink[s, h] - lse[t, h])
        p = torch.exp(sink.float().unsqueeze(-1) - lse.float().t().unsqueeze(0))
        p = torch.where(live.t().unsqueeze(0), p, torch.zeros_like(p))
        g = -(p * dpsum.t().unsqueeze(0)).sum(-1)
    elif sink_layout == "ssh":
        p = torch.exp(sink.float() - lse.float().unsqueeze(1))
        p = torch.where(live.unsqueeze(1), p, torch.zeros_like(p))
        g = -(p * dpsum.unsqueeze(1))
    else:
        raise ValueError(f"unsupported sink_layout {sink_layout}")
    g = g.to(sink.dtype)
    if dsink is not None:
        return dsink.copy_(g)
    return g
