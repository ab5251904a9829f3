"""Test-only attention backend for the CP runtime on CPU (gloo tests).
Implements the exact partial-attention accumulate/merge contract of the HIP
kernel (fp64, oracle math) so the runtime's comm/meta machinery can be
validated without a GPU — the reference's own bring-up strategy
(tests/README.md: SDPA backends as in-tree oracle)."""
import torch

from oracle import make_attn_mask


class OracleBackend:
    @staticmethod
    def fwd_partial(q, k, v, arg, out_acc, lse_acc, scale, softcap=0.0,
                    max_logits=None):
        L, hq, d = q.shape
        K = k.shape[0]
        hk = k.shape[1]
        g = hq // hk
        mask = make_attn_mask(L, K, arg.q_ranges, arg.k_ranges, arg.attn_type_map)
        qf = q.double().permute(1, 0, 2)
        kf = k.double().repeat_interleave(g, 1).permute(1, 0, 2)
        vf = v.double().repeat_interleave(g, 1).permute(1, 0, 2)
        s = qf @ kf.transpose(-1, -2) * scale
        if softcap > 0.0:
            s = softcap * torch.tanh(s / softcap)
        s = torch.where(mask.unsqueeze(0), s, torch.full_like(s, float("-inf")))
        if max_logits is not None:
            ml = s.amax(dim=(-1, -2)).float()  # [hq]
            torch.maximum(max_logits, ml, out=max_logits)
        lse_new = torch.logsumexp(s, -1)  # [hq, L]
        p = torch.nan_to_num(torch.exp(s - lse_new.unsqueeze(-1)), nan=0.0)
        o_new = (p @ vf).permute(1, 0, 2)  # [L, hq, d]
        lse_new = lse_new.permute(1, 0)  # [L, hq]
        # merge into accumulators (same math as the fwd lock-merge epilogue)
        lse_prev = lse_acc.double()
        mx = torch.maximum(lse_prev, lse_new)
        mn = torch.minimum(lse_prev, lse_new)
        lse_m = torch.where(
            mx == float("-inf"), mx, mx + torch.log1p(torch.exp(mn - mx))
        )
        w_prev = torch.nan_to_num(torch.exp(lse_prev - lse_m), nan=0.0)
        w_new = torch.nan_to_num(torch.exp(lse_new - lse_m), nan=0.0)
        out_acc.copy_(
            (w_prev.unsqueeze(-1) * out_acc.double()
             + w_new.unsqueeze(-1) * o_new).to(out_acc.dtype)
        )
        lse_acc.copy_(lse_m.to(lse_acc.dtype))

    @staticmethod
    def bwd_partial(dout, q, k, v, out, lse, dpsum, arg, dq, dk, dv, scale,
                    softcap=0.0):
        L, hq, d = q.shape
        K = k.shape[0]
        hk = k.shape[1]
        g = hq // hk
        mask = make_attn_mask(L, K, arg.q_ranges, arg.k_ranges, arg.attn_type_map)
        qf = q.double().permute(1, 0, 2)
        kf = k.double().repeat_interleave(g, 1).permute(1, 0, 2)
        vf = v.double().repeat_interleave(g, 1).permute(1, 0, 2)
        dof = dout.double().permute(1, 0, 2)
        s = qf @ kf.transpose(-1, -2) * scale
        if softcap > 0.0:
            th = torch.tanh(s / softcap)
            s = softcap * th
            dscale = 1.0 - th * th  # d(capped)/d(scaled) per element
        else:
            dscale = 1.0
        lse_t = lse.double().permute(1, 0).unsqueeze(-1)  # [hq, L, 1]
        p = torch.where(
            mask.unsqueeze(0), torch.exp(s * 1.0 - lse_t), torch.zeros_like(s)
        )
        p = torch.nan_to_num(p, nan=0.0, posinf=0.0)
        dp = dof @ vf.transpose(-1, -2)  # [hq, L, K]
        dps = dpsum.double().permute(1, 0).unsqueeze(-1)
        ds = p * (dp - dps) * dscale * scale
        dq_p = (ds @ kf).permute(1, 0, 2)
        dk_p = (ds.transpose(-1, -2) @ qf).permute(1, 0, 2)  # [K, hq, d]
        dv_p = (p.transpose(-1, -2) @ dof).permute(1, 0, 2)
        dq += dq_p.to(dq.dtype)
        # GQA: sum query-head groups into kv heads
        dk += dk_p.reshape(K, hk, g, d).sum(2).to(dk.dtype)
        dv += dv_p.reshape(K, hk, g, d).sum(2).to(dv.dtype)
