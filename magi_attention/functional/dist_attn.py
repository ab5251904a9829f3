# DistAttnRuntime / DistAttnFunc — the CP execution engine.
# (Reference: functional/dist_attn.py:141 DistAttnRuntime, :2993 DistAttnFunc,
#  :3608 dist_attn_func — multi-stage overlap of group-cast K/V fetch with
#  partial FFA compute, out/lse accumulated in-kernel; backward re-fetches
#  remote KV, computes partial dK/dV per stage and group-reduces them to owner
#  ranks (:1463 _fetch_remote_kv, :1924/_2068 reduce paths).)
#
# MI355X-native: FFA HIP kernel accumulates (out fp32, lse) in place across
# stage calls (lock-merge epilogue), so no separate out-correction pass is
# needed; comm is RCCL a2av over xGMI (comm/primitive.py).
from __future__ import annotations

from dataclasses import dataclass
from typing import List, Optional, Tuple

import torch
import torch.distributed as dist

from .. import env
from ..common.forward_meta import AttnForwardMeta
from ..comm.primitive import (
    WorkWithPostProcessFn,
    group_cast,
    group_reduce,
    group_reduce_out_lse,
    hier_group_cast,
    hier_group_reduce,
)
from ..meta.containers import AttnArg, CalcMeta, CommMeta, QoCommMeta

# test-only hook: lets CPU (gloo) tests run the runtime with an oracle
# attention implementation. NEVER set in production; the product path on GPU
# is the HIP kernel and raises if the extension is missing.
_test_attn_backend = None


def register_test_attn_backend(backend) -> None:
    global _test_attn_backend
    _test_attn_backend = backend


@dataclass
class DistAttnRuntime:
    calc_meta: CalcMeta
    comm_meta: CommMeta
    cp_group: dist.ProcessGroup
    total_local_q: int
    softmax_scale: Optional[float] = None
    # hierarchical 2D-mesh comm (set iff stages_cast_hier planned)
    intra_group: Optional[dist.ProcessGroup] = None
    inter_group: Optional[dist.ProcessGroup] = None
    # QO-comm plan (MAGI_ATTENTION_QO_COMM=1): remote slices computed at the
    # K-host; q (fwd) / q,do,lse,dpsum (bwd) travel instead of K/V
    qo_meta: Optional[QoCommMeta] = None
    # native HIP-IPC grpcoll transport (MAGI_ATTENTION_NATIVE_GRPCOLL=1)
    _native: object = None

    @property
    def overlap_degree(self) -> int:
        return self.comm_meta.overlap_degree

    @property
    def use_hier(self) -> bool:
        return (self.comm_meta.stages_cast_hier is not None
                and self.intra_group is not None)

    def _native_grpcoll(self, kv_local: torch.Tensor):
        """Lazy per-runtime IPC window setup (handles exchanged once)."""
        if self._native is None:
            from ..comm.native_grpcoll import NativeGrpColl

            object.__setattr__(self, "_native", NativeGrpColl(
                self.comm_meta.stages_native,
                self.comm_meta.stage_tokens_all,
                self.cp_group,
                kv_rows=2 * (kv_local.shape[0] // 2),
                h=kv_local.shape[1], d=kv_local.shape[2],
                dtype=kv_local.dtype,
            ))
        return self._native

    @property
    def use_native(self) -> bool:
        return (self.comm_meta.stages_native is not None
                and env.is_native_grpcoll_enable())

    def _cast(self, kv_local: torch.Tensor, s: int) -> WorkWithPostProcessFn:
        if self.use_native and kv_local.is_cuda:
            return self._native_grpcoll(kv_local).cast(kv_local, s)
        if self.use_hier:
            return hier_group_cast(
                kv_local, self.comm_meta.stages_cast_hier[s],
                self.intra_group, self.inter_group,
            )
        return group_cast(kv_local, self.comm_meta.stages_cast[s],
                          self.cp_group)

    def _reduce(self, partial, dst, s: int) -> WorkWithPostProcessFn:
        if self.use_native and dst.is_cuda and self._native is not None:
            return self._native.reduce(partial, dst, s)
        if self.use_hier:
            return hier_group_reduce(
                partial, dst, self.comm_meta.stages_reduce_hier[s],
                self.intra_group, self.inter_group,
            )
        return group_reduce(partial, dst, self.comm_meta.stages_reduce[s],
                            self.cp_group)

    # ---------------- forward ----------------
    def attn_fwd(
        self, q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
        sink: Optional[torch.Tensor] = None,
        softmax_scale: Optional[float] = None,
        softcap: float = 0.0,
        return_max_logits: bool = False,
    ) -> Tuple[torch.Tensor, torch.Tensor, Optional[torch.Tensor]]:
        tq, hq, d = q.shape
        fp8 = q.dtype == torch.float8_e4m3fn
        if fp8:
            assert self.qo_meta is None, "fp8 + QO-comm lands later"
        scale = softmax_scale or self.softmax_scale or d ** (-0.5)

        out_acc = torch.zeros(tq, hq, d, dtype=torch.float32, device=q.device)
        lse_acc = torch.full(
            (tq, hq), float("-inf"), dtype=torch.float32, device=q.device
        )
        max_logits = None
        if return_max_logits:
            max_logits = torch.full((hq,), float("-inf"), dtype=torch.float32,
                                    device=q.device)

        if self.qo_meta is not None:
            # QO-comm: cast q to the K-hosts, compute partial (out,lse) for
            # remote q rows, lse-merge them back to owners (reference
            # dist_attn.py:1570 _fetch_remote_q / :1924 _reduce_partial_out_lse)
            qo = self.qo_meta
            works = [
                group_cast(q, qo.stages_cast1[s], self.cp_group)
                for s in range(self.overlap_degree)
            ]
            self._fwd_partial(q, k, v, qo.calc.host_arg, out_acc, lse_acc,
                              scale, softcap, max_logits)
            rworks: List[WorkWithPostProcessFn] = []
            hp = env.is_forward_high_precision_reduce_enable()
            for s in range(self.overlap_degree):
                arg = qo.calc.stage_args[s]
                stage_q = works[s].wait_post_process()
                S = qo.stages_cast1[s].stage_tokens
                out_s = torch.zeros(S, hq, d, dtype=torch.float32,
                                    device=q.device)
                lse_s = torch.full((S, hq), float("-inf"),
                                   dtype=torch.float32, device=q.device)
                if S > 0 and not arg.is_empty():
                    self._fwd_partial(stage_q, k, v, arg, out_s, lse_s,
                                      scale, softcap, max_logits)
                # collective: every rank joins every stage (it may RECEIVE
                # partials for q rows it owns even with nothing to compute)
                rworks.append(group_reduce_out_lse(
                    out_s, lse_s, out_acc, lse_acc, qo.stages_reduce1[s],
                    self.cp_group, high_precision=hp, wire_dtype=q.dtype,
                ))
            for w in rworks:
                w.wait_post_process()
        else:
            kv_local = torch.cat([k, v], dim=0)
            # pre-issue ALL remote-stage group-casts (reference dist_attn.py:419-436)
            works = [
                self._cast(kv_local, s) for s in range(self.overlap_degree)
            ]
            self._fwd_partial(q, k, v, self.calc_meta.host_arg, out_acc,
                              lse_acc, scale, softcap, max_logits)
            for s in range(self.overlap_degree):
                arg = self.calc_meta.stage_args[s]
                stage_kv = works[s].wait_post_process()
                S = self.comm_meta.stages_cast[s].stage_tokens
                if S == 0 or arg.is_empty():
                    continue
                self._fwd_partial(
                    q, stage_kv[:S], stage_kv[S:], arg, out_acc, lse_acc,
                    scale, softcap, max_logits,
                )
        if sink is not None:
            # fold the replicated sink ONCE into this rank's (out, lse) — the
            # q rows are rank-disjoint, so per-rank postprocess is exact
            # (reference dist_attn.py: calc_lse_sink_compiled path)
            self._sink_post(out_acc, lse_acc, sink)
        if max_logits is not None:
            dist.all_reduce(max_logits, op=dist.ReduceOp.MAX,
                            group=self.cp_group)
        # the fp8 extension returns bf16 out (same policy as the single-GPU
        # wrapper: flex_flash_attn.py)
        out = out_acc.to(torch.bfloat16 if fp8 else q.dtype)
        return out, lse_acc, max_logits

    def _sink_post(self, out_acc, lse_acc, sink):
        if _test_attn_backend is not None and not out_acc.is_cuda:
            lse_sink = torch.logsumexp(sink.double(), dim=0)  # [hq]
            lse_old = lse_acc.double()
            lse_new = torch.logaddexp(lse_old, lse_sink.expand_as(lse_old))
            w = torch.where(
                lse_old == float("-inf"), torch.zeros_like(lse_old),
                torch.exp(lse_old - lse_new),
            )
            out_acc.mul_(w.unsqueeze(-1).to(out_acc.dtype))
            lse_acc.copy_(lse_new.to(lse_acc.dtype))
            return
        from .flex_flash_attn import _apply_sink_postprocess

        tq, hq, d = out_acc.shape
        _apply_sink_postprocess(out_acc, lse_acc, sink, "sh", tq, hq, d)

    def _fwd_partial(self, q, k, v, arg: AttnArg, out_acc, lse_acc, scale,
                     softcap=0.0, max_logits=None):
        if arg.is_empty():
            return
        if _test_attn_backend is not None and not q.is_cuda:
            _test_attn_backend.fwd_partial(q, k, v, arg, out_acc, lse_acc,
                                           scale, softcap, max_logits)
            return
        from .flex_flash_attn import _flex_flash_attn_forward

        qr, kr, tm = arg.to_device(q.device)
        qk_starts = None
        if (env.is_auto_range_merge_enable()
                and not env.is_deterministic_mode_enable()):
            (qr, kr, tm, qk_starts), _ = arg.to_device_merged(q.device)
        _flex_flash_attn_forward(
            q=q, k=k, v=v, sink=None, sink_layout="sh",
            out=out_acc, lse=lse_acc,
            q_ranges=qr, k_ranges=kr, attn_type_map=tm,
            softmax_scale=scale, softcap=softcap, out_type=torch.float32,
            disable_fwd_atomic_reduction=False,
            deterministic=env.is_deterministic_mode_enable(),
            # CU reservation only pays when comm kernels need the CUs
            sm_margin=(env.ffa_forward_sm_margin()
                       if dist.get_world_size(self.cp_group) > 1 else 0),
            max_seqlen_q=arg.max_seqlen_q,
            max_logits=max_logits, qk_starts=qk_starts,
        )

    # ---------------- backward ----------------
    def attn_bwd(
        self,
        dout: torch.Tensor,
        q: torch.Tensor,
        k: torch.Tensor,
        v: torch.Tensor,
        out: torch.Tensor,
        lse: torch.Tensor,
        sink: Optional[torch.Tensor] = None,
        softmax_scale: Optional[float] = None,
        softcap: float = 0.0,
    ):
        tq, hq, d = q.shape
        L = k.shape[0]
        scale = softmax_scale or self.softmax_scale or d ** (-0.5)
        in_dtype = q.dtype
        if in_dtype == torch.float8_e4m3fn:
            # mixed-precision policy (single-GPU wrapper, flex_flash_attn.py
            # :521): backward runs the bf16 kernels over upcast operands
            q = q.bfloat16()
            k = k.bfloat16()
            v = v.bfloat16()
            dout = dout.bfloat16()
            if out.dtype == torch.float8_e4m3fn:
                out = out.bfloat16()

        dq_acc = torch.zeros(tq, hq, d, dtype=torch.float32, device=q.device)
        dkv_acc = torch.zeros(2 * L, *k.shape[1:], dtype=torch.float32,
                              device=q.device)
        dpsum = self._bwd_dpsum(dout, out)
        dsink = None
        if sink is not None:
            dsink = self._dsink(sink, lse, dpsum)
        hide_tail = env.is_bwd_hide_tail_reduce()
        rworks: List[WorkWithPostProcessFn] = []

        if self.qo_meta is not None:
            # QO-comm backward (reference dist_attn.py:1659
            # _fetch_remote_qo_do_lse): cast (q,do) + (lse,dpsum) to the
            # K-hosts; partial dK/dV accumulate LOCALLY (no dKV wire), only
            # partial dq returns to q owners. We cast dpsum instead of the
            # reference's o — the kernels only consume dpsum, which the
            # owner has already computed for all its rows.
            qo = self.qo_meta
            qdo = torch.cat([q, dout], dim=0)
            lsedp = torch.cat([lse, dpsum], dim=0)
            works2 = [group_cast(qdo, qo.stages_cast2[s], self.cp_group)
                      for s in range(self.overlap_degree)]
            worksl = [group_cast(lsedp, qo.stages_cast2[s], self.cp_group)
                      for s in range(self.overlap_degree)]
            self._bwd_partial(
                dout, q, k, v, out, lse, dpsum, qo.calc.host_arg,
                dq_acc, dkv_acc[:L], dkv_acc[L:], scale, softcap,
            )
            # low-precision (default) wire = the PARAM dtype (bf16 for bf16
            # training; fp64 test runs stay exact), fp32 with the HP flag
            wire = (None if env.is_backward_high_precision_reduce_enable()
                    else k.dtype)
            for s in range(self.overlap_degree):
                arg = qo.calc.stage_args[s]
                buf = works2[s].wait_post_process()
                lbuf = worksl[s].wait_post_process()
                S = qo.stages_cast2[s].stage_tokens
                dq_stage = torch.zeros(S, hq, d, dtype=torch.float32,
                                       device=q.device)
                if S > 0 and not arg.is_empty():
                    self._bwd_partial(
                        buf[S:], buf[:S], k, v, out, lbuf[:S], lbuf[S:], arg,
                        dq_stage, dkv_acc[:L], dkv_acc[L:], scale, softcap,
                    )
                rworks.append(group_reduce(
                    dq_stage, dq_acc, qo.stages_reduce1[s], self.cp_group,
                    wire_dtype=wire,
                ))
        else:
            kv_local = torch.cat([k, v], dim=0)
            # re-fetch remote KV (reference backward:3455 -> _fetch_remote_kv:1463)
            works = [self._cast(kv_local, s)
                     for s in range(self.overlap_degree)]
            self._bwd_partial(
                dout, q, k, v, out, lse, dpsum, self.calc_meta.host_arg,
                dq_acc, dkv_acc[:L], dkv_acc[L:], scale, softcap,
            )
            # reference default: bf16 wire for partial dKV; the
            # high-precision-reduce flag doubles the wire for fp32
            # (env/comm.py:107)
            # low-precision (default) wire = the PARAM dtype (bf16 for bf16
            # training; fp64 test runs stay exact), fp32 with the HP flag
            wire = (None if env.is_backward_high_precision_reduce_enable()
                    else k.dtype)
            for s in range(self.overlap_degree):
                arg = self.calc_meta.stage_args[s]
                stage_kv = works[s].wait_post_process()
                S = self.comm_meta.stages_cast[s].stage_tokens
                # group_reduce is a COLLECTIVE: every rank must join every
                # stage even with an empty local partial (this rank may still
                # RECEIVE dK/dV contributions for KV rows it hosts).
                dkv_stage = torch.zeros(
                    2 * S, *k.shape[1:], dtype=torch.float32, device=q.device
                )
                if S > 0 and not arg.is_empty():
                    self._bwd_partial(
                        dout, q, stage_kv[:S], stage_kv[S:], out, lse, dpsum,
                        arg, dq_acc, dkv_stage[:S], dkv_stage[S:], scale,
                        softcap,
                    )
                rworks.append(self._reduce(dkv_stage, dkv_acc, s))
        if hide_tail:
            # MAGI_ATTENTION_BWD_HIDE_TAIL_REDUCE: run the local-grad dtype
            # casts UNDER the tail-stage reduce instead of after it
            # (reference _hide_tail_stage_reduce_backward dist_attn.py:2503)
            dq = dq_acc.to(q.dtype) if self.qo_meta is None else None
            for w in rworks:
                w.wait_post_process()
            if dq is None:
                dq = dq_acc.to(q.dtype)
        else:
            for w in rworks:
                w.wait_post_process()
            dq = dq_acc.to(q.dtype)
        if dsink is not None:
            # "sh" sink is replicated; its gradient sums over ALL q rows
            dist.all_reduce(dsink, op=dist.ReduceOp.SUM, group=self.cp_group)
        if in_dtype == torch.float8_e4m3fn:
            dq = dq_acc.to(in_dtype)
        dk = dkv_acc[:L].to(in_dtype)
        dv = dkv_acc[L:].to(in_dtype)
        return dq, dk, dv, dsink

    def _dsink(self, sink, lse, dpsum):
        if _test_attn_backend is not None and not lse.is_cuda:
            live = torch.isfinite(lse.double())
            # p[s, h, t] = exp(sink[s, h] - lse[t, h])
            p = torch.exp(
                sink.double().unsqueeze(-1) - lse.double().t().unsqueeze(0)
            )
            p = torch.where(live.t().unsqueeze(0), p, torch.zeros_like(p))
            return (-(p * dpsum.double().t().unsqueeze(0)).sum(-1)).to(
                torch.float32
            )
        from .. import _ffa_lib
        from .._ffa_lib import MagiSinkArgs, check, current_stream_ptr, ptr

        tq, hq = lse.shape
        sink_f = sink.contiguous().float()
        dsink = torch.zeros_like(sink_f)
        sargs = MagiSinkArgs(
            lse=ptr(lse), sink=ptr(sink_f), dsink=ptr(dsink), dpsum=ptr(dpsum),
            total_rows=tq, n_heads=hq, d=0, s_sink=sink_f.shape[0], ssh=0,
            stream=current_stream_ptr(),
        )
        check(_ffa_lib.lib().magi_ffa_dsink(sargs), "magi_ffa_dsink[dist]")
        return dsink

    def _bwd_dpsum(self, dout, out):
        if _test_attn_backend is not None and not dout.is_cuda:
            return (dout.float() * out.float()).sum(-1)
        from .. import _ffa_lib
        from .._ffa_lib import MagiFfaBwdArgs, check, current_stream_ptr, ptr

        tq, hq, d = dout.shape
        dpsum = torch.empty(tq, hq, dtype=torch.float32, device=dout.device)
        args = MagiFfaBwdArgs(
            dout=ptr(dout.contiguous()), out=ptr(out.contiguous()),
            dpsum=ptr(dpsum),
            total_q=tq, hq=hq, d=d,
            out_is_fp32=int(out.dtype == torch.float32),
            stream=current_stream_ptr(),
        )
        check(_ffa_lib.lib().magi_ffa_bwd_preprocess(args), "bwd_preprocess")
        return dpsum

    def _bwd_partial(self, dout, q, k, v, out, lse, dpsum, arg: AttnArg,
                     dq, dk, dv, scale, softcap=0.0):
        if arg.is_empty():
            return
        if _test_attn_backend is not None and not q.is_cuda:
            _test_attn_backend.bwd_partial(
                dout, q, k, v, out, lse, dpsum, arg, dq, dk, dv, scale, softcap
            )
            return
        from .. import _ffa_lib
        from .._ffa_lib import MagiFfaBwdArgs, check, current_stream_ptr, ptr

        qr, kr, tm = arg.to_device(q.device)
        tq, hq, d = q.shape
        tk, hk, _ = k.shape
        max_k = int(max(b - a for a, b in arg.k_ranges))
        args = MagiFfaBwdArgs(
            dout=ptr(dout), q=ptr(q), k=ptr(k.contiguous()),
            v=ptr(v.contiguous()), out=ptr(out), lse=ptr(lse),
            dq=ptr(dq), dk=ptr(dk), dv=ptr(dv), dpsum=ptr(dpsum),
            q_ranges=ptr(qr), k_ranges=ptr(kr), attn_type_map=ptr(tm),
            n_ranges=qr.shape[0], total_q=tq, total_k=tk,
            hq=hq, hk=hk, d=d, max_seqlen_k=max_k,
            out_is_fp32=int(out.dtype == torch.float32),
            softmax_scale=scale, softcap=softcap,
            cu_margin=(env.ffa_backward_sm_margin()
                       if dist.get_world_size(self.cp_group) > 1 else 0),
            stream=current_stream_ptr(),
        )
        from .flex_flash_attn import run_bwd_deterministic, run_bwd_passes

        if env.is_deterministic_mode_enable():
            run_bwd_deterministic(args, qr, kr, tm, hq, hk, q.device)
        elif env.is_auto_range_merge_enable():
            dq_t, dkv_t = arg.to_device_merged(q.device)
            run_bwd_passes(args, q.device, dq_tables=dq_t, dkv_tables=dkv_t)
        else:
            run_bwd_passes(args, q.device)


class DistAttnFunc(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, v, sink, runtime: DistAttnRuntime,
                softmax_scale, softcap, return_max_logits):
        out, lse, max_logits = runtime.attn_fwd(
            q, k, v, sink=sink, softmax_scale=softmax_scale, softcap=softcap,
            return_max_logits=return_max_logits,
        )
        ctx.save_for_backward(q, k, v, out, lse, sink)
        ctx.runtime = runtime
        ctx.softmax_scale = softmax_scale
        ctx.softcap = softcap
        if max_logits is not None:
            ctx.mark_non_differentiable(max_logits)
        return out, lse, max_logits

    @staticmethod
    def backward(ctx, dout, *_):
        q, k, v, out, lse, sink = ctx.saved_tensors
        dq, dk, dv, dsink = ctx.runtime.attn_bwd(
            dout.contiguous(), q, k, v, out, lse, sink=sink,
            softmax_scale=ctx.softmax_scale, softcap=ctx.softcap,
        )
        if sink is not None and dsink is not None:
            dsink = dsink.to(sink.dtype)
        return dq, dk, dv, dsink, None, None, None, None


def dist_attn_func(
    q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
    runtime: DistAttnRuntime,
    sink: Optional[torch.Tensor] = None,
    softmax_scale: Optional[float] = None,
    softcap: float = 0.0,
    return_max_logits: bool = False,
):
    """Reference functional/dist_attn.py:3608."""
    out, lse, max_logits = DistAttnFunc.apply(
        q, k, v, sink, runtime, softmax_scale, softcap, return_max_logits
    )
    return out, AttnForwardMeta(lse=lse, max_logits=max_logits)
