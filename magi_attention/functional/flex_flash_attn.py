# Reference surface: magi_attention/functional/flex_flash_attn.py
# (_flex_flash_attn_forward:334, _flex_flash_attn_backward:594,
#  FlexFlashAttnFunc:699, flex_flash_attn_func:1066).
# MI355X-native rebuild: the JIT'd CUDA module is replaced by the C-ABI HIP
# library (include/magi_ffa.h) bound via ctypes in magi_attention/_ffa_lib.py.
from __future__ import annotations

import ctypes
import math
from typing import Optional, Tuple

import torch

_side_stream: dict = {}


def _get_side_stream(device) -> torch.cuda.Stream:
    key = device.index or 0
    if key not in _side_stream:
        _side_stream[key] = torch.cuda.Stream(device=device)
    return _side_stream[key]


def run_bwd_deterministic(args, q_ranges, k_ranges, attn_type_map,
                          hq, hk, device) -> None:
    """Deterministic backward: sequential launches over q-disjoint groups
    (dq pass), k-disjoint groups x GQA head sub-launches (dkv pass) — every
    accumulator element receives its additions in a fixed, stream-ordered
    sequence."""
    lib = _ffa_lib.lib()
    gqa = hq // hk
    q_groups = _color_ranges(q_ranges.cpu().tolist())
    k_groups = _color_ranges(k_ranges.cpu().tolist())
    base_margin = args.cu_margin & 0xFFFF

    keep = []

    def launch(fn, groups, ranges_pairs, what, head_splits=1):
        for g in groups:
            idx = torch.tensor(g, dtype=torch.long, device=device)
            sq = _subset(ranges_pairs[0], idx)
            sk = _subset(ranges_pairs[1], idx)
            st = _subset(attn_type_map, idx)
            keep.append((sq, sk, st))  # outlive the async launches
            args.q_ranges = ptr(sq)
            args.k_ranges = ptr(sk)
            args.attn_type_map = ptr(st)
            args.n_ranges = len(g)
            for j in range(head_splits):
                if head_splits > 1:
                    args.cu_margin = base_margin | (gqa << 24) | (j << 16)
                check(fn(args), what)
            args.cu_margin = base_margin

    launch(lib.magi_ffa_bwd_dq, q_groups, (q_ranges, k_ranges), "bwd_dq[det]")
    hs = gqa if gqa > 1 else 1
    if env.is_bwd_split_dkv(args.max_seqlen_k, args.d):
        launch(lib.magi_ffa_bwd_dv, k_groups, (q_ranges, k_ranges),
               "bwd_dv[det]", head_splits=hs)
        launch(lib.magi_ffa_bwd_dk, k_groups, (q_ranges, k_ranges),
               "bwd_dk[det]", head_splits=hs)
    else:
        launch(lib.magi_ffa_bwd_dkv, k_groups, (q_ranges, k_ranges),
               "bwd_dkv[det]", head_splits=hs)


def run_bwd_passes(args, device, dq_tables=None, dkv_tables=None) -> None:
    """Launch the independent backward passes (dq / fused-dkv by default;
    dq / dv / dk with MAGI_BWD_SPLIT_DKV=1) on two streams so their waves
    co-schedule across the chip. dq_tables / dkv_tables override the range
    tables per pass for auto_range_merge: (ranges_outer, ranges_inner,
    attn_type_map, seg_starts) — outer = the pass's OWN loop dim."""
    lib = _ffa_lib.lib()
    main = torch.cuda.current_stream(device)
    cosched = env.is_bwd_cosched()
    side = _get_side_stream(device) if cosched else main
    if cosched:
        ev = torch.cuda.Event()
        ev.record(main)          # dpsum + inputs ready
        side.wait_event(ev)
    keep = (dq_tables, dkv_tables)  # outlive the async launches

    def set_tables(t, outer_is_q):
        if t is None:
            args.seg_starts = None
            return
        outer, inner, tm, starts = t
        args.q_ranges = ptr(outer if outer_is_q else inner)
        args.k_ranges = ptr(inner if outer_is_q else outer)
        args.attn_type_map = ptr(tm)
        args.seg_starts = ptr(starts)

    args.stream = ctypes.c_void_p(side.cuda_stream)
    set_tables(dq_tables, True)
    check(lib.magi_ffa_bwd_dq(args), "magi_ffa_bwd_dq")
    args.stream = ctypes.c_void_p(main.cuda_stream)
    set_tables(dkv_tables, False)
    if env.is_bwd_split_dkv(args.max_seqlen_k, args.d):
        check(lib.magi_ffa_bwd_dv(args), "magi_ffa_bwd_dv")
        check(lib.magi_ffa_bwd_dk(args), "magi_ffa_bwd_dk")
    else:
        check(lib.magi_ffa_bwd_dkv(args), "magi_ffa_bwd_dkv")
    if cosched:
        ev2 = torch.cuda.Event()
        ev2.record(side)
        main.wait_event(ev2)
    del keep

from .. import _ffa_lib, env
from .._ffa_lib import (
    MagiFfaBwdArgs,
    MagiFfaFwdArgs,
    check,
    current_stream_ptr,
    ptr,
)
from ..common.forward_meta import AttnForwardMeta

import contextlib


@contextlib.contextmanager
def maybe_profile_ffa_ctx(name: str, enable: bool = False):
    """Optional named-event timing around an FFA call (reference
    flex_flash_attn.py maybe_profile_ffa_ctx; magi_attn_ext event timers)."""
    if not enable:
        yield
        return
    from .. import magi_attn_ext

    magi_attn_ext.start_event(name)
    try:
        yield
    finally:
        magi_attn_ext.stop_event(name)


def merge_ranges(
    outer_ranges: torch.Tensor,
    inner_ranges: torch.Tensor,
    attn_type_map: torch.Tensor,
):
    """Sort + deduplicate (outer, inner) attention block pairs (reference
    flex_flash_attn.py:79 merge_ranges, built on the magi_attn_ext ops).
    Returns (merged_outer [n,2] zero-padded, sorted_outer, sorted_inner,
    sorted_attn_type_map, range_map inverse indices, unique_count [1])."""
    from .. import magi_attn_ext

    idx = magi_attn_ext.argsort_ranges(outer_ranges)
    so, si, st = magi_attn_ext.reorder_ranges_and_attn_type_maps(
        outer_ranges, inner_ranges, attn_type_map, idx
    )
    uniq, inverse, count = magi_attn_ext.unique_consecutive_pairs(so)
    n = outer_ranges.shape[0]
    merged = torch.zeros(n, 2, dtype=torch.int32, device=outer_ranges.device)
    merged[: uniq.shape[0]] = uniq
    return merged, so, si, st, inverse, count


def _seg_starts(inverse: torch.Tensor, n: int) -> torch.Tensor:
    """[n+1] cumulative segment starts from sorted inverse indices; entries
    past the unique count repeat the total, giving EMPTY segments — so the
    kernel grid can stay sized by n with no host sync on unique_count."""
    counts = torch.bincount(inverse.long(), minlength=n)
    starts = torch.zeros(n + 1, dtype=torch.int32, device=inverse.device)
    starts[1:] = counts.cumsum(0).to(torch.int32)
    return starts


LOCK_GRAN = 128
_lock_cache: dict[tuple[int, int], torch.Tensor] = {}


def maybe_contiguous(x: Optional[torch.Tensor]) -> Optional[torch.Tensor]:
    return x.contiguous() if x is not None and not x.is_contiguous() else x


def _get_locks(total_q: int, hq: int, device: torch.device) -> torch.Tensor:
    """Lock words for the fwd merge epilogue. Kernel always releases to 0, so
    the buffer can be cached and reused without re-zeroing."""
    slots = (total_q + LOCK_GRAN - 1) // LOCK_GRAN
    key = (device.index or 0, 0)
    buf = _lock_cache.get(key)
    need = slots * hq
    if buf is None or buf.numel() < need:
        buf = torch.zeros(max(need, 1 << 16), dtype=torch.int32, device=device)
        _lock_cache[key] = buf
    return buf


def _max_seqlen_of(ranges: torch.Tensor) -> int:
    if ranges.numel() == 0:
        return 0
    return int((ranges[:, 1] - ranges[:, 0]).amax().item())


def _color_ranges(ranges: list) -> list:
    """Greedy interval partitioning: split slice indices into groups whose
    ranges are pairwise DISJOINT. Deterministic mode runs one kernel launch
    per group, sequentially on the stream, so reduction order is fixed by
    construction — placement-independent, unlike device-side lock ordering
    (the reference's deterministic.h approach assumes dispatch order, which
    CDNA4 does not guarantee)."""
    order = sorted(range(len(ranges)), key=lambda i: (ranges[i][0], ranges[i][1]))
    groups: list = []   # list of (last_end, [indices])
    for i in order:
        s_, e_ = ranges[i]
        placed = False
        for g in groups:
            if g[0] <= s_:
                g[1].append(i)
                g[0] = e_
                placed = True
                break
        if not placed:
            groups.append([e_, [i]])
    return [sorted(g[1]) for g in groups]


def _subset(t: Optional[torch.Tensor], idx: torch.Tensor) -> Optional[torch.Tensor]:
    return None if t is None else t.index_select(0, idx)


def _flex_flash_attn_forward(
    q: torch.Tensor,
    k: torch.Tensor,
    v: torch.Tensor,
    sink: Optional[torch.Tensor],
    sink_layout: str,
    out: Optional[torch.Tensor],
    lse: Optional[torch.Tensor],
    q_ranges: torch.Tensor,
    k_ranges: torch.Tensor,
    attn_type_map: Optional[torch.Tensor],
    softmax_scale: float,
    softcap: float,
    out_type: Optional[torch.dtype],
    disable_fwd_atomic_reduction: bool,
    deterministic: bool,
    sm_margin: int,
    max_seqlen_q: Optional[int] = None,
    max_logits: Optional[torch.Tensor] = None,
    qk_starts: Optional[torch.Tensor] = None,
    **_unused,
) -> tuple[torch.Tensor, AttnForwardMeta]:
    is_fp8 = q.dtype == torch.float8_e4m3fn
    assert q.dtype in (torch.bfloat16, torch.float8_e4m3fn), (
        "bf16 or fp8-e4m3 inputs"
    )
    q, k, v, q_ranges, k_ranges = [
        maybe_contiguous(x) for x in (q, k, v, q_ranges, k_ranges)
    ]
    attn_type_map = maybe_contiguous(attn_type_map)

    tq, hq, d = q.shape
    tk, hk, _ = k.shape

    if out is None:
        dtype = out_type or (q.dtype if disable_fwd_atomic_reduction else torch.float32)
        out = torch.zeros(tq, hq, d, dtype=dtype, device=q.device)
    if lse is None:
        lse = torch.full((tq, hq), float("-inf"), dtype=torch.float32, device=q.device)

    out_is_fp32 = out.dtype == torch.float32
    if not disable_fwd_atomic_reduction:
        assert out_is_fp32, "atomic (merge) forward requires an fp32 out accumulator"
        locks = _get_locks(tq, hq, q.device)
    else:
        assert out.dtype in (torch.float32, torch.bfloat16)
        locks = None

    if max_seqlen_q is None:
        max_seqlen_q = _max_seqlen_of(q_ranges)

    args = MagiFfaFwdArgs(
        q=ptr(q), k=ptr(k), v=ptr(v), out=ptr(out), lse=ptr(lse),
        q_ranges=ptr(q_ranges), k_ranges=ptr(k_ranges),
        attn_type_map=ptr(attn_type_map), locks=ptr(locks),
        max_logits=ptr(max_logits), qk_starts=ptr(qk_starts),
        n_ranges=q_ranges.shape[0], total_q=tq, total_k=tk,
        hq=hq, hk=hk, d=d, max_seqlen_q=max_seqlen_q,
        softmax_scale=softmax_scale, softcap=softcap,
        out_is_fp32=int(out_is_fp32),
        disable_atomic_reduction=int(disable_fwd_atomic_reduction),
        cu_margin=sm_margin, stream=current_stream_ptr(),
    )
    fwd_fn = (_ffa_lib.lib().magi_ffa_fwd_fp8 if is_fp8
              else _ffa_lib.lib().magi_ffa_fwd)
    if is_fp8:
        assert not disable_fwd_atomic_reduction and out_is_fp32
    if qk_starts is not None:
        assert not deterministic, (
            "deterministic + auto_range_merge lands in a later round"
        )
        assert not is_fp8, "fp8 + auto_range_merge lands in a later round"
    if deterministic and q_ranges.shape[0] > 1:
        # fixed merge order: one launch per q-disjoint slice group.
        # NOTE: subset tensors must outlive the async kernel launches — hold
        # references until the end of the loop (the ctypes kernel is invisible
        # to torch's stream-aware allocator).
        groups = _color_ranges(q_ranges.cpu().tolist())
        keep = []
        for g in groups:
            idx = torch.tensor(g, dtype=torch.long, device=q.device)
            sq = _subset(q_ranges, idx)
            sk = _subset(k_ranges, idx)
            st = _subset(attn_type_map, idx)
            keep.append((sq, sk, st))
            args.q_ranges = ptr(sq)
            args.k_ranges = ptr(sk)
            args.attn_type_map = ptr(st)
            args.n_ranges = len(g)
            check(fwd_fn(args), "magi_ffa_fwd[det]")
        del keep
    else:
        check(fwd_fn(args), "magi_ffa_fwd")
    if sink is not None:
        _apply_sink_postprocess(out, lse, sink, sink_layout, tq, hq, d)
    return out, AttnForwardMeta(lse=lse, max_logits=max_logits)


def _check_sink(sink, sink_layout, tq, hq):
    assert sink_layout in ("sh", "ssh"), f"unsupported sink_layout {sink_layout}"
    sink = sink.contiguous().float()
    if sink_layout == "sh":
        assert sink.dim() == 2 and sink.shape[1] == hq, "sink must be [s_sink, hq]"
        s_sink = sink.shape[0]
    else:
        assert sink.dim() == 3 and sink.shape[0] == tq and sink.shape[2] == hq, (
            "sink must be [total_q, s_sink, hq]"
        )
        s_sink = sink.shape[1]
    assert 1 <= s_sink <= 8, "seqlen_sink must be in [1, 8] (reference kMaxSeqlenSink)"
    return sink, s_sink


def _apply_sink_postprocess(out, lse, sink, sink_layout, tq, hq, d):
    """Fold the sink logits into (out, lse) ONCE, after all slice merges
    (reference flash_fwd_postprocess_kernel.h:39)."""
    sink, s_sink = _check_sink(sink, sink_layout, tq, hq)
    from .._ffa_lib import MagiSinkArgs

    args = MagiSinkArgs(
        out=ptr(out), lse=ptr(lse), sink=ptr(sink),
        total_rows=tq, n_heads=hq, d=d, s_sink=s_sink,
        ssh=int(sink_layout == "ssh"),
        out_is_fp32=int(out.dtype == torch.float32),
        stream=current_stream_ptr(),
    )
    check(_ffa_lib.lib().magi_ffa_sink_postprocess(args), "sink_postprocess")


def _flex_flash_attn_forward_index(
    q: torch.Tensor,
    k: torch.Tensor,
    v: torch.Tensor,
    indices_2d: torch.Tensor,
    softmax_scale: float,
    softcap: float,
) -> tuple[torch.Tensor, AttnForwardMeta]:
    """Index-attention (token-gather) forward: the reference's
    index_attn_indices direct-to-kernel path (flex_flash_attn.py:1358-1390).
    indices_2d [total_q, max_topk] int32 lists the GLOBAL K rows each q token
    row attends, shared by all hq query heads; -1 = contiguous tail padding.
    Forward only (the reference backward receives None for index_attn)."""
    from .._ffa_lib import MagiFfaIndexArgs

    assert q.dtype == torch.bfloat16, "index_attn requires bf16 inputs"
    q, k, v, indices_2d = [maybe_contiguous(x) for x in (q, k, v, indices_2d)]
    tq, hq, d = q.shape
    tk, hk, _ = k.shape
    assert hk == 1, (
        "index_attn expects KV heads folded into the K row dimension "
        "(k of shape [total_k, 1, d]; global id = (b*S_kv + t)*NHK + h)"
    )
    assert indices_2d.dtype == torch.int32 and indices_2d.dim() == 2
    assert indices_2d.shape[0] == tq, (
        f"indices_2d rows ({indices_2d.shape[0]}) must match q token rows ({tq})"
    )
    max_topk = indices_2d.shape[1]

    out = torch.zeros(tq, hq, d, dtype=q.dtype, device=q.device)
    lse = torch.full((tq, hq), float("-inf"), dtype=torch.float32, device=q.device)

    args = MagiFfaIndexArgs(
        q=ptr(q), k=ptr(k), v=ptr(v), out=ptr(out), lse=ptr(lse),
        indices_2d=ptr(indices_2d),
        total_q=tq, total_k=tk, max_topk=max_topk,
        hq=hq, hk=hk, d=d,
        softmax_scale=softmax_scale, softcap=softcap,
        out_is_fp32=0, stream=current_stream_ptr(),
    )
    check(_ffa_lib.lib().magi_ffa_fwd_index(args), "magi_ffa_fwd_index")
    return out, AttnForwardMeta(lse=lse, max_logits=None)


def _flex_flash_attn_backward(
    dout: torch.Tensor,
    q: torch.Tensor,
    k: torch.Tensor,
    v: torch.Tensor,
    sink: Optional[torch.Tensor],
    sink_layout: str,
    out: torch.Tensor,
    lse: torch.Tensor,
    dq: Optional[torch.Tensor],
    dk: Optional[torch.Tensor],
    dv: Optional[torch.Tensor],
    dsink: Optional[torch.Tensor],
    q_ranges: torch.Tensor,
    k_ranges: torch.Tensor,
    attn_type_map: Optional[torch.Tensor],
    softmax_scale: float,
    softcap: float,
    dq_type: Optional[torch.dtype],
    dk_type: Optional[torch.dtype],
    dv_type: Optional[torch.dtype],
    disable_bwd_dkv_atomic_reduction: bool,
    deterministic: bool,
    sm_margin: int,
    max_seqlen_k: Optional[int] = None,
    auto_range_merge: bool = False,
    **_unused,
) -> tuple[torch.Tensor, torch.Tensor, torch.Tensor, Optional[torch.Tensor]]:
    dout, q, k, v, out, q_ranges, k_ranges = [
        maybe_contiguous(x) for x in (dout, q, k, v, out, q_ranges, k_ranges)
    ]
    attn_type_map = maybe_contiguous(attn_type_map)
    tq, hq, d = q.shape
    tk, hk, _ = k.shape

    # fp32 accumulators, atomically reduced by the kernel
    dq = torch.zeros_like(q, dtype=torch.float32) if dq is None else dq
    dk = torch.zeros_like(k, dtype=torch.float32) if dk is None else dk
    dv = torch.zeros_like(v, dtype=torch.float32) if dv is None else dv
    assert dq.dtype == dk.dtype == dv.dtype == torch.float32
    dpsum = torch.empty(tq, hq, dtype=torch.float32, device=q.device)

    if max_seqlen_k is None:
        max_seqlen_k = _max_seqlen_of(k_ranges)

    args = MagiFfaBwdArgs(
        dout=ptr(dout), q=ptr(q), k=ptr(k), v=ptr(v), out=ptr(out),
        lse=ptr(lse), dq=ptr(dq), dk=ptr(dk), dv=ptr(dv), dpsum=ptr(dpsum),
        q_ranges=ptr(q_ranges), k_ranges=ptr(k_ranges),
        attn_type_map=ptr(attn_type_map),
        n_ranges=q_ranges.shape[0], total_q=tq, total_k=tk,
        hq=hq, hk=hk, d=d, max_seqlen_k=max_seqlen_k,
        out_is_fp32=int(out.dtype == torch.float32),
        softmax_scale=softmax_scale, softcap=softcap,
        cu_margin=sm_margin, stream=current_stream_ptr(),
    )
    lib = _ffa_lib.lib()
    check(lib.magi_ffa_bwd_preprocess(args), "magi_ffa_bwd_preprocess")
    if sink is not None:
        # dsink needs only (sink, lse, dpsum); the q/k/v gradients pick the
        # sink up automatically through the corrected lse saved by forward
        sink_f, s_sink = _check_sink(sink, sink_layout, tq, hq)
        if dsink is None:
            dsink = torch.zeros_like(sink_f)
        from .._ffa_lib import MagiSinkArgs

        sargs = MagiSinkArgs(
            lse=ptr(lse), sink=ptr(sink_f), dsink=ptr(dsink), dpsum=ptr(dpsum),
            total_rows=tq, n_heads=hq, d=d, s_sink=s_sink,
            ssh=int(sink_layout == "ssh"), stream=current_stream_ptr(),
        )
        check(lib.magi_ffa_dsink(sargs), "magi_ffa_dsink")
    if deterministic:
        assert not auto_range_merge, (
            "deterministic + auto_range_merge lands in a later round"
        )
        run_bwd_deterministic(args, q_ranges, k_ranges, attn_type_map,
                              hq, hk, q.device)
    elif auto_range_merge:
        # dq pass: unique q ranges with k segments; dkv pass: unique k
        # ranges with q segments (reference bwd_kq_map)
        mq, _, sk_i, sq_t, inv_q, _ = merge_ranges(
            q_ranges, k_ranges, attn_type_map
        )
        qk_starts = _seg_starts(inv_q, q_ranges.shape[0])
        mk, _, sq_i, sk_t, inv_k, _ = merge_ranges(
            k_ranges, q_ranges, attn_type_map
        )
        kq_starts = _seg_starts(inv_k, k_ranges.shape[0])
        run_bwd_passes(
            args, q.device,
            dq_tables=(mq, sk_i, sq_t, qk_starts),
            dkv_tables=(mk, sq_i, sk_t, kq_starts),
        )
    else:
        run_bwd_passes(args, q.device)
    return dq, dk, dv, dsink


class FlexFlashAttnFunc(torch.autograd.Function):
    @staticmethod
    def forward(
        ctx,
        q, k, v, sink, sink_layout,
        q_ranges, k_ranges, attn_type_map,
        softmax_scale=None, softcap=0.0, deterministic=False, sm_margin=0,
        disable_fwd_atomic_reduction=False,
        disable_bwd_dkv_atomic_reduction=False,
        ref_block_size=None, max_seqlen_q=None,
        auto_range_merge=False, swap_ab=False, pack_gqa=False, cat_gqa=False,
        sparse_load=False, index_attn=False, swap_bwd_qk_loop=False,
        return_max_logits=False, index_attn_indices_2d=None,
        index_attn_max_topk=0, max_seqlen_k=None,
    ):
        softmax_scale = (
            q.shape[-1] ** (-0.5) if softmax_scale is None else softmax_scale
        )
        assert q_ranges is not None and k_ranges is not None
        assert q_ranges.size(0) == k_ranges.size(0)
        fwd_q_ranges, fwd_k_ranges, fwd_tm = q_ranges, k_ranges, attn_type_map
        fwd_qk_starts = None
        if auto_range_merge:
            assert attn_type_map is not None, (
                "auto_range_merge requires an explicit attn_type_map"
            )
            mq, _, sk_i, st, inv_q, _ = merge_ranges(
                q_ranges, k_ranges, attn_type_map
            )
            fwd_q_ranges, fwd_k_ranges, fwd_tm = mq, sk_i, st
            fwd_qk_starts = _seg_starts(inv_q, q_ranges.shape[0])
        max_logits = None
        if return_max_logits:
            assert q.shape[1] <= 128, "num_qheads must be <= 128 (reference cap)"
            max_logits = torch.full((q.shape[1],), float("-inf"),
                                    dtype=torch.float32, device=q.device)

        out, meta = _flex_flash_attn_forward(
            q=q, k=k, v=v, sink=sink, sink_layout=sink_layout,
            out=None, lse=None,
            q_ranges=fwd_q_ranges, k_ranges=fwd_k_ranges,
            attn_type_map=fwd_tm,
            softmax_scale=softmax_scale, softcap=softcap, out_type=None,
            disable_fwd_atomic_reduction=disable_fwd_atomic_reduction,
            deterministic=deterministic, sm_margin=sm_margin,
            max_seqlen_q=max_seqlen_q, max_logits=max_logits,
            qk_starts=fwd_qk_starts,
        )
        lse = meta.lse
        if q.dtype == torch.float8_e4m3fn:
            out = out.to(torch.bfloat16)  # fp8 extension returns bf16 out
        elif out.dtype != q.dtype:
            out = out.to(q.dtype)
        ctx.save_for_backward(q, k, v, out, lse, q_ranges, k_ranges,
                              attn_type_map, sink)
        ctx.softmax_scale = softmax_scale
        ctx.softcap = softcap
        ctx.deterministic = deterministic
        ctx.sm_margin = sm_margin
        ctx.sink_layout = sink_layout
        ctx.auto_range_merge = auto_range_merge
        # avoid a per-backward device sync to size the bwd grid
        ctx.max_seqlen_k = max_seqlen_k
        if max_logits is not None:
            ctx.mark_non_differentiable(max_logits)
        return out, lse, max_logits

    @staticmethod
    def backward(ctx, dout, *_):
        (q, k, v, out, lse, q_ranges, k_ranges, attn_type_map,
         sink) = ctx.saved_tensors
        in_dtype = q.dtype
        if in_dtype == torch.float8_e4m3fn:
            # fp8 backward runs on the bf16 kernels over upcast operands
            # (standard mixed-precision practice; fp8 MFMA backward is a
            # later-round item). Gradients are produced in bf16 precision and
            # cast to the input dtype at return, as autograd requires.
            q = q.to(torch.bfloat16)
            k = k.to(torch.bfloat16)
            v = v.to(torch.bfloat16)
            out = out.to(torch.bfloat16) if out.dtype == torch.float8_e4m3fn else out
            dout = dout.to(torch.bfloat16)
        dq, dk, dv, dsink = _flex_flash_attn_backward(
            dout=dout, q=q, k=k, v=v, sink=sink, sink_layout=ctx.sink_layout,
            out=out, lse=lse, dq=None, dk=None, dv=None, dsink=None,
            q_ranges=q_ranges, k_ranges=k_ranges, attn_type_map=attn_type_map,
            softmax_scale=ctx.softmax_scale, softcap=ctx.softcap,
            dq_type=None, dk_type=None, dv_type=None,
            disable_bwd_dkv_atomic_reduction=False,
            deterministic=ctx.deterministic, sm_margin=ctx.sm_margin,
            max_seqlen_k=ctx.max_seqlen_k,
            auto_range_merge=ctx.auto_range_merge,
        )
        dq = dq.to(in_dtype)
        dk = dk.to(in_dtype)
        dv = dv.to(in_dtype)
        if sink is not None and dsink is not None:
            dsink = dsink.to(sink.dtype)
        return (dq, dk, dv, dsink) + (None,) * 24


def flex_flash_attn_func(
    q: torch.Tensor,
    k: torch.Tensor,
    v: torch.Tensor,
    q_ranges: Optional[torch.Tensor] = None,
    k_ranges: Optional[torch.Tensor] = None,
    attn_type_map: Optional[torch.Tensor] = None,
    *,
    index_attn_indices: Optional[torch.Tensor] = None,
    q_block_size: int = 1,
    k_block_size: int = 1,
    sink: Optional[torch.Tensor] = None,
    sink_layout: str = "sh",
    softmax_scale: Optional[float] = None,
    softcap: float = 0.0,
    deterministic: bool = False,
    sm_margin: int = 0,
    disable_fwd_atomic_reduction: bool = False,
    disable_bwd_dkv_atomic_reduction: bool = False,
    ref_block_size: Optional[Tuple[int, int]] = None,
    max_seqlen_q: Optional[int] = None,
    auto_range_merge: bool = False,
    swap_ab: bool = False,
    pack_gqa: bool = False,
    cat_gqa: bool = False,
    sparse_load: bool = False,
    index_attn: bool = False,
    swap_bwd_qk_loop: bool = False,
    return_max_logits: bool = False,
    max_seqlen_k: Optional[int] = None,
) -> tuple[torch.Tensor, AttnForwardMeta]:
    """Single-GPU flex-flash-attention (drop-in for the reference
    flex_flash_attn_func, flex_flash_attn.py:1066; full mask semantics in the
    reference docstring :1247-1341). Returns (out, AttnForwardMeta(lse=...)).

    index_attn_indices (reference :1358-1390): (total_q, num_kv_heads,
    max_topk) int32 GLOBAL K row ids, -1 tail padding, mutually exclusive
    with q_ranges/k_ranges; max_topk must be a multiple of 64 (the MI355X
    staged-tile size; the reference requires 128, so any reference-legal
    input is accepted). Forward only."""
    # ── sparse-input validation (mirrors reference :1344-1384) ──
    _has_ranges = q_ranges is not None
    _has_index = index_attn_indices is not None
    assert int(_has_ranges) + int(_has_index) == 1, (
        "Exactly one of (q_ranges + k_ranges) or index_attn_indices must be "
        "provided"
    )
    assert not (sparse_load and _has_index), (
        "sparse_load and index_attn_indices are mutually exclusive"
    )
    if _has_index:
        assert index_attn_indices.dim() == 3, (
            f"index_attn_indices must be 3D (total_q, num_kv_heads, max_topk), "
            f"got shape {tuple(index_attn_indices.shape)}"
        )
        assert q_block_size == 1 and k_block_size == 1, (
            "only q_block_size=1 / k_block_size=1 (token granularity) is "
            "supported for index_attn"
        )
        max_topk = index_attn_indices.shape[2]
        assert max_topk % 64 == 0, (
            f"index_attn max_topk={max_topk} must be a multiple of 64 "
            f"(pad with -1)"
        )
        assert not (
            torch.is_grad_enabled()
            and any(t.requires_grad for t in (q, k, v))
        ), "index_attn is forward-only (matches the reference)"
        if softmax_scale is None:
            softmax_scale = q.shape[-1] ** (-0.5)
        d = q.shape[-1]
        if d not in (64, 128):
            # zero-pad odd head dims into the next bucket (same scheme as the
            # dense path below; exact for attention)
            assert d < 128, f"index_attn head_dim {d} > 128 unsupported"
            bucket = 64 if d <= 64 else 128
            pad = bucket - d
            out, meta = _flex_flash_attn_forward_index(
                torch.nn.functional.pad(q, (0, pad)),
                torch.nn.functional.pad(k, (0, pad)),
                torch.nn.functional.pad(v, (0, pad)),
                index_attn_indices.reshape(-1, max_topk),
                softmax_scale, softcap,
            )
            return out[..., :d], meta
        return _flex_flash_attn_forward_index(
            q, k, v, index_attn_indices.reshape(-1, max_topk),
            softmax_scale, softcap,
        )
    d = q.shape[-1]
    if d not in (64, 128, 192):
        # arbitrary head dims run in the next bucket with zero feature
        # padding — exact for attention (padding contributes 0 to scores;
        # padded V columns are sliced off; gradients slice back through the
        # autograd pad). Reference buckets head_size <=64/<=128/<=192
        # (flash_api.cpp:322-334); the 192 bucket lands with a D=192 kernel.
        assert d < 192, f"head_dim {d} > 192 is beyond the reference's buckets"
        assert q.dtype == torch.bfloat16, "padded head dims require bf16"
        bucket = 64 if d <= 64 else (128 if d <= 128 else 192)
        if softmax_scale is None:
            softmax_scale = d ** (-0.5)  # scale from the REAL head dim
        pad = bucket - d
        qp = torch.nn.functional.pad(q, (0, pad))
        kp = torch.nn.functional.pad(k, (0, pad))
        vp = torch.nn.functional.pad(v, (0, pad))
        out, meta = flex_flash_attn_func(
            qp, kp, vp, q_ranges, k_ranges, attn_type_map,
            sink=sink, sink_layout=sink_layout, softmax_scale=softmax_scale,
            softcap=softcap, deterministic=deterministic, sm_margin=sm_margin,
            disable_fwd_atomic_reduction=disable_fwd_atomic_reduction,
            max_seqlen_q=max_seqlen_q, max_seqlen_k=max_seqlen_k,
            auto_range_merge=auto_range_merge,
            return_max_logits=return_max_logits,
        )
        return out[..., :d], meta
    out, lse, max_logits = FlexFlashAttnFunc.apply(
        q, k, v, sink, sink_layout, q_ranges, k_ranges, attn_type_map,
        softmax_scale, softcap, deterministic, sm_margin,
        disable_fwd_atomic_reduction, disable_bwd_dkv_atomic_reduction,
        ref_block_size, max_seqlen_q, auto_range_merge, swap_ab, pack_gqa,
        cat_gqa, sparse_load, index_attn, swap_bwd_qk_loop, return_max_logits,
        None, 0, max_seqlen_k,
    )
    return out, AttnForwardMeta(lse=lse, max_logits=max_logits)
