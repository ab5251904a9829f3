# Rolled re-dispatch of an already-dispatched tensor (reference
# functional/roll.py:448 roll_p2p / :422 roll_simple_p2p): shifts the PADDED
# global sequence cyclically without materialising the full tensor — peak
# memory O(N/P) and each row crosses the wire exactly once.
#
# MI355X-native realisation: instead of the reference's batched isend/irecv
# ring, the permutation is packed into ONE all_to_all_single (RCCL over
# xGMI spreads the multi-destination traffic across the 7 p2p links); the
# run tables are derived deterministically on every rank from DispatchMeta.
from __future__ import annotations

from collections import defaultdict

import torch
import torch.distributed as dist

from ..comm.primitive import _a2av, _rows_copy
from ..meta.containers import RowChunkMap


def _roll_tables(meta, shift: int):
    """Per-pair contiguous-run tables for rolling the padded global sequence
    by `shift`. Returns (send_pack, in_splits, recv_unpack, out_splits)."""
    ck = meta.chunk_size
    parts = meta.partitions
    T = meta.total_seqlen
    W = meta.cp_size
    rank = meta.cp_rank
    shift %= T
    owner = {}
    loff = {}
    for r, pl in enumerate(parts):
        for i, c in enumerate(pl):
            owner[c] = r
            loff[c] = i * ck

    # send: my source rows -> destination ranks, runs split at chunk borders
    send = defaultdict(list)  # dst rank -> [(src_local, run, src_global)]
    for i, c in enumerate(parts[rank]):
        off = 0
        while off < ck:
            g = c * ck + off
            d = (g + shift) % T
            dc = d // ck
            run = min(ck - off, (dc + 1) * ck - d)
            send[owner[dc]].append((i * ck + off, run, g))
            off += run
    # recv: runs arriving from each source rank, which the sender packed in
    # ITS src-local order — reproduce that order here for the unpack table
    recv = defaultdict(list)  # src rank -> [(src_local, run, dst_local)]
    for i, c in enumerate(parts[rank]):
        off = 0
        while off < ck:
            d = c * ck + off
            s = (d - shift) % T
            sc = s // ck
            run = min(ck - off, (sc + 1) * ck - s)
            recv[owner[sc]].append(
                (loff[sc] + (s - sc * ck), run, i * ck + off)
            )
            off += run

    in_r, out_s, in_splits = [], [], []
    cur = 0
    for r in range(W):
        tok = 0
        for sl, run, _g in sorted(send.get(r, [])):
            in_r.append((sl, sl + run))
            out_s.append(cur)
            cur += run
            tok += run
        in_splits.append(tok)
    send_pack = RowChunkMap(in_r, out_s, cur)

    un_r, un_s, out_splits = [], [], []
    rcur = 0
    for r in range(W):
        runs = sorted(recv.get(r, []))  # sender's src-local order
        tok = sum(x[1] for x in runs)
        for _sl, run, dl in runs:
            un_r.append((rcur, rcur + run))
            un_s.append(dl)
            rcur += run
        out_splits.append(tok)
    recv_unpack = RowChunkMap(un_r, un_s, rcur)
    return send_pack, in_splits, recv_unpack, out_splits


def _roll_once(x_local, shift, meta, group, seq_dim):
    assert seq_dim == 0, "seq_dim != 0 lands in a later round"
    send_pack, in_sp, recv_unpack, out_sp = _roll_tables(meta, shift)
    tail = x_local.shape[1:]
    send = x_local.new_empty((send_pack.total_rows, *tail))
    _rows_copy(x_local, send, send_pack)
    recv = x_local.new_empty((sum(out_sp), *tail))
    _a2av(recv, send, out_sp, in_sp, group, async_op=False)
    out = torch.empty_like(x_local)
    _rows_copy(recv, out, recv_unpack)
    return out


class _RollP2P(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x_local, shift, meta, group, seq_dim):
        ctx.shift = shift
        ctx.meta = meta
        ctx.group = group
        ctx.seq_dim = seq_dim
        return _roll_once(x_local, shift, meta, group, seq_dim)

    @staticmethod
    def backward(ctx, grad):
        gx = _roll_once(grad.contiguous(), -ctx.shift, ctx.meta, ctx.group,
                        ctx.seq_dim)
        return gx, None, None, None, None


def roll_p2p(x_local, shift, meta, group, seq_dim: int = 0):
    """Reference functional/roll.py:448 (autograd-capable)."""
    return _RollP2P.apply(x_local, shift, meta, group, seq_dim)


def roll_simple_p2p(x_local, shift, meta, group, seq_dim: int = 0):
    """Reference functional/roll.py:422 — same semantics, same transport
    here (the a2av realisation already moves each row once)."""
    return _RollP2P.apply(x_local, shift, meta, group, seq_dim)
