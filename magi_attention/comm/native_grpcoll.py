"""Native group-collective transport: HIP-IPC peer windows + pull kernels
over xGMI (MAGI_ATTENTION_NATIVE_GRPCOLL=1).

Role of the reference's `magi_attn_comm` / csrc/comm/grpcoll intranode path
(buffer.cpp + intranode_kernel.cuh:46,573 — NVLink-IPC push channels with
head/tail token queues). MI355X-first PULL redesign (csrc/grpcoll.hip):

- every rank owns persistent WINDOW buffers (kv copy for the cast, partial
  dKV for the reduce) exported once via hipIpcGetMemHandle (dmabuf IPC) and
  opened by all peers (handles travel over one init-time
  all_gather_object);
- a cast = SDMA copy of kv into my window + a 1-block release-signal, then
  one bounded-grid pull kernel on the consumer gathers its planned row
  ranges straight out of the peers' windows (plan: NativeStageMeta);
- the reduce mirrors it with a fused fp32 sum into the local accumulator;
- ordering is entirely device-side (system-scope seq flags + ack counters
  for window reuse backpressure): no host sync between compute and comm.
"""
from __future__ import annotations

import ctypes
from typing import List, Optional

import torch
import torch.distributed as dist

from .. import _ffa_lib
from .._ffa_lib import MagiGrpCollPullArgs, check, ptr
from ..meta.containers import NativeStageMeta
from .primitive import WorkWithPostProcessFn

_HANDLE_BYTES = 64


def _get_handle(t: torch.Tensor):
    """(handle bytes, offset of the tensor inside its BASE allocation) —
    torch's caching allocator suballocates, and IPC handles name the base."""
    base = ctypes.c_void_p()
    size = ctypes.c_uint64()
    check(_ffa_lib.lib().magi_ipc_base(ptr(t), ctypes.byref(base),
                                       ctypes.byref(size)), "ipc_base")
    off = ptr(t).value - base.value
    buf = ctypes.create_string_buffer(_HANDLE_BYTES)
    check(_ffa_lib.lib().magi_ipc_get_handle(
        ctypes.c_void_p(base.value), ctypes.cast(buf, ctypes.c_void_p)),
        "ipc_get_handle")
    return buf.raw, off


# opening the SAME base handle twice in one process is an error — windows of
# one peer may share a caching-allocator block, so dedupe opens per process
_open_cache: dict = {}


def _open_handle(h) -> int:
    hb, off = h
    if hb not in _open_cache:
        buf = ctypes.create_string_buffer(hb, _HANDLE_BYTES)
        out = ctypes.c_void_p()
        check(_ffa_lib.lib().magi_ipc_open(
            ctypes.cast(buf, ctypes.c_void_p), ctypes.byref(out)), "ipc_open")
        _open_cache[hb] = out.value
    return _open_cache[hb] + off


def _stream_ptr() -> ctypes.c_void_p:
    return ctypes.c_void_p(torch.cuda.current_stream().cuda_stream)


class NativeGrpColl:
    """Per-(runtime, group) IPC window manager + pull launches."""

    def __init__(
        self,
        stages: List[NativeStageMeta],
        stage_tokens_all: List[List[int]],
        group: dist.ProcessGroup,
        kv_rows: int,               # 2*L of this rank
        h: int,
        d: int,
        dtype: torch.dtype,
    ):
        self.stages = stages
        self.group = group
        self.rank = dist.get_rank(group)
        self.world = dist.get_world_size(group)
        assert self.world <= 8, "native grpcoll: single node (<=8 peers)"
        self.h, self.d = h, d
        self.deg = len(stages)
        dev = torch.device("cuda", torch.cuda.current_device())
        # windows: kv (param dtype) + per-stage partial (fp32)
        self.kv_win = torch.empty(kv_rows, h, d, dtype=dtype, device=dev)
        part_rows = max(
            (2 * stage_tokens_all[s][self.rank] for s in range(self.deg)),
            default=0,
        )
        self.part_win = torch.empty(max(part_rows, 1), h, d,
                                    dtype=torch.float32, device=dev)
        # flags: [deg cast_seq][deg cast_ack][deg red_seq][deg red_ack]
        self.flags = torch.zeros(4 * max(self.deg, 1), dtype=torch.int32,
                                 device=dev)
        # separate sequence counters: cast rounds run every fwd AND bwd,
        # reduce rounds only in bwd — sharing one counter deadlocks the
        # reduce backpressure on its never-advanced ack counters
        self.seq_cast = 0
        self.seq_red = 0

        my = (_get_handle(self.kv_win), _get_handle(self.part_win),
              _get_handle(self.flags))
        allh: List = [None] * self.world
        dist.all_gather_object(allh, my, group=group)
        self.kv_ptrs, self.part_ptrs, self.flag_ptrs = [], [], []
        for r, (hk, hp, hf) in enumerate(allh):
            if r == self.rank:
                self.kv_ptrs.append(ptr(self.kv_win).value)
                self.part_ptrs.append(ptr(self.part_win).value)
                self.flag_ptrs.append(ptr(self.flags).value)
            else:
                self.kv_ptrs.append(_open_handle(hk))
                self.part_ptrs.append(_open_handle(hp))
                self.flag_ptrs.append(_open_handle(hf))

    def _flag(self, r: int, slot: int) -> int:
        return self.flag_ptrs[r] + 4 * slot

    def _pull(self, pieces_dev, n_pieces, peer_bases, flag_slot, wait_value,
              dst, reduce: bool, elem_size: int):
        a = MagiGrpCollPullArgs()
        a.pieces = ptr(pieces_dev).value
        a.n_pieces = n_pieces
        a.row_elems = self.h * self.d
        a.elem_size = elem_size
        for r in range(self.world):
            a.peer_ptrs[r] = peer_bases[r]
            a.peer_flags[r] = self._flag(r, flag_slot)
        a.wait_value = wait_value
        a.n_peers = self.world
        a.dst = ptr(dst).value
        a.reduce = 1 if reduce else 0
        a.stream = _stream_ptr().value
        check(_ffa_lib.lib().magi_grpcoll_pull(ctypes.byref(a)),
              "grpcoll_pull")

    def _signal(self, slot: int, value: int):
        check(_ffa_lib.lib().magi_grpcoll_signal(
            ctypes.c_void_p(self._flag(self.rank, slot)), value,
            _stream_ptr()), "grpcoll_signal")

    def _wait_acks(self, ack_slot: int, target: int):
        if target <= 0:
            return
        check(_ffa_lib.lib().magi_grpcoll_wait(
            ctypes.c_void_p(self._flag(self.rank, ack_slot)), target,
            _stream_ptr()), "grpcoll_wait")

    def _ack_peers(self, peers: List[int], ack_slot: int):
        if not peers:
            return
        arr = (ctypes.c_void_p * 8)()
        for i, r in enumerate(peers):
            arr[i] = self._flag(r, ack_slot)
        check(_ffa_lib.lib().magi_grpcoll_ack(
            ctypes.cast(arr, ctypes.c_void_p), len(peers), _stream_ptr()),
            "grpcoll_ack")

    # ---- cast: my kv -> window; pull my stage buffer from peers ----
    def cast(self, kv_local: torch.Tensor, s: int) -> WorkWithPostProcessFn:
        if s == 0:
            self.seq_cast += 1
        meta = self.stages[s]
        slot_seq, slot_ack = s, self.deg + s
        # backpressure: every consumer of my window acked the previous round
        self._wait_acks(slot_ack, meta.cast_consumers * (self.seq_cast - 1))
        self.kv_win[: kv_local.shape[0]].copy_(kv_local)
        self._signal(slot_seq, self.seq_cast)
        S = meta.stage_tokens
        stage = kv_local.new_zeros((2 * S, self.h, self.d))
        cast_dev, _ = meta.to_device(kv_local.device)
        if meta.cast_pieces:
            self._pull(cast_dev, len(meta.cast_pieces), self.kv_ptrs,
                       slot_seq, self.seq_cast, stage, False,
                       kv_local.element_size())
            srcs = sorted({p[0] for p in meta.cast_pieces})
            self._ack_peers(srcs, slot_ack)
        return WorkWithPostProcessFn(None, lambda: stage)

    # ---- reduce: my partial -> window; owners pull-sum from peers ----
    def reduce(self, partial: torch.Tensor, dst: torch.Tensor, s: int
               ) -> WorkWithPostProcessFn:
        if s == 0:
            self.seq_red += 1
        meta = self.stages[s]
        slot_seq, slot_ack = 2 * self.deg + s, 3 * self.deg + s
        self._wait_acks(slot_ack, meta.reduce_consumers * (self.seq_red - 1))
        if partial.shape[0] > 0:
            self.part_win[: partial.shape[0]].copy_(partial)
        self._signal(slot_seq, self.seq_red)
        _, red_dev = meta.to_device(dst.device)
        if meta.reduce_pieces:
            self._pull(red_dev, len(meta.reduce_pieces), self.part_ptrs,
                       slot_seq, self.seq_red, dst, True, 4)
            srcs = sorted({p[0] for p in meta.reduce_pieces})
            self._ack_peers(srcs, slot_ack)
        return WorkWithPostProcessFn(None, lambda: dst)
