# Group collectives on RCCL (reference comm/primitive/grpcoll/
# _group_collective.py:81 group_cast, :255 group_reduce, realised as the a2av
# impl _a2av_grpcoll_impl.py:69,228: pack -> dist.all_to_all_single -> unpack,
# async with a post-process hook comm/work.py WorkWithPostProcessFn).
#
# MI355X-native: pack/unpack are the HIP range kernels (csrc/range_ops.hip) on
# GPU and torch index copies on CPU (gloo tests); the wire is
# dist.all_to_all_single, which on ROCm is RCCL over xGMI.
from __future__ import annotations

from typing import Callable, List, Optional

import torch
import torch.distributed as dist

from ..meta.containers import GroupCastArg, GroupReduceArg, RowChunkMap


class WorkWithPostProcessFn:
    """An async collective handle plus the unpack step that completes it
    (reference comm/work.py)."""

    def __init__(self, work, post_process: Callable[[], torch.Tensor]):
        self._work = work
        self._post = post_process
        self._done = False
        self._result: Optional[torch.Tensor] = None

    def wait_post_process(self) -> torch.Tensor:
        if not self._done:
            if self._work is not None:
                self._work.wait()
            self._result = self._post()
            self._done = True
        return self._result


def _rows_copy(src: torch.Tensor, dst: torch.Tensor, cmap: RowChunkMap,
               reduce: str = "copy", src_is_gpu: bool = False):
    """dst rows [out_start..] (+)= src rows [in_range..]."""
    if not cmap.in_ranges:
        return dst
    if src.is_cuda:
        from ..common.range_op import range_gather, range_reduce

        in_r, out_s = cmap.to_device(src.device)
        if reduce == "copy":
            range_gather(src, in_r, out_s, cmap.total_rows, output=dst)
        else:
            range_reduce(src, dst, in_r, out_s, op="sum",
                         total_rows=cmap.total_rows)
        return dst
    for (a, b), o in zip(cmap.in_ranges, cmap.out_starts):
        if reduce == "copy":
            dst[o:o + (b - a)] = src[a:b]
        else:
            dst[o:o + (b - a)] += src[a:b]
    return dst


def group_cast(
    kv_local: torch.Tensor,  # [2L, h, d] (k rows then v rows)
    arg: GroupCastArg,
    group: dist.ProcessGroup,
    async_op: bool = True,
) -> WorkWithPostProcessFn:
    """Multicast my hosted K/V token rows to the ranks that need them this
    stage; returns a handle whose wait_post_process() yields the stage kv
    buffer [2S, h, d] in globally-sorted order."""
    h_tail = kv_local.shape[1:]
    send = kv_local.new_empty((arg.send_pack.total_rows, *h_tail))
    _rows_copy(kv_local, send, arg.send_pack)
    recv = kv_local.new_empty((sum(arg.output_split_sizes), *h_tail))
    if dist.get_world_size(group) == 1:
        work = None
        recv = send
    else:
        flat = 1
        for t in h_tail:
            flat *= t
        work = dist.all_to_all_single(
            recv.view(recv.shape[0], flat),
            send.view(send.shape[0], flat),
            output_split_sizes=arg.output_split_sizes,
            input_split_sizes=arg.input_split_sizes,
            group=group,
            async_op=async_op,
        )
        if not async_op:
            work = None

    def post() -> torch.Tensor:
        stage = kv_local.new_zeros((2 * arg.stage_tokens, *h_tail))
        _rows_copy(recv, stage, arg.recv_unpack)
        return stage

    return WorkWithPostProcessFn(work, post)


def group_reduce(
    partial: torch.Tensor,   # [2S, h, d] partial dK/dV of the stage buffer
    dst: torch.Tensor,       # [2L, h, d] local dK/dV accumulator (fp32)
    arg: GroupReduceArg,
    group: dist.ProcessGroup,
    async_op: bool = True,
) -> WorkWithPostProcessFn:
    """Return partial dK/dV rows to their owner ranks and sum-reduce them into
    the owner-local accumulator."""
    h_tail = partial.shape[1:]
    send = partial.new_empty((arg.send_pack.total_rows, *h_tail))
    _rows_copy(partial, send, arg.send_pack)
    recv = partial.new_empty((arg.total_recv, *h_tail))
    if dist.get_world_size(group) == 1:
        work = None
        recv = send
    else:
        flat = 1
        for t in h_tail:
            flat *= t
        work = dist.all_to_all_single(
            recv.view(recv.shape[0], flat),
            send.view(send.shape[0], flat),
            output_split_sizes=arg.output_split_sizes,
            input_split_sizes=arg.input_split_sizes,
            group=group,
            async_op=async_op,
        )
        if not async_op:
            work = None

    def post() -> torch.Tensor:
        _rows_copy(recv, dst, arg.recv_reduce, reduce="sum")
        return dst

    return WorkWithPostProcessFn(work, post)


def all_gather_v(
    local: torch.Tensor,
    sizes: List[int],
    group: dist.ProcessGroup,
) -> torch.Tensor:
    """Variable all-gather along dim 0 (reference comm/primitive/_all_gather_v.py)."""
    world = dist.get_world_size(group)
    if world == 1:
        return local
    tail = local.shape[1:]
    assert all(s == sizes[0] for s in sizes), "equal shards expected"
    parts = [local.new_empty((s, *tail)) for s in sizes]
    dist.all_gather(parts, local.contiguous(), group=group)
    return torch.cat(parts, dim=0)


def reduce_scatter_v(
    full: torch.Tensor,
    sizes: List[int],
    group: dist.ProcessGroup,
) -> torch.Tensor:
    world = dist.get_world_size(group)
    rank = dist.get_rank(group)
    if world == 1:
        return full
    dist.all_reduce(full, group=group)
    start = sum(sizes[:rank])
    return full[start:start + sizes[rank]].clone()
