# dispatch/undispatch of activations along the CP permutation
# (reference functional/dispatch.py:193 dispatch_func, :224 undispatch_func,
#  _DispatchFunc:123: forward = local chunk select with NO communication,
#  backward = all-gather-v + unpermute :45; undispatch is the inverse :166).
from __future__ import annotations

from typing import List

import torch
import torch.distributed as dist

from ..comm.primitive import all_gather_v


def _positions(partitions: List[List[int]], chunk_size: int, rank: int,
               device) -> torch.Tensor:
    idx = []
    for c in partitions[rank]:
        idx.extend(range(c * chunk_size, (c + 1) * chunk_size))
    return torch.tensor(idx, dtype=torch.long, device=device)


class _DispatchFunc(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, partitions, chunk_size, group):
        rank = dist.get_rank(group)
        pos = _positions(partitions, chunk_size, rank, x.device)
        ctx.partitions = partitions
        ctx.chunk_size = chunk_size
        ctx.group = group
        ctx.total = x.shape[0]
        return x.index_select(0, pos).contiguous()

    @staticmethod
    def backward(ctx, grad):
        group = ctx.group
        world = dist.get_world_size(group)
        sizes = [len(ctx.partitions[r]) * ctx.chunk_size for r in range(world)]
        gathered = all_gather_v(grad.contiguous(), sizes, group)
        dx = grad.new_zeros((ctx.total, *grad.shape[1:]))
        cursor = 0
        for r in range(world):
            pos = _positions(ctx.partitions, ctx.chunk_size, r, grad.device)
            dx.index_copy_(0, pos, gathered[cursor:cursor + sizes[r]])
            cursor += sizes[r]
        return dx, None, None, None


class _UndispatchFunc(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x_local, partitions, chunk_size, group):
        rank = dist.get_rank(group)
        world = dist.get_world_size(group)
        sizes = [len(partitions[r]) * chunk_size for r in range(world)]
        gathered = all_gather_v(x_local.contiguous(), sizes, group)
        total = sum(sizes)
        full = x_local.new_zeros((total, *x_local.shape[1:]))
        cursor = 0
        for r in range(world):
            pos = _positions(partitions, chunk_size, r, x_local.device)
            full.index_copy_(0, pos, gathered[cursor:cursor + sizes[r]])
            cursor += sizes[r]
        ctx.partitions = partitions
        ctx.chunk_size = chunk_size
        ctx.group = group
        return full

    @staticmethod
    def backward(ctx, grad):
        rank = dist.get_rank(ctx.group)
        pos = _positions(ctx.partitions, ctx.chunk_size, rank, grad.device)
        return grad.index_select(0, pos).contiguous(), None, None, None


def dispatch_func(x, partitions, chunk_size, group):
    return _DispatchFunc.apply(x, partitions, chunk_size, group)


def undispatch_func(x_local, partitions, chunk_size, group):
    return _UndispatchFunc.apply(x_local, partitions, chunk_size, group)
