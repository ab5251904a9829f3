"""Autograd-coupled gather/scatter pairs (reference surface:
comm/functional/_gather_scatter_v.py) — the differentiable dispatch/undispatch
transport: all-gather forward pairs with scatter backward and vice versa.
Built over this rebuild's uneven all_gather_v (comm/primitive.py:342)."""
from __future__ import annotations

import torch
import torch.distributed as dist

from .. import primitive


def _gather(x: torch.Tensor, group, dim: int, split_sizes) -> torch.Tensor:
    world = dist.get_world_size(group)
    sizes = split_sizes if split_sizes is not None else [x.shape[dim]] * world
    moved = x.movedim(dim, 0) if dim != 0 else x
    out = primitive.all_gather_v(moved.contiguous(), list(sizes), group)
    return out.movedim(0, dim) if dim != 0 else out


def _scatter(x: torch.Tensor, group, dim: int, split_sizes) -> torch.Tensor:
    rank, world = dist.get_rank(group), dist.get_world_size(group)
    if split_sizes is None:
        return torch.chunk(x, chunks=world, dim=dim)[rank].contiguous()
    return torch.split(x, list(split_sizes), dim=dim)[rank].contiguous()


class _AllGatherFwdScatterBwd(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, group, dim, split_sizes):
        ctx.group, ctx.dim, ctx.split_sizes = group, dim, split_sizes
        return _gather(x, group, dim, split_sizes)

    @staticmethod
    def backward(ctx, grad):
        return _scatter(grad, ctx.group, ctx.dim, ctx.split_sizes), None, None, None


class _ScatterFwdAllGatherBwd(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, group, dim, split_sizes):
        ctx.group, ctx.dim, ctx.split_sizes = group, dim, split_sizes
        return _scatter(x, group, dim, split_sizes)

    @staticmethod
    def backward(ctx, grad):
        return _gather(grad, ctx.group, ctx.dim, ctx.split_sizes), None, None, None


def all_gather_fwd_scatter_bwd(
    input: torch.Tensor,
    group: dist.ProcessGroup,
    dim: int,
    split_sizes: list[int] | None = None,
) -> torch.Tensor:
    return _AllGatherFwdScatterBwd.apply(input, group, dim, split_sizes)


def scatter_fwd_all_gather_bwd(
    input: torch.Tensor,
    group: dist.ProcessGroup,
    dim: int,
    split_sizes: list[int] | None = None,
) -> torch.Tensor:
    return _ScatterFwdAllGatherBwd.apply(input, group, dim, split_sizes)


__all__ = ["all_gather_fwd_scatter_bwd", "scatter_fwd_all_gather_bwd"]
