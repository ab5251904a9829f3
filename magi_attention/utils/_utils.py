"""General helpers (reference surface: utils/_utils.py — the subset that
the reference's tests, examples and experiments actually import, plus the
dtype/list/seed utilities they build on). Re-implemented for the MI355X
rebuild; torch.cuda here is ROCm/HIP."""
from __future__ import annotations

import hashlib
import os
import random
from typing import Any, Callable, Iterable, Sequence, TypeAlias, Union

import numpy as np
import torch
import torch.distributed as dist

__all__ = [
    "ceil_div",
    "rprint_rank",
    "write_rank",
    "setup_dist_env",
    "clearup_dist_env",
    "seqlens2cu_seqlens",
    "cu_seqlens2seqlens",
    "flatten_nested_list",
    "perm_idxs2unperm_idxs",
    "wrap_to_list",
    "is_list_value_all",
    "is_list_value_any",
    "is_list_type_all",
    "pad_and_pack_tensors",
    "get_factors",
    "transpose_matrix",
    "make_slice_mask_from_ffa_attn_type",
    "make_attn_mask_from_ffa_args",
    "fp_dtype_bits",
    "is_fp_dtype_at_least",
    "to_higher_fp_dtype",
    "max_fp_dtype",
    "argmin",
    "argmax",
    "argsort",
    "str2seed",
    "set_random_seed",
    "is_same_process_group",
    "missing_dependency",
    "NestedIntList",
]

NestedIntList: TypeAlias = Union[list[int], tuple[int, ...], Sequence["NestedIntList"]]


def ceil_div(a: int, b: int) -> int:
    return -(-a // b)


def rprint_rank(msg: str, rank: int | None = None, width: int = 50) -> None:  # pragma: no cover
    if rank is None or dist.get_rank() == rank:
        r = dist.get_rank()
        header = f"\n{'-' * width}{' ' * 5}rank={r}{' ' * 5}{'-' * width}\n\n"
        try:
            from rich import print as _print
        except ImportError:
            _print = print
        _print(header + msg, flush=True)


def write_rank(msg: str, path: str, rank: int | None = None, width: int = 50) -> None:  # pragma: no cover
    if rank is None or dist.get_rank() == rank:
        r = dist.get_rank()
        with open(path, "a") as f:
            f.write(f"\n{'-' * width}{' ' * 5}rank={r}{' ' * 5}{'-' * width}\n\n" + msg)


def setup_dist_env(
    backend: str = "nccl",
    base_seed: int | None = None,
    seed_bias: Callable = lambda rank: 0,
):
    """torchrun-style env setup; returns (rank, local_rank, world_size,
    num_nodes, num_local_ranks, world_group, device, seed)."""
    num_nodes = int(os.environ.get("NNODES", "1"))
    num_local_ranks = int(os.environ.get("NPROC_PER_NODE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    world_size = int(os.environ.get("WORLD_SIZE", "1"))

    torch.cuda.set_device(local_rank)
    device = torch.cuda.current_device()
    dist.init_process_group(backend=backend, rank=rank, world_size=world_size)

    seed = None
    if base_seed is not None:
        seed = base_seed + seed_bias(rank)
        set_random_seed(seed)
    return (rank, local_rank, world_size, num_nodes, num_local_ranks,
            dist.group.WORLD, device, seed)


def clearup_dist_env() -> None:
    dist.destroy_process_group()


def seqlens2cu_seqlens(seqlens: list[int]) -> list[int]:
    out = [0]
    for s in seqlens:
        out.append(out[-1] + s)
    return out


def cu_seqlens2seqlens(cu_seqlens: list[int]) -> list[int]:
    return [b - a for a, b in zip(cu_seqlens, cu_seqlens[1:])]


def flatten_nested_list(nested_list: NestedIntList) -> list[int]:
    flat: list[int] = []
    stack = list(nested_list[::-1])
    while stack:
        item = stack.pop()
        if isinstance(item, (list, tuple)):
            stack.extend(item[::-1])
        else:
            flat.append(item)  # type: ignore[arg-type]
    return flat


def perm_idxs2unperm_idxs(perm_idxs: list[int]) -> list[int]:
    inv = [0] * len(perm_idxs)
    for i, p in enumerate(perm_idxs):
        inv[p] = i
    return inv


def wrap_to_list(x: Any, broadcast_to_length: int = 1) -> list[Any]:
    return list(x) if isinstance(x, (list, tuple)) else [x] * broadcast_to_length


def _list_check(vals, val, just_same, allow_empty, agg):
    if len(vals) == 0:
        return allow_empty
    if just_same:
        assert val is None, "val should be None when just_same is True"
        val = vals[0]
    return agg(x == val for x in vals)


def is_list_value_all(_list, val=None, just_same=False, allow_empty=False) -> bool:
    return _list_check(_list, val, just_same, allow_empty, all)


def is_list_value_any(_list, val=None, just_same=False, allow_empty=False) -> bool:
    return _list_check(_list, val, just_same, allow_empty, any)


def is_list_type_all(_list, _type=None, just_same=False, allow_empty=False) -> bool:
    if len(_list) == 0:
        return allow_empty
    if just_same:
        assert _type is None, "_type should be None when just_same is True"
        _type = type(_list[0])
    return all(isinstance(x, _type) for x in _list)


def pad_and_pack_tensors(
    tensors: list[torch.Tensor],
    target_length: int,
    padding_value: float = 0.0,
    dtype: torch.dtype | None = None,
    device: torch.device | None = None,
) -> torch.Tensor:
    """Right-pad 1-D tensors to target_length and stack into [n, target]."""
    if not tensors:
        return torch.empty(0, target_length, dtype=dtype, device=device)
    dtype = dtype or tensors[0].dtype
    device = device or tensors[0].device
    out = torch.full((len(tensors), target_length), padding_value,
                     dtype=dtype, device=device)
    for i, t in enumerate(tensors):
        if t.dim() != 1:
            raise ValueError(f"Input tensor at index {i} is not 1D: {t.dim()}D")
        if t.numel() > target_length:
            raise ValueError(
                f"Tensor at index {i} has length {t.numel()}, which is "
                f"greater than target_length {target_length}. "
                "Cannot pad to a smaller length."
            )
        out[i, : t.numel()] = t
    return out


def get_factors(x: int) -> list[int]:
    return [i for i in range(1, x + 1) if x % i == 0]


def transpose_matrix(matrix: list[list[Any]]) -> list[list[Any]]:
    return [list(row) for row in zip(*matrix)]


def make_slice_mask_from_ffa_attn_type(
    seqlen_q: int,
    seqlen_k: int,
    attn_type_idx: int = 0,
    device: str | int = "cuda",
) -> torch.Tensor:
    """Dense bool mask of one slice for the FFA attn type (0 full, 1 causal
    bottom-right, 2 inv-causal top-left, 3 bi-causal)."""
    if attn_type_idx not in (0, 1, 2, 3):
        raise ValueError(f"Invalid attn_type_idx={attn_type_idx}")
    n = max(seqlen_q, seqlen_k)
    ones = torch.ones((n, n), dtype=torch.bool, device=device)
    if attn_type_idx == 0:
        return ones[:seqlen_q, :seqlen_k]
    br = torch.tril(ones)[n - seqlen_q :, n - seqlen_k :]
    tl = torch.triu(ones)[:seqlen_q, :seqlen_k]
    return br if attn_type_idx == 1 else (tl if attn_type_idx == 2 else br & tl)


def make_attn_mask_from_ffa_args(
    q_ranges,
    k_ranges,
    attn_type_map: list[int],
    total_seqlen_q: int,
    total_seqlen_k: int,
    device: str | int = "cuda",
) -> torch.Tensor:
    """Dense bool mask [tq, tk] from FFA (q_ranges, k_ranges, attn_type_map).
    Assignment (overwrite) semantics per slice, like the reference."""
    mask = torch.zeros((total_seqlen_q, total_seqlen_k), dtype=torch.bool,
                       device=device)
    for qr, kr, t in zip(q_ranges, k_ranges, attn_type_map):
        mask[qr.start : qr.end, kr.start : kr.end] = (
            make_slice_mask_from_ffa_attn_type(qr.seqlen, kr.seqlen, t, device)
        )
    return mask


def fp_dtype_bits(dtype: torch.dtype) -> int:
    if dtype == getattr(torch, "float4_e2m1fn_x2", None):
        return 4  # packed pair of fp4 values per byte
    return torch.finfo(dtype).bits


def is_fp_dtype_at_least(tensor: torch.Tensor, lowest_precision: torch.dtype) -> bool:
    return torch.finfo(tensor.dtype).bits >= torch.finfo(lowest_precision).bits


def to_higher_fp_dtype(tensor: torch.Tensor, lowest_precision: torch.dtype) -> torch.Tensor:
    if not is_fp_dtype_at_least(tensor, lowest_precision):
        return tensor.to(lowest_precision)
    return tensor


def max_fp_dtype(*dtypes: torch.dtype) -> torch.dtype:
    return max(dtypes, key=lambda d: torch.finfo(d).bits)


def argmin(iterable: Iterable[Any], key: Callable = lambda x: x) -> int:
    return min(enumerate(iterable), key=lambda p: key(p[1]))[0]


def argmax(iterable: Iterable[Any], key: Callable = lambda x: x) -> int:
    return max(enumerate(iterable), key=lambda p: key(p[1]))[0]


def argsort(iterable: Iterable[Any], key: Callable = lambda x: x) -> list[int]:
    return [i for i, _ in sorted(enumerate(iterable), key=lambda p: key(p[1]))]


def str2seed(s: str) -> int:
    h = int(hashlib.sha256(s.encode("utf-8")).hexdigest(), 16)
    return h % (2**32)


def set_random_seed(seed: int) -> None:
    random.seed(seed)
    np.random.seed(seed)
    torch.manual_seed(seed)
    torch.cuda.manual_seed(seed)


def is_same_process_group(a, b) -> bool:
    """True when both process-group handles denote the same group (either
    identical objects, or both None = default group)."""
    if a is b:
        return True
    if a is None or b is None:
        try:
            return (a or dist.group.WORLD) is (b or dist.group.WORLD)
        except Exception:
            return False
    return False


def missing_dependency(func_name: str, dep_name: str):  # pragma: no cover
    def _raise(*args, **kwargs):
        raise ImportError(
            f"`{func_name}` requires optional dependency `{dep_name}`, "
            f"but it is not installed."
        )

    return _raise
