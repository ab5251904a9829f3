"""Public test harness (reference surface: magi_attention/testing/ —
parameterize, assert_close, ref_attn_func, GroundTruthDispatcher,
dist_common, utils)."""
from __future__ import annotations

import functools
import itertools
import os
from typing import Any, Callable

import torch.distributed as dist

from ..utils import str2seed
from . import dist_common, utils
from .dist_common import RUN_IN_MP
from .gt_dispatcher import GroundTruthDispatcher
from .precision import assert_close
from .ref_attn import ref_attn_func

__all__ = [
    "dist_common",
    "utils",
    "GroundTruthDispatcher",
    "assert_close",
    "ref_attn_func",
    "parameterize",
]


def parameterize(argument: str, values: list[Any]) -> Callable:
    """pytest.mark.parametrize analogue with multi-process awareness
    (reference testing/__init__.py:40). Stacked decorators compose into one
    cartesian product; the OUTERMOST one runs every combination, fail-fast.
    With RUN_IN_MP=1 inside an initialized process group, combinations are
    sharded across ranks by a hash of the combination (replication is the
    default, since collective tests need every rank on every case)."""

    def _wrapper(func: Callable):
        inner_params = getattr(func, "_param_info", [])
        all_params = [(argument, values)] + inner_params
        original_func = getattr(func, "_original_func", func)

        @functools.wraps(func)
        def _parameterized_func(*args, **kwargs):
            if dist.is_available() and dist.is_initialized():
                rank, world, is_dist = dist.get_rank(), dist.get_world_size(), True
            else:
                rank, world, is_dist = 0, 1, False
            shard = is_dist and os.environ.get(RUN_IN_MP, "0") == "1"

            arg_names = [n for n, _ in all_params]
            value_lists = [v for _, v in all_params]
            for combination in itertools.product(*value_lists):
                if shard and str2seed(str(combination)) % world != rank:
                    continue
                case_kwargs = dict(zip(arg_names, combination))
                try:
                    original_func(*args, **{**kwargs, **case_kwargs})
                except Exception as e:
                    details = []
                    for name, vlist in all_params:
                        val = case_kwargs[name]
                        try:
                            idx = vlist.index(val)
                        except ValueError:
                            idx = "?"
                        shown = (
                            val[dist_common.NAME]
                            if isinstance(val, dict) and dist_common.NAME in val
                            else val
                        )
                        details.append(f"      {name}[{idx}] = {shown}")
                    msg = "".join(
                        [
                            "\n-->",
                            f" [Rank {rank}] " if is_dist else " ",
                            "Test case failed:\n",
                            "    Parameters:\n" + "\n".join(details) + "\n",
                            f"    Error: {type(e).__name__}: {e}",
                        ]
                    )
                    raise type(e)(msg) from e

        _parameterized_func._param_info = all_params  # type: ignore[attr-defined]
        _parameterized_func._original_func = original_func  # type: ignore[attr-defined]
        return _parameterized_func

    return _wrapper
