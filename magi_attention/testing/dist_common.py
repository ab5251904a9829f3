"""Multi-process test scaffolding (reference surface:
testing/dist_common.py): DistTestBase over torch's MultiProcessTestCase,
with_comms/with_run_in_mp wrappers and the env-var test-case filters.
On this target PG_DEFAULT_BACKEND resolves to RCCL ("nccl" on ROCm) when
GPUs are present, gloo otherwise."""
from __future__ import annotations

import datetime
import os
import sys
from fnmatch import fnmatch
from functools import wraps
from typing import Any, Callable

import torch
import torch.distributed as dist

try:  # torch's MultiProcessTestCase needs the optional `expecttest` package
    from torch.testing._internal.common_distributed import (
        TIMEOUT_OVERRIDE,
        TEST_SKIPS,
        MultiProcessTestCase,
    )

    _MP_IMPORT_ERROR = None
except ImportError as _e:  # pragma: no cover
    _MP_IMPORT_ERROR = _e
    TIMEOUT_OVERRIDE: dict = {}  # type: ignore[no-redef]
    TEST_SKIPS: dict = {}  # type: ignore[no-redef]

    class MultiProcessTestCase:  # type: ignore[no-redef]
        def __init__(self, *a, **k):
            raise ImportError(
                "DistTestBase requires torch.testing._internal."
                "common_distributed (optional dependency `expecttest` is "
                f"missing): {_MP_IMPORT_ERROR}"
            )

from ..utils import set_random_seed
from .utils import switch_envvar_decorator

NAME = "name"
SKIP_WORLD_SIZE = "skip_world_size"
PROFILE_ONLY = "profile_only"
INTERFACE = "interface"

TEST_ATTN_CONFIG = "MAGI_ATTENTION_TEST_ATTN_CONFIG"
_TEST_FILTER_ENV_PREFIX = "MAGI_ATTENTION_TEST_"
TEST_WORLD_SIZE = "MAGI_ATTENTION_TEST_WORLD_SIZE"

_TEST_FILTER_ENVVARS: dict[str, str] = {
    "attn_config": "MAGI_ATTENTION_TEST_ATTN_CONFIG",
    "overlap_config": "MAGI_ATTENTION_TEST_OVERLAP_CONFIG",
    "num_heads": "MAGI_ATTENTION_TEST_NUM_HEADS",
    "head_dim": "MAGI_ATTENTION_TEST_HEAD_DIM",
    "dtype": "MAGI_ATTENTION_TEST_DTYPE",
    "random_type_mapping": "MAGI_ATTENTION_TEST_RANDOM_TYPE_MAPPING",
}


def _match_patterns(value_str: str, raw_env: str) -> bool:
    patterns = [p.strip() for p in raw_env.split(",") if p.strip()]
    return any(fnmatch(value_str, pat) for pat in patterns)


def should_run_world_size(world_size: int) -> bool:
    """MAGI_ATTENTION_TEST_WORLD_SIZE: unset = run all; else a
    comma-separated allow-list of world sizes."""
    raw = os.environ.get(TEST_WORLD_SIZE, "").strip()
    if not raw:
        return True
    return world_size in {int(s.strip()) for s in raw.split(",") if s.strip()}


def skip_if_world_size_filtered(func):
    """Subprocess-level skip when self.world_size is filtered out (reports
    as skipped, like skip_if_lt_x_gpu)."""

    @wraps(func)
    def wrapper(self, *args, **kwargs):
        if should_run_world_size(self.world_size):
            return func(self, *args, **kwargs)
        sys.exit(TEST_SKIPS["generic"].exit_code)

    return wrapper


def should_run_attn_config(name: str) -> bool:
    raw = os.environ.get(TEST_ATTN_CONFIG, "").strip()
    if not raw:
        return True
    return _match_patterns(name, raw)


def _case_value_str(value: object) -> str:
    """Canonical string of one parametrize value for filtering: named
    dicts match by their NAME entry, tuples join with underscores."""
    if isinstance(value, dict) and NAME in value:
        return str(value[NAME])
    if isinstance(value, tuple):
        return "_".join(str(v) for v in value)
    return str(value)


def should_run_test_case(**parametrize_args: object) -> bool:
    """AND-combined fnmatch filters per parametrize dimension, driven by
    MAGI_ATTENTION_TEST_<DIMENSION> env vars."""
    for dim_name, value in parametrize_args.items():
        envvar = _TEST_FILTER_ENVVARS.get(
            dim_name, _TEST_FILTER_ENV_PREFIX + dim_name.upper()
        )
        raw = os.environ.get(envvar, "").strip()
        if raw and not _match_patterns(_case_value_str(value), raw):
            return False
    return True


DEVICE_TYPE = (
    "cuda" if torch.cuda.is_available() and torch.cuda.device_count() > 1 else "cpu"
)
PG_DEFAULT_BACKEND = "nccl" if DEVICE_TYPE == "cuda" else "gloo"
NUM_DEVICES = 4

RUN_IN_MP = "MAGI_ATTENTION_PARAMETERIZE_RUN_IN_MP"


_SUPPORTED_BACKENDS = ("nccl", "gloo", "mpi", "cpu:gloo,cuda:nccl")


class DistTestBase(MultiProcessTestCase):
    """Multi-process test base: spawns world_size workers, each of which
    calls init_pg() (via with_comms) to join a file-rendezvous process group
    with a rank-offset seed. On ROCm the "nccl" backend is RCCL."""

    seed = property(lambda self: 42)
    world_size = property(lambda self: NUM_DEVICES)
    backend = property(lambda self: PG_DEFAULT_BACKEND)

    def init_pg(self) -> None:
        be = self.backend
        if "nccl" in be and torch.cuda.device_count() < self.world_size:
            raise RuntimeError(
                f"nccl backend requires {self.world_size} GPUs, but only "
                f"{torch.cuda.device_count()} are available"
            )
        if be not in _SUPPORTED_BACKENDS:
            raise RuntimeError(f"Backend {be} not supported!")
        dist.init_process_group(
            backend=be,
            world_size=self.world_size,
            rank=self.rank,
            init_method=f"file://{self.file_name}",
            timeout=datetime.timedelta(minutes=30),
        )
        if "nccl" in be:
            torch.cuda.set_device(self.rank)
        self._set_random_seed()

    def destroy_pg(self) -> None:
        dist.barrier()
        dist.destroy_process_group()

    def _set_random_seed(self) -> None:
        set_random_seed(self.seed + self.rank)

    def setUp(self) -> None:
        super().setUp()
        timeout = getattr(self, "timeout", None)
        if timeout is not None:
            TIMEOUT_OVERRIDE.update({self.id().split(".")[-1]: timeout})
        self._spawn_processes()


TestFunc = Callable[..., Any]


def with_comms(func: TestFunc) -> TestFunc:
    assert func is not None

    @wraps(func)
    def wrapper(self, *args: tuple[object], **kwargs: dict[str, Any]) -> None:
        if torch.cuda.is_available() and torch.cuda.device_count() >= self.world_size:
            self.device_type = "cuda"
        else:
            self.device_type = "cpu"
        self.init_pg()
        func(self, *args, **kwargs)
        self.destroy_pg()

    return wrapper


def with_run_in_mp(func: TestFunc) -> TestFunc:
    """with_comms + the parameterize case-distribution mode enabled."""
    return switch_envvar_decorator(envvar_name=RUN_IN_MP, enable=True)(
        with_comms(func)
    )
