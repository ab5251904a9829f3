"""FlagCombGenerator (reference surface: testing/flag_generator.py:25):
generates the "valuable" subset of a flag-combination space for tests —
defaults first, then the all-non-default point, then one-hot deviations,
then per-group sweeps and random fill — instead of the full product."""
from __future__ import annotations

import itertools
import math
import random
from typing import Any, Callable, Generator, Literal

import torch.distributed as dist

FlagCombStrategy = Literal["constant", "sequential", "random", "heuristic"]


class FlagCombGenerator:
    def __init__(
        self,
        flags: list[str],
        options: dict[str, list[Any]] = {},
        defaults: dict[str, Any] = {},
        groups: list[tuple[str, ...]] = [],
        strategy: FlagCombStrategy = "heuristic",
        cycle_times: int = -1,
    ):
        self.flags = list(dict.fromkeys(flags).keys())
        self.options = {f: options.get(f, [False, True]) for f in self.flags}
        self.defaults = {
            f: defaults.get(f, self.options[f][0]) for f in self.flags
        }
        for f in self.flags:
            dv = self.defaults[f]
            assert dv in self.options[f], (
                f"The default value for flag={f} ({dv}) must be in the "
                f"options ({self.options[f]})"
            )
            # default first, keep relative order of the rest
            self.options[f] = [dv] + [v for v in self.options[f] if v != dv]
        self.groups = groups
        self.strategy = strategy
        assert cycle_times > 0 or cycle_times == -1, (
            f"`cycle_times` must be greater than 0 or -1, got {cycle_times}"
        )
        self.cycle_times = cycle_times
        self.comb_set: set[tuple[Any, ...]] = set()
        self._internal_iter = self.iter()
        self._deferred_combs: list[dict[str, Any]] = []

    # ---------------- info ----------------

    @property
    def num_flags(self) -> int:
        return len(self.flags)

    @property
    def num_combs(self) -> int:
        return math.prod(len(v) for v in self.options.values())

    def is_comb_covered(self, comb: tuple[Any, ...]) -> bool:
        return self._key(comb) in self.comb_set

    @classmethod
    def to_test_case(cls, flag_comb: dict[str, Any]) -> str:
        parts = []
        for flag, value in flag_comb.items():
            shown = value["name"] if isinstance(value, dict) and "name" in value else value
            parts.append(f"{flag}=[{shown}]")
        return " x ".join(parts)

    @classmethod
    def sync_group(
        cls, flag_comb: dict[str, Any], group: dist.ProcessGroup
    ) -> dict[str, Any]:
        """Broadcast rank 0's combination so every rank tests the same one."""
        obj = [flag_comb if dist.get_rank(group) == 0 else None]
        dist.broadcast_object_list(obj, group=group, group_src=0)
        return obj[0]

    def get_next_valid_comb(
        self,
        test_config: dict[str, Any],
        is_valid_fn: Callable[[dict[str, Any], dict[str, Any]], bool],
    ) -> dict[str, Any]:
        """Next combination legal under test_config; illegal draws are
        deferred for later configs rather than consumed."""
        for i, comb in enumerate(self._deferred_combs):
            if is_valid_fn(comb, test_config):
                return self._deferred_combs.pop(i)
        attempts = 0
        while True:
            try:
                comb = next(self._internal_iter)
            except StopIteration:
                raise RuntimeError(
                    f"FlagCombGenerator exhausted: none of the "
                    f"{len(self._deferred_combs)} deferred combo(s) is valid "
                    f"for {test_config=}."
                ) from None
            if is_valid_fn(comb, test_config):
                return comb
            self._deferred_combs.append(comb)
            attempts += 1
            if attempts > self.num_combs:
                raise RuntimeError(
                    f"FlagCombGenerator: no valid flag combination found "
                    f"after {attempts} draws for {test_config=}; deferred "
                    f"{len(self._deferred_combs)} combo(s)."
                )

    # ---------------- iteration ----------------

    def __iter__(self) -> Generator[dict[str, Any], None, None]:
        return self.iter()

    def __reversed__(self):
        return self.iter(reverse=True)

    def iter(self, reverse: bool = False) -> Generator[dict[str, Any], None, None]:
        cycles = itertools.count() if self.cycle_times == -1 else range(self.cycle_times)
        for _ in cycles:
            yield from self._iter(reverse)

    def _iter(self, reverse: bool = False) -> Generator[dict[str, Any], None, None]:
        self.comb_set = set()
        if self.strategy == "constant":
            yield from self._emit(tuple(self.defaults.values()))
        elif self.strategy == "sequential":
            combs = itertools.product(*self.options.values())
            for comb in reversed(list(combs)) if reverse else combs:
                if not self.is_comb_covered(comb):
                    yield from self._emit(comb)
        elif self.strategy == "random":
            yield from self._iter_random()
        elif self.strategy == "heuristic":
            yield from self._iter_heuristic(reverse)
        else:
            raise ValueError(f"Unknown strategy {self.strategy}")

    def _iter_random(
        self,
        fixed_values: dict[str, Any] = {},
        max_iter_times: int = -1,
    ) -> Generator[dict[str, Any], None, None]:
        drawn = 0
        while len(self.comb_set) < self.num_combs:
            comb = tuple(
                fixed_values.get(f, random.choice(self.options[f]))
                for f in self.flags
            )
            yield from self._emit(comb)
            drawn += 1
            if 0 < max_iter_times <= drawn:
                break

    def _iter_heuristic(
        self, reverse: bool = False
    ) -> Generator[dict[str, Any], None, None]:
        base = tuple(self.defaults.values())
        yield from self._emit(base)  # 1. all defaults
        flipped = tuple(opts[-1] for opts in self.options.values())
        if not self.is_comb_covered(flipped):  # 2. all non-defaults
            yield from self._emit(flipped)
        # 3. one-hot deviations from the default point
        for i, f in enumerate(self.flags):
            for opt in self.options[f]:
                if opt == base[i]:
                    continue
                comb = base[:i] + (opt,) + base[i + 1 :]
                if not self.is_comb_covered(comb):
                    yield from self._emit(comb)
        # 4. per-group full sweeps (random fill outside the group)
        sweeps = [
            itertools.product(*[self.options[f] for f in g]) for g in self.groups
        ]
        for row in itertools.zip_longest(*sweeps):
            fixed: dict[str, Any] = {}
            for g, values in zip(self.groups, row):
                if values is not None:
                    fixed.update(dict(zip(g, values)))
            yield from self._iter_random(fixed_values=fixed, max_iter_times=1)
        # 5. random fill until the space is covered
        yield from self._iter_random()

    # ---------------- internals ----------------

    @staticmethod
    def _key(comb: tuple[Any, ...]) -> tuple[Any, ...]:
        return tuple(id(v) if isinstance(v, (dict, list)) else v for v in comb)

    def _emit(self, comb: tuple[Any, ...]) -> Generator[dict[str, Any], None, None]:
        self.comb_set.add(self._key(comb))
        yield dict(zip(self.flags, comb))
