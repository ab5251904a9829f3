# DispatchSolver — chunk -> rank assignment balancing per-chunk mask AREA
# (reference meta/solver/dispatch_solver.py:578; MinHeap alg :61-358).
from __future__ import annotations

import heapq
from dataclasses import dataclass
from typing import List, Sequence

from ...common.enum import DispatchAlgType
from ...config import (  # noqa: F401  (re-exported for API parity)
    DispatchAlg,
    DispatchConfig,
    MinHeapDispatchAlg,
    SequentialDispatchAlg,
    ToppHeapDispatchAlg,
)


@dataclass
class DispatchSolution:
    partitions: List[List[int]]  # rank -> sorted chunk ids
    loads: List[float]


class DispatchSolver:
    """Assign num_chunks chunks (with given workloads) to cp_size ranks,
    each receiving exactly num_chunks // cp_size chunks, minimising the max
    per-rank total workload."""

    def __init__(self, alg: DispatchAlg | None = None):
        self.alg = alg or MinHeapDispatchAlg()

    def solve(
        self, workloads: Sequence[float], cp_size: int,
        affinities: "List | None" = None,
        uneven_shard: bool = False,
    ) -> DispatchSolution:
        """affinities: optional per-chunk AttnRanges (the k rows the chunk's
        mask slices touch) for the affinity tie-breaks of TOPP_HEAP
        (reference dispatch_solver.py:477 IOUAffinity: prefer the rank whose
        accumulated coverage overlaps the chunk most — fewer distinct remote
        rows => less group-cast traffic). uneven_shard: per-rank chunk counts
        may differ by one (reference DispatchConfig.uneven_shard)."""
        n = len(workloads)
        if not uneven_shard:
            assert n % cp_size == 0, f"{n} chunks not divisible by cp {cp_size}"
        cap = [(n + cp_size - 1 - r) // cp_size if uneven_shard
               else n // cp_size for r in range(cp_size)]
        t = self.alg.type
        if t == DispatchAlgType.SEQUENTIAL_SELECT:
            parts, loads = [], []
            c0 = 0
            for r in range(cp_size):
                parts.append(list(range(c0, c0 + cap[r])))
                c0 += cap[r]
                loads.append(sum(workloads[c] for c in parts[r]))
            return DispatchSolution(parts, loads)
        if t == DispatchAlgType.SORTED_SEQUENTIAL_SELECT:
            # sort desc, deal in snake order (reference :328) — cheap LPT-ish
            order = sorted(range(n), key=lambda c: -workloads[c])
            parts = [[] for _ in range(cp_size)]
            loads = [0.0] * cp_size
            i = 0
            fwd = True
            for c in order:
                tries = 0
                while len(parts[i]) >= cap[i] and tries <= 2 * cp_size:
                    i, fwd = self._snake_next(i, fwd, cp_size)
                    tries += 1
                parts[i].append(c)
                loads[i] += workloads[c]
                i, fwd = self._snake_next(i, fwd, cp_size)
            for p in parts:
                p.sort()
            return DispatchSolution(parts, loads)
        if t == DispatchAlgType.TOPP_HEAP and affinities is not None:
            # MinHeap capacity-LPT with an IOU-affinity tie-break (reference
            # dispatch_solver.py:990 _solve_with_topphp): candidates = the
            # top max(1, ceil(cp*top_p)) least-loaded ranks with capacity;
            # pick the one with the largest coverage overlap
            import math

            top_p = getattr(self.alg, "top_p", 0.5)
            m = max(1, math.ceil(cp_size * top_p))
            order = sorted(range(n), key=lambda c: -workloads[c])
            parts = [[] for _ in range(cp_size)]
            loads = [0.0] * cp_size
            acc = [None] * cp_size  # accumulated AttnRanges per rank
            for c in order:
                avail = [r for r in range(cp_size) if len(parts[r]) < cap[r]]
                cand = sorted(avail, key=lambda r: loads[r])[:m]
                best, best_ov = cand[0], -1.0
                for r in cand:
                    ov = (_intersect_size(acc[r], affinities[c])
                          if acc[r] is not None else 0.0)
                    if ov > best_ov:
                        best, best_ov = r, ov
                parts[best].append(c)
                loads[best] += workloads[c]
                acc[best] = _merge_aff(acc[best], affinities[c])
            for p in parts:
                p.sort()
            return DispatchSolution(parts, loads)
        # MIN_HEAP / LOWER_BOUND (default, and fallback): greedy LPT with
        # capacity — sort chunks by workload desc, assign each to the
        # least-loaded rank that still has capacity.
        order = sorted(range(n), key=lambda c: -workloads[c])
        heap = [(0.0, r) for r in range(cp_size)]
        heapq.heapify(heap)
        parts: List[List[int]] = [[] for _ in range(cp_size)]
        loads = [0.0] * cp_size
        for c in order:
            # pop until a rank with capacity
            popped = []
            while True:
                load, r = heapq.heappop(heap)
                if len(parts[r]) < cap[r]:
                    break
                popped.append((load, r))
            parts[r].append(c)
            loads[r] = load + workloads[c]
            heapq.heappush(heap, (loads[r], r))
            for item in popped:
                heapq.heappush(heap, item)
        for p in parts:
            p.sort()
        return DispatchSolution(parts, loads)

    @staticmethod
    def _snake_next(i: int, fwd: bool, cp: int):
        if fwd:
            if i + 1 < cp:
                return i + 1, True
            return i, False
        if i > 0:
            return i - 1, False
        return i, True


def _intersect_size(acc, rr) -> float:
    if acc is None or rr is None:
        return 0.0
    tot = 0
    for p in rr.find_overlap_ranges(acc):
        tot += p.seqlen
    return float(tot)


def _merge_aff(acc, rr):
    from ...common.ranges import AttnRanges

    u = AttnRanges()
    if acc is not None:
        for p in acc:
            u.append(p.clone())
    if rr is not None:
        for p in rr:
            u.append(p.clone())
    return u.merge()
