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
        self, workloads: Sequence[float], cp_size: int
    ) -> DispatchSolution:
        n = len(workloads)
        assert n % cp_size == 0, f"{n} chunks not divisible by cp {cp_size}"
        per = n // cp_size
        if self.alg.type == DispatchAlgType.SEQUENTIAL_SELECT:
            parts = [list(range(r * per, (r + 1) * per)) for r in range(cp_size)]
            loads = [sum(workloads[c] for c in p) for p in parts]
            return DispatchSolution(parts, loads)
        # MIN_HEAP (default, and fallback for other alg names): greedy LPT
        # with capacity — sort chunks by workload desc, assign each to the
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
                if len(parts[r]) < per:
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
