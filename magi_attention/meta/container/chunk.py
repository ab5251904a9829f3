"""AttnChunk (reference meta/container/chunk.py): the slices intersecting
one fixed-size token chunk."""
from __future__ import annotations

from dataclasses import dataclass, field

from ...common.ranges import AttnRanges
from .slice import AttnSlice


@dataclass(repr=False)
class AttnChunk:
    chunk_id: int | None = None
    q_slices: list[AttnSlice] = field(default_factory=list)
    sample_ids: list[int] = field(default_factory=list)

    @property
    def q_ranges(self) -> AttnRanges:
        rr = AttnRanges()
        for s in self.q_slices:
            rr.append(s.q_range)
        return rr

    @property
    def k_ranges(self) -> AttnRanges:
        rr = AttnRanges()
        for s in self.q_slices:
            rr.append(s.k_range)
        return rr

    @property
    def attn_slices(self) -> list[AttnSlice]:
        return self.q_slices

    @property
    def area(self) -> int:
        return sum(s.area for s in self.q_slices)

    @property
    def iou(self) -> float:
        union = self.k_ranges.union_size()
        return self.k_ranges.intersect_size() / union if union else 0.0

    def iou_with(self, other: "AttnChunk") -> float:
        return self.k_ranges.intersect_size_with(
            other.k_ranges
        ) / self.k_ranges.union_size_with(other.k_ranges)

    def __eq__(self, other: object) -> bool:
        if not isinstance(other, AttnChunk):
            return False
        return self.chunk_id == other.chunk_id and self.q_slices == other.q_slices

    def __repr__(self, indent: str = "") -> str:  # pragma: no cover
        lines = [
            f"{indent}AttnChunk(chunk_id={self.chunk_id}, area={self.area}, "
            f"q_slices=["
        ]
        for i, s in enumerate(self.q_slices):
            tee = "└──" if i == len(self.q_slices) - 1 else "├──"
            lines.append(f"{indent}    {tee} {s}")
        lines.append(f"{indent}])")
        return "\n".join(lines)
