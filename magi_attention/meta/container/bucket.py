"""AttnBucket (reference meta/container/bucket.py): the chunks one cp rank
hosts, with aggregate areas and k-overlap IOU."""
from __future__ import annotations

from dataclasses import dataclass, field

from ...common.ranges import AttnRanges
from .chunk import AttnChunk
from .slice import AttnSlice


@dataclass(repr=False)
class AttnBucket:
    cp_rank: int | None = None
    q_chunks: list[AttnChunk] = field(default_factory=list)

    @property
    def q_ranges(self) -> AttnRanges:
        rr = AttnRanges()
        for c in self.q_chunks:
            rr.extend(c.q_ranges)
        return rr

    @property
    def k_ranges(self) -> AttnRanges:
        rr = AttnRanges()
        for c in self.q_chunks:
            rr.extend(c.k_ranges)
        return rr

    @property
    def attn_slices(self) -> list[AttnSlice]:
        out: list[AttnSlice] = []
        for c in self.q_chunks:
            out.extend(c.attn_slices)
        return out

    @property
    def area(self) -> int:
        return sum(c.area for c in self.q_chunks)

    @property
    def areas(self) -> list[int]:
        return [c.area for c in self.q_chunks]

    @property
    def iou(self) -> float:
        union = self.k_ranges.union_size()
        return self.k_ranges.intersect_size() / union if union else 0.0

    def iou_with(self, other: "AttnBucket") -> float:
        return self.k_ranges.intersect_size_with(
            other.k_ranges
        ) / self.k_ranges.union_size_with(other.k_ranges)

    def __eq__(self, other: object) -> bool:
        if not isinstance(other, AttnBucket):
            return False
        return self.cp_rank == other.cp_rank and self.q_chunks == other.q_chunks

    def __repr__(self, indent: str = "") -> str:  # pragma: no cover
        lines = [
            f"{indent}AttnBucket(cp_rank={self.cp_rank}, area={self.area}, "
            f"q_chunks=["
        ]
        for i, c in enumerate(self.q_chunks):
            tee = "└──" if i == len(self.q_chunks) - 1 else "├──"
            lines.append(f"{indent}    {tee} {c.__repr__(indent + '    ')}")
        lines.append(f"{indent}])")
        return "\n".join(lines)
