"""AttnRectangles: container of AttnRectangle band regions (reference
surface: common/rectangles.py:31) — the list algebra the dynamic mask
representation builds on: bulk cuts, segment clips, union seqlens, areas."""
from __future__ import annotations

from typing import Any, Iterator, Union

from .enum import AttnMaskType
from .range import AttnRange
from .ranges import AttnRanges
from .rectangle import AttnRectangle

__all__ = ["AttnRectangles"]


class AttnRectangles:
    def __init__(self) -> None:
        self._rects: list[AttnRectangle] = []

    # ---------------- validity ----------------

    def is_valid(self) -> bool:
        return all(rect.is_valid() for rect in self._rects)

    def check_valid(self) -> None:
        if not self.is_valid():
            raise ValueError(f"Some of the {self._rects=} is invalid")

    # ---------------- construction ----------------

    def append(self, attn_rect: AttnRectangle, check: bool = False) -> None:
        if check:
            attn_rect.check_valid()
        self._rects.append(attn_rect)

    def extend(self, attn_rects: "AttnRectangles", check: bool = False) -> None:
        if check:
            attn_rects.check_valid()
        self._rects.extend(attn_rects._rects)

    @staticmethod
    def from_ranges(
        q_ranges,
        k_ranges,
        mask_types,
        check: bool = False,
    ) -> "AttnRectangles":
        """Build from parallel q/k range lists + mask types; empty ranges and
        degenerate bi-causal entries (q longer than k) are dropped."""
        qrs = AttnRanges.from_ranges(q_ranges, check)
        krs = AttnRanges.from_ranges(k_ranges, check)
        mts = [
            AttnMaskType.from_int_type(t) if isinstance(t, int) else t
            for t in mask_types
        ]
        assert len(qrs) == len(krs) == len(mts), (
            "q_ranges, k_ranges, mask_types length should be equal"
        )
        out = AttnRectangles()
        for qr, kr, mt in zip(qrs, krs, mts):
            if qr.is_empty() or kr.is_empty():
                continue
            if mt == AttnMaskType.BICAUSAL and qr.seqlen > kr.seqlen:
                continue
            out._rects.append(
                AttnRectangle(q_range=qr, k_range=kr, mask_type=mt)
            )
        if check:
            out.check_valid()
        return out

    # ---------------- unions / seqlens ----------------

    def get_qo_ranges_union(self) -> AttnRanges:
        u = AttnRanges()
        for rect in self._rects:
            u.append(rect.q_range)
        return u.merge()

    def get_kv_ranges_union(self) -> AttnRanges:
        u = AttnRanges()
        for rect in self._rects:
            u.append(rect.k_range)
        return u.merge()

    def total_seqlen_qo(self) -> int:
        return self.get_qo_ranges_union().total_seqlen

    def total_seqlen_kv(self) -> int:
        return self.get_kv_ranges_union().total_seqlen

    # ---------------- bulk cuts / clips ----------------

    def _split(self, cutter) -> tuple["AttnRectangles", "AttnRectangles"]:
        left, right = AttnRectangles(), AttnRectangles()
        for rect in self._rects:
            a, b = cutter(rect)
            if a is not None:
                left._rects.append(a)
            if b is not None:
                right._rects.append(b)
        return left, right

    def cut_q(self, cut_pos: int) -> tuple["AttnRectangles", "AttnRectangles"]:
        return self._split(lambda r: r.cut_q(cut_pos=cut_pos))

    def cut_k(self, cut_pos: int) -> tuple["AttnRectangles", "AttnRectangles"]:
        return self._split(lambda r: r.cut_k(cut_pos=cut_pos))

    def get_rects_within_q_segment(self, q_start: int, q_end: int) -> "AttnRectangles":
        out = AttnRectangles()
        for rect in self._rects:
            seg = rect.get_rect_within_q_segment(q_start, q_end)
            if seg is not None:
                out._rects.append(seg)
        return out

    def get_rects_within_k_segment(self, k_start: int, k_end: int) -> "AttnRectangles":
        out = AttnRectangles()
        for rect in self._rects:
            seg = rect.get_rect_within_k_segment(k_start, k_end)
            if seg is not None:
                out._rects.append(seg)
        return out

    # ---------------- measures ----------------

    def area(self) -> int:
        return sum(rect.area() for rect in self._rects)

    @property
    def size(self) -> int:
        return len(self._rects)

    def is_empty(self) -> bool:
        return len(self._rects) == 0

    # ---------------- dunder ----------------

    def __len__(self) -> int:
        return len(self._rects)

    def __getitem__(self, idx: int | slice):
        if isinstance(idx, slice):
            sub = AttnRectangles()
            sub._rects = list(self._rects[idx])
            return sub
        return self._rects[idx]

    def __setitem__(self, idx, value: Union[AttnRectangle, "AttnRectangles"]):
        if isinstance(idx, slice):
            assert isinstance(value, AttnRectangles) and (
                idx.stop - idx.start == len(value)
            )
            self._rects[idx] = value._rects
        else:
            assert isinstance(value, AttnRectangle)
            self._rects[idx] = value

    def __iter__(self) -> Iterator[AttnRectangle]:
        return iter(self._rects)

    def __eq__(self, other: Any) -> bool:
        if isinstance(other, AttnRectangles):
            return self._rects == other._rects
        return False

    def __hash__(self) -> int:
        return hash(tuple(self._rects))

    def __repr__(self) -> str:  # pragma: no cover
        if self.is_empty():
            return "[-1, -1) x [-1, -1): None"
        return f"{self._rects}"
