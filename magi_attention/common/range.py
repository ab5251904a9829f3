# MI355X-native rebuild of the reference API surface
# (reference: magi_attention/common/range.py — names kept, fresh implementation).
from __future__ import annotations

from typing import Any, Tuple

NaiveRange = Tuple[int, int]


class RangeError(Exception):
    pass


class AttnRange:
    """A half-open token index range [start, end) used for attention metadata."""

    __slots__ = ("_start", "_end")

    def __init__(self, start: int, end: int) -> None:
        # reference semantics (common/range.py:61): start <= end; negative
        # indices are legal (AttnRectangle d_ranges are k-q diagonals)
        if end < start:
            raise RangeError(
                f"The attn_range {(start, end)} is invalid against the rule: "
                f"'start <= end'"
            )
        self._start = int(start)
        self._end = int(end)

    # -- properties -------------------------------------------------------
    # setters are UNCHECKED like the reference's: rectangle shrinking sets
    # temporarily-inverted bounds and re-validates afterwards
    @property
    def start(self) -> int:
        return self._start

    @start.setter
    def start(self, value: int) -> None:
        self._start = int(value)

    @property
    def end(self) -> int:
        return self._end

    @end.setter
    def end(self, value: int) -> None:
        self._end = int(value)

    # -- validity (reference common/range.py:238-281) ----------------------
    def is_valid_close(self, start: int | None = None, end: int | None = None) -> bool:
        start = self._start if start is None else start
        end = self._end if end is None else end
        return start <= end

    def is_valid_open(self, start: int | None = None, end: int | None = None) -> bool:
        start = self._start if start is None else start
        end = self._end if end is None else end
        return start < end

    def check_valid(self, start: int | None = None, end: int | None = None) -> None:
        if not self.is_valid_close(start, end):
            raise RangeError(
                f"The attn_range {(start, end)} is invalid against the rule: "
                f"'start <= end'"
            )

    @property
    def seqlen(self) -> int:
        return self._end - self._start

    # -- constructors ------------------------------------------------------
    @classmethod
    def from_range(cls, naive_range, check: bool = False) -> "AttnRange":
        return cls(naive_range[0], naive_range[1])

    def clone(self) -> "AttnRange":
        return AttnRange(self._start, self._end)

    # -- algebra ----------------------------------------------------------
    def to_naive_range(self) -> NaiveRange:
        return (self._start, self._end)

    def offset(self, offset: int) -> "AttnRange":
        return AttnRange(self._start + offset, self._end + offset)

    def truncate(self, start: int | None = None, end: int | None = None) -> "AttnRange":
        s = self._start if start is None else max(self._start, start)
        e = self._end if end is None else min(self._end, end)
        if e < s:
            s = e = min(max(s, 0), self._end)
            return AttnRange(0, 0)
        return AttnRange(s, e)

    def intersect(self, other: "AttnRange") -> "AttnRange":
        s = max(self._start, other._start)
        e = min(self._end, other._end)
        return AttnRange(s, e) if e > s else AttnRange(0, 0)

    def intersect_size(self, other: "AttnRange") -> int:
        return max(0, min(self._end, other._end) - max(self._start, other._start))

    def union(self, other: "AttnRange") -> list["AttnRange"]:
        """One merged range when the two overlap or nest, else both
        (reference common/range.py union; adjacency without overlap keeps
        both, like the reference)."""
        if self.is_empty() or other.is_empty():
            return [self, other]
        if self.is_subrange_of(other):
            return [other]
        if other.is_subrange_of(self):
            return [self]
        if self.is_overlap_with(other):
            return [AttnRange(min(self._start, other._start),
                              max(self._end, other._end))]
        return [self, other]

    def union_size(self, other: "AttnRange") -> int:
        return self.seqlen + other.seqlen - self.intersect_size(other)

    def diff_by(self, other: "AttnRange") -> list["AttnRange"]:
        """Parts of self not covered by other."""
        res = []
        if other._start > self._start:
            res.append(AttnRange(self._start, min(self._end, other._start)))
        if other._end < self._end:
            res.append(AttnRange(max(self._start, other._end), self._end))
        return [r for r in res if not r.is_empty()]

    def is_subrange_of(self, other: "AttnRange") -> bool:
        return self._start >= other._start and self._end <= other._end

    def is_overlap_with(self, other: "AttnRange") -> bool:
        return self.intersect_size(other) > 0

    def is_empty(self) -> bool:
        return self._end == self._start

    # -- dunder ------------------------------------------------------------
    def __len__(self) -> int:
        return self.seqlen

    def __eq__(self, other: Any) -> bool:
        return (
            isinstance(other, AttnRange)
            and self._start == other._start
            and self._end == other._end
        )

    def __hash__(self) -> int:
        return hash((self._start, self._end))

    def __repr__(self) -> str:  # pragma: no cover
        return f"[{self._start},{self._end})"
