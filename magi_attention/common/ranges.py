# MI355X-native rebuild of the reference API surface
# (reference: magi_attention/common/ranges.py — names kept, fresh implementation).
from __future__ import annotations

from typing import Any, Iterator, List, Sequence, Tuple, Union

import torch

from .range import AttnRange, RangeError

NaiveRanges = List[Tuple[int, int]]


def is_valid_cu_seqlens(cu_seqlens: Sequence[int], seq_len: int) -> bool:
    if len(cu_seqlens) < 2 or cu_seqlens[0] != 0 or cu_seqlens[-1] != seq_len:
        return False
    return all(a <= b for a, b in zip(cu_seqlens, cu_seqlens[1:]))


def check_valid_cu_seqlens(cu_seqlens: Sequence[int], seq_len: int) -> None:
    if not is_valid_cu_seqlens(cu_seqlens, seq_len):
        raise ValueError(f"invalid cu_seqlens {cu_seqlens} for seq_len {seq_len}")


class AttnRanges:
    """An ordered list of AttnRange with set-algebra helpers used by the planner."""

    def __init__(self) -> None:
        self._ranges: list[AttnRange] = []

    # -- basic list ops ----------------------------------------------------
    def append(self, attn_range: AttnRange, check: bool = False) -> None:
        self._ranges.append(attn_range)

    def insert(self, idx: int, attn_range: AttnRange, check: bool = False) -> None:
        self._ranges.insert(idx, attn_range)

    def extend(self, attn_ranges: "AttnRanges", check: bool = False) -> None:
        self._ranges.extend(r.clone() for r in attn_ranges)

    def pop(self, idx: int = -1) -> AttnRange:
        return self._ranges.pop(idx)

    def clear_empty(self) -> "AttnRanges":
        out = AttnRanges()
        out._ranges = [r.clone() for r in self._ranges if not r.is_empty()]
        return out

    def sort(self) -> "AttnRanges":
        out = AttnRanges()
        out._ranges = sorted(
            (r.clone() for r in self._ranges), key=lambda r: (r.start, r.end)
        )
        return out

    def merge(self) -> "AttnRanges":
        """Sort + coalesce overlapping/adjacent ranges."""
        out = AttnRanges()
        for r in self.sort().clear_empty():
            if out._ranges and r.start <= out._ranges[-1].end:
                out._ranges[-1].end = max(out._ranges[-1].end, r.end)
            else:
                out._ranges.append(r.clone())
        return out

    def chunk(self, chunk_size: int, check: bool = True) -> list["AttnRanges"]:
        """Split the (merged) token space of self into consecutive chunks of
        `chunk_size` tokens; each chunk is an AttnRanges of its pieces."""
        chunks: list[AttnRanges] = []
        cur = AttnRanges()
        cur_len = 0
        for r in self.merge():
            s = r.start
            while s < r.end:
                take = min(chunk_size - cur_len, r.end - s)
                cur.append(AttnRange(s, s + take))
                cur_len += take
                s += take
                if cur_len == chunk_size:
                    chunks.append(cur)
                    cur = AttnRanges()
                    cur_len = 0
        if cur_len:
            chunks.append(cur)
        return chunks

    def truncate(self, start: int | None = None, end: int | None = None) -> "AttnRanges":
        out = AttnRanges()
        for r in self._ranges:
            t = r.truncate(start, end)
            if not t.is_empty():
                out.append(t)
        return out

    # -- predicates --------------------------------------------------------
    def is_sorted(self) -> bool:
        return all(
            a.start <= b.start for a, b in zip(self._ranges, self._ranges[1:])
        )

    def is_merged(self) -> bool:
        return self == self.merge()

    def is_non_overlap(self) -> bool:
        s = self.sort()
        return all(a.end <= b.start for a, b in zip(s._ranges, s._ranges[1:]))

    def is_valid(self) -> bool:
        """All member ranges satisfy start <= end (reference ranges.py:158)."""
        return all(r.is_valid_close() for r in self._ranges)

    def check_valid(self) -> None:
        if not self.is_valid():
            raise ValueError(
                f"Some of the {self._ranges=} is invalid against the rule: "
                f"'start <= end'"
            )

    def is_cu_seqlens(self, seqlen: int) -> bool:
        if not self._ranges:
            return False
        if self._ranges[0].start != 0 or self._ranges[-1].end != seqlen:
            return False
        return all(a.end == b.start for a, b in zip(self._ranges, self._ranges[1:]))

    def to_cu_seqlens(self, seq_len: int) -> list[int]:
        assert self.is_cu_seqlens(seq_len), f"{self} is not cu_seqlens-shaped"
        return [0] + [r.end for r in self._ranges]

    # -- local coordinate mapping -----------------------------------------
    def make_range_local(self, attn_range: AttnRange, is_self_merged: bool = False) -> AttnRange:
        """Map a global range (contained in self) to the local coordinate system
        obtained by concatenating self's (merged) ranges in order."""
        base = self if is_self_merged else self.merge()
        offset = 0
        for r in base._ranges:
            if attn_range.is_subrange_of(r):
                return AttnRange(
                    offset + attn_range.start - r.start,
                    offset + attn_range.end - r.start,
                )
            offset += r.seqlen
        raise RangeError(f"{attn_range} is not contained in {base}")

    def make_ranges_local(
        self, attn_ranges: "AttnRanges", is_self_merged: bool = False
    ) -> "AttnRanges":
        base = self if is_self_merged else self.merge()
        out = AttnRanges()
        for r in attn_ranges:
            out.append(base.make_range_local(r, is_self_merged=True))
        return out

    # -- set algebra -------------------------------------------------------
    def find_hole_ranges(
        self, other: "AttnRanges", is_other_merged: bool = False
    ) -> "AttnRanges":
        """Parts of self NOT covered by other (both treated as token sets).
        Reference semantics: dist_attn_solver.py:463-470 remote-K computation."""
        mine = self.merge()
        cover = other if is_other_merged else other.merge()
        out = AttnRanges()
        for r in mine:
            cur = r.start
            for c in cover:
                if c.end <= cur:
                    continue
                if c.start >= r.end:
                    break
                if c.start > cur:
                    out.append(AttnRange(cur, min(c.start, r.end)))
                cur = max(cur, c.end)
                if cur >= r.end:
                    break
            if cur < r.end:
                out.append(AttnRange(cur, r.end))
        return out

    def find_overlap_ranges(self, other: "AttnRanges") -> "AttnRanges":
        mine = self.merge()
        theirs = other.merge()
        out = AttnRanges()
        for r in mine:
            for c in theirs:
                i = r.intersect(c)
                if not i.is_empty():
                    out.append(i)
        return out.merge()

    def intersect_size(self) -> int:
        """Total overlap amount among self's own ranges."""
        total = sum(r.seqlen for r in self._ranges)
        return total - self.merge().total_seqlen

    def union_size(self) -> int:
        """Total token count covered by the union of self's own ranges."""
        return self.merge().total_seqlen

    def intersect_size_with(self, other: "AttnRanges") -> int:
        return sum(r.seqlen for r in self.find_overlap_ranges(other))

    def union_size_with(self, other: "AttnRanges") -> int:
        u = self.merge()
        u.extend(other)
        return u.merge().total_seqlen

    # -- conversions -------------------------------------------------------
    def to_tensor(self, device: str | torch.device = "cpu") -> torch.Tensor:
        return torch.tensor(
            [[r.start, r.end] for r in self._ranges] if self._ranges else [],
            dtype=torch.int32,
            device=device,
        ).reshape(-1, 2)

    def to_naive_ranges(self) -> NaiveRanges:
        return [r.to_naive_range() for r in self._ranges]

    @classmethod
    def from_ranges(cls, ranges, check: bool = False) -> "AttnRanges":
        out = cls()
        for r in ranges:
            if isinstance(r, AttnRange):
                out.append(r.clone())
            else:
                out.append(AttnRange(r[0], r[1]))
        return out

    @classmethod
    def from_cu_seqlens(cls, cu_seqlens: Sequence[int], seq_len: int) -> "AttnRanges":
        check_valid_cu_seqlens(cu_seqlens, seq_len)
        out = cls()
        for a, b in zip(cu_seqlens, cu_seqlens[1:]):
            out.append(AttnRange(a, b))
        return out

    def clone(self) -> "AttnRanges":
        out = AttnRanges()
        out._ranges = [r.clone() for r in self._ranges]
        return out

    # -- stats -------------------------------------------------------------
    @property
    def total_seqlen(self) -> int:
        return sum(r.seqlen for r in self._ranges)

    @property
    def max_seqlen(self) -> int:
        return max((r.seqlen for r in self._ranges), default=0)

    @property
    def start(self) -> int:
        return min((r.start for r in self._ranges), default=0)

    @property
    def end(self) -> int:
        return max((r.end for r in self._ranges), default=0)

    @property
    def size(self) -> int:
        return len(self._ranges)

    @property
    def points(self) -> list[int]:
        pts: set[int] = set()
        for r in self._ranges:
            pts.add(r.start)
            pts.add(r.end)
        return sorted(pts)

    def is_empty(self) -> bool:
        return len(self._ranges) == 0

    # -- dunder ------------------------------------------------------------
    def __len__(self) -> int:
        return len(self._ranges)

    def __getitem__(self, idx: Union[int, slice]):
        if isinstance(idx, slice):
            out = AttnRanges()
            out._ranges = [r.clone() for r in self._ranges[idx]]
            return out
        return self._ranges[idx]

    def __setitem__(self, idx, value) -> None:
        if isinstance(idx, slice):
            self._ranges[idx] = list(value)
        else:
            self._ranges[idx] = value

    def __iter__(self) -> Iterator[AttnRange]:
        return iter(self._ranges)

    def __eq__(self, other: Any) -> bool:
        return isinstance(other, AttnRanges) and self._ranges == other._ranges

    def __hash__(self) -> int:
        return hash(tuple((r.start, r.end) for r in self._ranges))

    def __repr__(self) -> str:  # pragma: no cover
        return f"AttnRanges({self._ranges})"
