"""AttnRectangle: a diagonal-band-bounded attention region
(reference surface: common/rectangle.py:28) — the geometry unit of the
dynamic mask representation: integer points (q, k) with

    q in [q_range.start, q_range.end)   (half-open)
    k in [k_range.start, k_range.end)   (half-open)
    k - q in [d_range.start, d_range.end]   (CLOSED: diagonal indices)

A mask type at construction seeds the band: causal/bi-causal pin the top
diagonal to the bottom-right corner (k_end - q_end), inv/bi-causal pin the
bottom diagonal to the top-left corner (k_start - q_start); the band is
then shrunk to the feasible hull. The area arithmetic is re-derived
(clamped-linear sums over q), not a translation; parity with the reference
is pinned by brute-force counting in tests/test_rectangle_cpu.py.
"""
from __future__ import annotations

from typing import Any, Union

from .enum import AttnMaskType
from .range import AttnRange

__all__ = ["AttnRectangle"]

INT_MAX = 10**9
INT_MIN = -(10**9)


def _tri_sum(m: int, n: int) -> int:
    """sum of q for q in [m, n] (0 when empty)."""
    if m > n:
        return 0
    return (n * (n + 1) - (m - 1) * m) // 2


def _sum_min(m: int, n: int, cap: int, off: int) -> int:
    """sum over q in [m, n] of min(cap, q + off)."""
    if m > n:
        return 0
    # q + off <= cap  <=>  q <= cap - off
    pivot = min(n, cap - off)
    lin = _tri_sum(m, pivot) + off * max(0, pivot - m + 1)  # linear part
    flat = cap * max(0, n - max(m, pivot + 1) + 1)          # capped part
    return lin + flat


def _sum_max(m: int, n: int, floor: int, off: int) -> int:
    """sum over q in [m, n] of max(floor, q + off)."""
    return -_sum_min(-n, -m, -floor, -off)


class AttnRectangle:
    def __init__(
        self,
        q_range: AttnRange,
        k_range: AttnRange,
        d_range: AttnRange | None = None,
        mask_type: AttnMaskType | int = AttnMaskType.FULL,
    ) -> None:
        self._q_range = q_range
        self._k_range = k_range
        self._d_range = AttnRange(INT_MIN, INT_MAX) if d_range is None else d_range

        if isinstance(mask_type, AttnMaskType):
            mt = mask_type
        elif isinstance(mask_type, int):
            mt = AttnMaskType.from_int_type(mask_type)
        else:
            raise TypeError(
                f"mask_type must be AttnMaskType or int type, but got {type(mask_type)}"
            )

        if mt in (AttnMaskType.CAUSAL, AttnMaskType.BICAUSAL):
            # top diagonal pinned to the bottom-right corner
            self._d_range.end = min(self._d_range.end, k_range.end - q_range.end)
        else:
            self._d_range.end = min(self._d_range.end, k_range.end - 1 - q_range.start)
        if mt in (AttnMaskType.INVCAUSAL, AttnMaskType.BICAUSAL):
            # bottom diagonal pinned to the top-left corner
            self._d_range.start = max(self._d_range.start, k_range.start - q_range.start)
        else:
            self._d_range.start = max(
                self._d_range.start, k_range.start - (q_range.end - 1)
            )

        self.shrink_d_range()
        self.shrink_q_range()
        self.shrink_k_range()
        self.check_valid()

    # ---------------- properties ----------------

    @property
    def q_range(self):
        return self._q_range

    @q_range.setter
    def q_range(self, value) -> None:
        self.check_valid(q_range=value)
        self._q_range = value

    @property
    def k_range(self):
        return self._k_range

    @k_range.setter
    def k_range(self, value) -> None:
        self.check_valid(k_range=value)
        self._k_range = value

    @property
    def d_range(self):
        return self._d_range

    @d_range.setter
    def d_range(self, value) -> None:
        self.check_valid(d_range=value)
        self._d_range = value

    # ---------------- validity ----------------

    def is_valid(
        self,
        q_range: AttnRange | None = None,
        k_range: AttnRange | None = None,
        d_range: AttnRange | None = None,
    ) -> bool:
        q_range = self._q_range if q_range is None else q_range
        k_range = self._k_range if k_range is None else k_range
        d_range = self._d_range if d_range is None else d_range
        return (
            q_range.is_valid_open()
            and k_range.is_valid_open()
            and d_range.is_valid_close()
        )

    def check_valid(
        self,
        q_range: AttnRange | None = None,
        k_range: AttnRange | None = None,
        d_range: AttnRange | None = None,
    ) -> None:
        q_range = self._q_range if q_range is None else q_range
        k_range = self._k_range if k_range is None else k_range
        d_range = self._d_range if d_range is None else d_range
        if not self.is_valid(q_range, k_range, d_range):
            raise ValueError(
                f"Some of the {q_range=} {k_range=} {d_range=} is invalid, "
                f"no area include"
            )

    def get_valid_or_none(self) -> Union["AttnRectangle", None]:
        return self if self.is_valid() else None

    # ---------------- shrink to feasible hull ----------------

    def shrink_d_range(self) -> bool:
        self._d_range.start = max(
            self._d_range.start, self._k_range.start - (self._q_range.end - 1)
        )
        self._d_range.end = min(
            self._d_range.end, (self._k_range.end - 1) - self._q_range.start
        )
        return self._d_range.is_valid_close()

    def shrink_q_range(self) -> bool:
        self._q_range.start = max(
            self._q_range.start, self._k_range.start - self._d_range.end
        )
        self._q_range.end = min(
            self._q_range.end, self._k_range.end - self._d_range.start
        )
        return self._q_range.is_valid_open()

    def shrink_k_range(self) -> bool:
        self._k_range.start = max(
            self._k_range.start, self._d_range.start + self._q_range.start
        )
        self._k_range.end = min(
            self._k_range.end, self._d_range.end + self._q_range.end
        )
        return self._k_range.is_valid_open()

    def clone(self) -> "AttnRectangle":
        new = self.__class__.__new__(self.__class__)
        new._q_range = self._q_range.clone()
        new._k_range = self._k_range.clone()
        new._d_range = self._d_range.clone()
        return new

    # ---------------- cuts ----------------

    def cut_q(
        self, cut_pos: int
    ) -> tuple[Union["AttnRectangle", None], Union["AttnRectangle", None]]:
        if cut_pos <= self._q_range.start:
            return None, self
        if cut_pos >= self._q_range.end:
            return self, None
        left, right = self.clone(), self.clone()
        left._q_range.end = cut_pos
        right._q_range.start = cut_pos
        for part in (left, right):
            part.shrink_d_range()
            part.shrink_k_range()
        return left, right

    def cut_k(
        self, cut_pos: int
    ) -> tuple[Union["AttnRectangle", None], Union["AttnRectangle", None]]:
        if cut_pos <= self._k_range.start:
            return None, self
        if cut_pos >= self._k_range.end:
            return self, None
        left, right = self.clone(), self.clone()
        left._k_range.end = cut_pos
        right._k_range.start = cut_pos
        for part in (left, right):
            part.shrink_d_range()
            part.shrink_q_range()
        return left, right

    def get_rect_within_q_segment(
        self, q_start: int, q_end: int
    ) -> Union["AttnRectangle", None]:
        if q_end <= self._q_range.start or q_start >= self._q_range.end:
            return None
        seg = self.clone()
        seg._q_range.start = max(seg._q_range.start, q_start)
        seg._q_range.end = min(seg._q_range.end, q_end)
        seg.shrink_d_range()
        seg.shrink_k_range()
        return seg

    def get_rect_within_k_segment(
        self, k_start: int, k_end: int
    ) -> Union["AttnRectangle", None]:
        if k_end <= self._k_range.start or k_start >= self._k_range.end:
            return None
        seg = self.clone()
        seg._k_range.start = max(seg._k_range.start, k_start)
        seg._k_range.end = min(seg._k_range.end, k_end)
        seg.shrink_d_range()
        seg.shrink_q_range()
        return seg

    # ---------------- classification ----------------

    def intersection_q_id_on_left_boundary(self) -> int:
        """q where the bottom diagonal meets k_range.start."""
        return self._k_range.start - self._d_range.start

    def intersection_q_id_on_right_boundary(self) -> int:
        """q where the top diagonal meets k_range.end - 1."""
        return self._k_range.end - 1 - self._d_range.end

    def _band_unbounded_below(self) -> bool:
        return self._d_range.start <= self._k_range.start - (self._q_range.end - 1)

    def _band_unbounded_above(self) -> bool:
        return self._d_range.end >= (self._k_range.end - 1) - self._q_range.start

    def is_full(self) -> bool:
        return self._band_unbounded_below() and self._band_unbounded_above()

    def is_causal(self) -> bool:
        return (
            self._band_unbounded_below()
            and self._d_range.end == self._k_range.end - self._q_range.end
        )

    def is_inv_causal(self) -> bool:
        return (
            self._d_range.start == self._k_range.start - self._q_range.start
            and self._band_unbounded_above()
        )

    def is_bi_causal(self) -> bool:
        return (
            self._d_range.start == self._k_range.start - self._q_range.start
            and self._d_range.end == self._k_range.end - self._q_range.end
        )

    def to_qk_range_mask_type(self) -> list[tuple[AttnRange, AttnRange, int]]:
        """Decompose into (q_range, k_range, ffa attn_type) triples — the
        band becomes a stack of full / causal / inv-causal / bi-causal
        sub-rectangles, cut at the boundary-intersection q rows."""
        if self.is_full():
            return [(self._q_range, self._k_range, 0)]
        if self.is_causal():
            return [(self._q_range, self._k_range, 1)]
        if self.is_inv_causal():
            return [(self._q_range, self._k_range, 2)]
        if self.is_bi_causal():
            return [(self._q_range, self._k_range, 3)]

        q_l = self.intersection_q_id_on_left_boundary()
        q_r = self.intersection_q_id_on_right_boundary()
        if not (
            self._q_range.start <= q_l < self._q_range.end
            and self._q_range.start <= q_r < self._q_range.end
        ):
            raise ValueError(
                f"rect{self} without shrinkage call to_qk_range_mask_type"
            )

        # pick the cut(s) that split off the pure top/bottom parts
        if q_l == self._q_range.end - 1:
            cuts = [q_r + 1]        # causal top | full bottom
        elif q_r == self._q_range.start:
            cuts = [q_l]            # full top | inv-causal bottom
        elif q_r <= q_l:
            cuts = [q_l]            # causal/full top | inv-causal bottom
        elif q_r == q_l + 1:
            cuts = [q_r]            # causal top | inv-causal bottom
        else:
            cuts = [q_l, q_r + 1]   # causal | bi-causal | inv-causal
        out: list[tuple[AttnRange, AttnRange, int]] = []
        rest: Union["AttnRectangle", None] = self
        for pos in cuts:
            assert rest is not None
            part, rest = rest.cut_q(pos)
            if part is not None:
                out.extend(part.to_qk_range_mask_type())
            if rest is None:
                return out
        out.extend(rest.to_qk_range_mask_type())
        return out

    # ---------------- area ----------------

    def area(self) -> int:
        return self.count_areas(
            self._q_range.start, self._q_range.end,
            self._k_range.start, self._k_range.end,
            self._d_range.start, self._d_range.end,
        )

    @staticmethod
    def count_areas(lq: int, rq: int, lk: int, rk: int, ld: int, rd: int) -> int:
        """#{(q, k) : lq <= q < rq, lk <= k < rk, ld <= k - q <= rd} in O(1):
        for each q the k-row is [max(lk, q+ld), min(rk-1, q+rd)], non-empty
        exactly for q in [lk - rd, (rk-1) - ld]; sum the clamped-linear
        bounds with triangular-number algebra."""
        if rq <= lq or rk <= lk or rd < ld:
            return 0
        k_hi, k_lo = rk - 1, lk
        m = max(lq, k_lo - rd)
        n = min(rq - 1, k_hi - ld)
        if m > n:
            return 0
        upper = _sum_min(m, n, k_hi, rd)
        lower = _sum_max(m, n, k_lo, ld)
        return upper - lower + (n - m + 1)

    # ---------------- dunder ----------------

    def __len__(self) -> int:
        return 1

    def __eq__(self, other: Any) -> bool:
        if isinstance(other, AttnRectangle):
            return (
                self._q_range == other._q_range
                and self._k_range == other._k_range
                and self._d_range == other._d_range
            )
        return False

    def __hash__(self) -> int:
        return hash((self._q_range, self._k_range, self._d_range))

    def __repr__(self) -> str:  # pragma: no cover
        return f"{self._q_range} x {self._k_range} x {self._d_range}"
