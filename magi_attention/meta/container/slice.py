"""AttnSlice / MultiKAttnSlice (reference meta/container/slice.py): one
(q_range, k_range, mask_type) cell of the chunked mask IR, with closed-form
area and a k-overlap IOU used by affinity dispatch."""
from __future__ import annotations

from dataclasses import dataclass, field

from ...common.enum import AttnMaskType
from ...common.range import AttnRange
from ...common.ranges import AttnRanges


def _typed_area(sq: int, sk: int, mask_type: AttnMaskType) -> int:
    """Points of a sq x sk slice under the FFA mask type (bottom-right
    causal / top-left inv-causal / both for bi-causal)."""
    if mask_type == AttnMaskType.FULL:
        return sq * sk
    if mask_type in (AttnMaskType.CAUSAL, AttnMaskType.INVCAUSAL):
        if sk > sq:  # trapezoid
            return (2 * sk - sq + 1) * sq // 2
        return (1 + sk) * sk // 2  # triangle
    if mask_type == AttnMaskType.BICAUSAL:
        return (sk - sq + 1) * sq  # parallelogram band
    raise ValueError(
        f"Only support 'full', 'causal', 'inv_causal' and 'bi_causal' mask, "
        f"but got {mask_type}."
    )


@dataclass(repr=False)
class AttnSlice:
    slice_id: int | None = None
    mask_type: AttnMaskType | None = None
    q_range: AttnRange | None = None
    k_range: AttnRange | None = None
    _area: int | None = None

    @property
    def area(self) -> int:
        if self._area is None:
            self._area = _typed_area(
                self.q_range.seqlen, self.k_range.seqlen, self.mask_type
            )
        return self._area

    @area.setter
    def area(self, area: int) -> None:
        self._area = area

    def iou_with(self, other: "AttnSlice") -> float:
        return self.k_range.intersect_size(other.k_range) / self.k_range.union_size(
            other.k_range
        )

    def __eq__(self, other: object) -> bool:
        if not isinstance(other, AttnSlice):
            return False
        return (
            self.mask_type == other.mask_type
            and self.q_range == other.q_range
            and self.k_range == other.k_range
        )

    def __repr__(self) -> str:  # pragma: no cover
        return (
            f"AttnSlice(slice_id={self.slice_id}, "
            f"q_range={self.q_range}, k_range={self.k_range}, "
            f"mask_type={self.mask_type}, area={self.area})"
        )


@dataclass(repr=False)
class MultiKAttnSlice:
    """One q_range attending several k_ranges, each with its own type."""

    q_range: AttnRange = None  # type: ignore[assignment]
    k_ranges: AttnRanges = field(default_factory=AttnRanges)
    mask_types: list[AttnMaskType] = field(default_factory=list)
    slice_id: int | None = None
    _area: int | None = None

    def __post_init__(self):
        assert len(self.mask_types) == len(self.k_ranges), (
            f"The length of mask_types and k_ranges should be the same, "
            f"but got {len(self.mask_types)} and {len(self.k_ranges)}"
        )

    @property
    def area(self) -> int:
        if self._area is None:
            self._area = sum(
                _typed_area(self.q_range.seqlen, kr.seqlen, mt)
                for kr, mt in zip(self.k_ranges, self.mask_types)
            )
        return self._area

    @area.setter
    def area(self, area: int) -> None:
        self._area = area

    def __repr__(self) -> str:  # pragma: no cover
        return (
            f"MultiKAttnSlice(slice_id={self.slice_id}, "
            f"q_range={self.q_range}, k_ranges={self.k_ranges}, "
            f"mask_types={self.mask_types}, area={self.area})"
        )
