"""Protocol-conformance tests (reference surface: common/protocols.py +
tests/test_common/test_protocol_conformance.py): the shipped geometry
types must structurally satisfy the backend-interchangeability contracts."""
import pytest

from magi_attention.common import (
    AttnMaskType,
    AttnRange,
    AttnRanges,
    AttnRectangle,
    AttnRectangles,
)
from magi_attention.common.protocols import (
    AttnMaskTypeProtocol,
    AttnRangeProtocol,
    AttnRangesProtocol,
    AttnRectangleProtocol,
    AttnRectanglesProtocol,
)

CASES = [
    (AttnMaskType.FULL, AttnMaskTypeProtocol),
    (AttnRange(0, 4), AttnRangeProtocol),
    (AttnRanges.from_ranges([(0, 4)]), AttnRangesProtocol),
    (AttnRectangle(AttnRange(0, 4), AttnRange(0, 4), mask_type=1),
     AttnRectangleProtocol),
    (AttnRectangles.from_ranges([(0, 4)], [(0, 4)], [1]),
     AttnRectanglesProtocol),
]


@pytest.mark.parametrize("obj,proto", CASES,
                         ids=[type(o).__name__ for o, _ in CASES])
def test_protocol_conformance(obj, proto):
    assert isinstance(obj, proto), (
        f"{type(obj).__name__} does not satisfy {proto.__name__}"
    )


def test_union_and_ranges_validity():
    a, b = AttnRange(0, 4), AttnRange(2, 8)
    assert a.union(b) == [AttnRange(0, 8)]
    assert a.union(AttnRange(6, 8)) == [a, AttnRange(6, 8)]
    assert a.union(AttnRange(1, 3)) == [a]
    rr = AttnRanges.from_ranges([(0, 4), (8, 12)])
    assert rr.is_valid()
    rr.check_valid()
    rr[0].end = -5  # unchecked setter -> invalid state
    assert not rr.is_valid()
    with pytest.raises(ValueError):
        rr.check_valid()
