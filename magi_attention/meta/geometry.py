# Mask-slice geometry engine for the CP planner.
#
# Role of the reference's HostAttnSliceMaker/RemoteAttnSliceMaker
# (meta/solver/slice_maker.py:25-499): splitting causal/inv-causal/bi-causal
# slices that cross chunk or host/remote boundaries into sub-slices with
# recomputed mask types. Re-designed here around ONE primitive instead of
# per-shape case analysis:
#
#   every typed slice is a diagonal-band constraint  lo <= k - q <= up
#   (lo = ks - qs for INV/BICAUSAL else -inf; up = ke - qe for CAUSAL/BICAUSAL
#    else +inf; semantics flex_flash_attn.py:1247-1341), and `normalize()`
#   re-expresses any (q_window, k_window, lo, up) region as a minimal list of
#   exactly-aligned typed slices by cutting the q axis at the two thresholds
#   where each bound starts/stops binding and shrinking k to the envelope.
#
# Validated against dense masks in tests/test_geometry.py.
from __future__ import annotations

from dataclasses import dataclass
from typing import List, Optional, Tuple

FULL, CAUSAL, INV_CAUSAL, BI_CAUSAL = 0, 1, 2, 3


@dataclass(frozen=True)
class MaskSlice:
    """One (q_range, k_range, type) triple in GLOBAL coordinates."""

    qs: int
    qe: int
    ks: int
    ke: int
    t: int

    @property
    def sq(self) -> int:
        return self.qe - self.qs

    @property
    def sk(self) -> int:
        return self.ke - self.ks

    def bounds(self) -> Tuple[Optional[int], Optional[int]]:
        """(lo, up) of the band lo <= k - q <= up, None = unbounded."""
        lo = self.ks - self.qs if self.t in (INV_CAUSAL, BI_CAUSAL) else None
        up = self.ke - self.qe if self.t in (CAUSAL, BI_CAUSAL) else None
        return lo, up

    def area(self) -> int:
        sq, sk = self.sq, self.sk
        if sq <= 0 or sk <= 0:
            return 0
        if self.t == FULL:
            return sq * sk
        if self.t == CAUSAL:
            # row i (local) sees sk - sq + 1 + i keys (>=1 after normalize;
            # raw slices may have empty rows -> clamp)
            base = sk - sq + 1
            lo_i = max(0, -base + 1)  # first row with >=1 key
            n = sq - lo_i
            if n <= 0:
                return 0
            first = base + lo_i
            return n * first + n * (n - 1) // 2
        if self.t == INV_CAUSAL:
            # row i sees max(0, sk - i) keys
            n = min(sq, sk)
            return n * sk - n * (n - 1) // 2
        # BI_CAUSAL: row i sees clamp of [i, i + sk - sq] within [0, sk)
        total = 0
        w = sk - sq + 1
        if w >= 1:
            return sq * w
        # degenerate band (sk < sq): allowed j in [i, i + sk - sq] is empty
        return 0


def normalize(
    qlo: int, qhi: int, ka: int, kb: int,
    lo: Optional[int], up: Optional[int],
) -> List[MaskSlice]:
    """Express region {(q,k): q in [qlo,qhi), k in [ka,kb), lo<=k-q<=up} as
    exactly-aligned typed slices."""
    if qhi <= qlo or kb <= ka:
        return []
    if lo is not None and up is not None and lo > up:
        return []  # empty band (e.g. bi-causal with sk < sq shrunk to nothing)
    # clip empty rows
    if up is not None:
        qlo = max(qlo, ka - up)  # q+up >= ka
    if lo is not None:
        qhi = min(qhi, kb - lo)  # q+lo <= kb-1
    if qhi <= qlo:
        return []
    # binding thresholds on q
    t_lo = (ka - lo) if lo is not None else None  # rows >= t_lo: lower binds
    t_up = (kb - up) if up is not None else None  # rows <  t_up: upper binds
    cuts = {qlo, qhi}
    if t_lo is not None and qlo < t_lo < qhi:
        cuts.add(t_lo)
    if t_up is not None and qlo < t_up < qhi:
        cuts.add(t_up)
    pts = sorted(cuts)
    out: List[MaskSlice] = []
    for x, y in zip(pts, pts[1:]):
        inv_act = lo is not None and (t_lo is not None and x >= t_lo)
        cau_act = up is not None and (t_up is not None and y <= t_up)
        kx = x + lo if inv_act else ka
        ky = y + up if cau_act else kb
        kx = max(kx, ka)
        ky = min(ky, kb)
        if ky <= kx or y <= x:
            continue
        t = (
            BI_CAUSAL if (inv_act and cau_act)
            else CAUSAL if cau_act
            else INV_CAUSAL if inv_act
            else FULL
        )
        out.append(MaskSlice(x, y, kx, ky, t))
    return out


def transpose_slice(sl: MaskSlice) -> List[MaskSlice]:
    """The same mask region viewed with axes swapped: (q', k') = (k, q),
    band lo <= k-q <= up  ->  -up <= k'-q' <= -lo. Used by QO-comm planning
    (the solver runs on the transposed mask so remote-K machinery produces
    remote-Q tables; reference env/comm.py:72 MAGI_ATTENTION_QO_COMM)."""
    lo, up = sl.bounds()
    lo2 = -up if up is not None else None
    up2 = -lo if lo is not None else None
    return normalize(sl.ks, sl.ke, sl.qs, sl.qe, lo2, up2)


def q_window(sl: MaskSlice, a: int, b: int) -> List[MaskSlice]:
    """Sub-slices of sl restricted to q rows [a, b)."""
    lo, up = sl.bounds()
    return normalize(max(sl.qs, a), min(sl.qe, b), sl.ks, sl.ke, lo, up)


def k_window(sl: MaskSlice, a: int, b: int) -> List[MaskSlice]:
    """Sub-slices of sl restricted to k cols [a, b)."""
    lo, up = sl.bounds()
    return normalize(sl.qs, sl.qe, max(sl.ks, a), min(sl.ke, b), lo, up)


def slice_from_raw(qs: int, qe: int, ks: int, ke: int, t: int) -> List[MaskSlice]:
    """Normalize a raw user slice (which may contain empty rows) into aligned
    slices — identity for well-formed inputs."""
    lo = ks - qs if t in (INV_CAUSAL, BI_CAUSAL) else None
    up = ke - qe if t in (CAUSAL, BI_CAUSAL) else None
    return normalize(qs, qe, ks, ke, lo, up)


def area_in_rows(sl: MaskSlice, a: int, b: int) -> int:
    """Mask area of sl within q rows [a, b)."""
    return sum(s.area() for s in q_window(sl, a, b))


def to_dense(slices, total_q: int, total_k: int):
    """Dense bool mask (tests only)."""
    import torch

    m = torch.zeros(total_q, total_k, dtype=torch.bool)
    for sl in slices:
        lo, up = sl.bounds()
        for q in range(sl.qs, sl.qe):
            ka = sl.ks if lo is None else max(sl.ks, q + lo)
            kb = sl.ke if up is None else min(sl.ke, q + up + 1)
            if kb > ka:
                m[q, ka:kb] = True
    return m
