"""AttnMask: a dense, inspectable materialization of a varlen flex mask
(reference surface: magi_attention/common/mask.py:29 `AttnMask` — factory
methods `from_ranges`/`from_mask`, area/sub-mask queries, varlen/pure
classification). Behaviorally compatible; the row→ranges inference in
`from_mask` is re-derived (vectorized per-row span extraction + the same
FULL/CAUSAL row-merging rules), not a translation.

Like the reference, `from_ranges` accepts FULL and CAUSAL blocks only (the
class predates inv/bi-causal) and later blocks OVERWRITE earlier cells
within their rectangle (assignment semantics, not OR).
"""
from __future__ import annotations

from contextlib import contextmanager
from typing import Any, Iterable

import torch
from torch import nn

from .enum import AttnMaskType
from .range import AttnRange
from .ranges import AttnRanges


class AttnMask(nn.Module):
    """A 2-D mask matrix with per-cell meta info plus the (q_ranges,
    k_ranges, attn_mask_type) tuple list it was built from / inferred to."""

    _can_instantiate = False
    meta_info_dim_size = 1
    mask_flag_dim_idx = 0
    masked_flag = 0
    unmasked_flag = 1
    device = "cpu"

    def __new__(cls, *args, **kwargs):
        if not cls._can_instantiate:
            raise RuntimeError("Please use the factory methods to create an instance.")
        return super().__new__(cls)

    def __init__(
        self,
        mask_tensor: torch.Tensor,
        q_ranges: AttnRanges,
        k_ranges: AttnRanges,
        attn_mask_type: list[AttnMaskType],
        total_seqlen_q: int,
        total_seqlen_k: int,
    ) -> None:
        super().__init__()
        self.mask_tensor = mask_tensor
        self.q_ranges = q_ranges
        self.k_ranges = k_ranges
        self.attn_mask_type = attn_mask_type
        self.total_seqlen_q = total_seqlen_q
        self.total_seqlen_k = total_seqlen_k
        # normalized 0/1 int32 view of the mask-flag plane
        flags = mask_tensor[..., self.__class__.mask_flag_dim_idx]
        self.mask_flag_array = (
            (flags != self.__class__.masked_flag).to(torch.int32).numpy()
        )
        self._is_pure_full: bool | None = None
        self._is_pure_causal: bool | None = None
        self._is_empty: bool | None = None

    def tuples(self) -> Iterable[tuple[AttnRange, AttnRange, AttnMaskType]]:
        yield from zip(self.q_ranges, self.k_ranges, self.attn_mask_type)

    # ---------------- factories ----------------

    @classmethod
    def from_ranges(
        cls,
        q_ranges: AttnRanges,
        k_ranges: AttnRanges,
        attn_mask_type: list[AttnMaskType],
        total_seqlen_q: int | None = None,
        total_seqlen_k: int | None = None,
    ) -> "AttnMask":
        assert len(q_ranges) == len(k_ranges) == len(attn_mask_type), (
            f"The length should be equal, but got: {len(q_ranges)=}, "
            f"{len(k_ranges)=}, {len(attn_mask_type)=}"
        )
        tq = total_seqlen_q if total_seqlen_q is not None else q_ranges.end
        tk = total_seqlen_k if total_seqlen_k is not None else k_ranges.end

        with cls.can_instantiate_ctx():
            m = torch.full(
                (tq, tk, cls.meta_info_dim_size), cls.masked_flag,
                dtype=torch.int32,
            )
            for qr, kr, t in zip(q_ranges, k_ranges, attn_mask_type):
                if t == AttnMaskType.FULL:
                    block = torch.full(
                        (qr.seqlen, kr.seqlen), cls.unmasked_flag,
                        dtype=torch.int32,
                    )
                elif t == AttnMaskType.CAUSAL:
                    block = cls.make_causal_mask(qr.seqlen, kr.seqlen)
                    block = torch.where(
                        block != 0,
                        torch.tensor(cls.unmasked_flag, dtype=torch.int32),
                        torch.tensor(cls.masked_flag, dtype=torch.int32),
                    )
                else:
                    raise ValueError(f"Invalid mask type: {t}")
                # assignment (overwrite) semantics, matching the reference
                m[qr.start : qr.end, kr.start : kr.end, cls.mask_flag_dim_idx] = block
            return AttnMask(
                mask_tensor=m,
                q_ranges=q_ranges,
                k_ranges=k_ranges,
                attn_mask_type=attn_mask_type,
                total_seqlen_q=tq,
                total_seqlen_k=tk,
            )

    @classmethod
    def from_mask(
        cls,
        mask: list[list[int]] | torch.Tensor,
    ) -> "AttnMask":
        """Infer a canonical (q_ranges, k_ranges, attn_mask_type) tuple list
        from a dense 0/1 matrix whose unmasked cells are row-contiguous."""
        mask = torch.as_tensor(mask, dtype=torch.int32, device=cls.device)
        cls._check_mask_valid(mask)
        tq, tk = mask.shape

        # per-row [start, end) span of the unmasked run (empty -> (row, row),
        # the reference's empty-row convention)
        nz = mask != 0
        any_row = nz.any(dim=1)
        first = torch.where(any_row, nz.int().argmax(dim=1), 0)
        last = torch.where(
            any_row, tk - 1 - nz.flip(dims=[1]).int().argmax(dim=1), -1
        )
        spans: list[AttnRange] = [
            AttnRange(int(first[r]), int(last[r]) + 1)
            if bool(any_row[r])
            else AttnRange(r, r)
            for r in range(tq)
        ]

        with cls.can_instantiate_ctx():
            q_ranges, k_ranges = AttnRanges(), AttnRanges()
            types: list[AttnMaskType] = []

            def push(row: int, span: AttnRange) -> None:
                q_ranges.append(AttnRange(row, row + 1))
                k_ranges.append(span)
                # an empty row can only be the top of a causal block
                types.append(
                    AttnMaskType.CAUSAL if span.is_empty() else AttnMaskType.FULL
                )

            def steps_down(top: AttnRange, bot: AttnRange) -> bool:
                # two stacked rows forming a causal staircase
                return top.start == bot.start and top.end + 1 == bot.end

            for row, span in enumerate(spans):
                if row == 0:
                    push(row, span)
                    continue
                lq, lk, lt = q_ranges[-1], k_ranges[-1], types[-1]
                if lt == AttnMaskType.FULL:
                    if span == lk:
                        q_ranges[-1] = AttnRange(lq.start, lq.end + 1)
                    elif steps_down(lk, span):
                        if lq.seqlen == 1:
                            # lone row becomes the top of a causal block
                            q_ranges[-1] = AttnRange(lq.start, lq.end + 1)
                            k_ranges[-1] = AttnRange(lk.start, lk.end + 1)
                            types[-1] = AttnMaskType.CAUSAL
                        else:
                            # split the full block's last row off into a new
                            # causal block together with this row
                            q_ranges[-1] = AttnRange(lq.start, lq.end - 1)
                            q_ranges.append(AttnRange(row - 1, row + 1))
                            k_ranges.append(span)
                            types.append(AttnMaskType.CAUSAL)
                    else:
                        push(row, span)
                else:  # CAUSAL
                    if lk.is_empty():
                        if span.is_empty():
                            q_ranges[-1] = AttnRange(lq.start, lq.end + 1)
                        elif span.seqlen == 1:
                            q_ranges[-1] = AttnRange(lq.start, lq.end + 1)
                            k_ranges[-1] = span
                        else:
                            push(row, span)
                    elif steps_down(lk, span):
                        q_ranges[-1] = AttnRange(lq.start, lq.end + 1)
                        k_ranges[-1] = AttnRange(lk.start, lk.end + 1)
                    else:
                        push(row, span)

            return AttnMask(
                mask_tensor=mask.unsqueeze(-1),
                q_ranges=q_ranges,
                k_ranges=k_ranges,
                attn_mask_type=types,
                total_seqlen_q=tq,
                total_seqlen_k=tk,
            )

    # ---------------- checks / queries ----------------

    @classmethod
    def _check_mask_valid(cls, mask: torch.Tensor) -> None:
        assert len(mask.shape) == 2, f"The mask should be 2d, but got: {mask.shape=}"
        nz = mask != 0
        counts = nz.sum(dim=1)
        tk = mask.shape[1]
        first = nz.int().argmax(dim=1)
        last = tk - 1 - nz.flip(dims=[1]).int().argmax(dim=1)
        ok = (counts == 0) | (counts == last - first + 1)
        assert bool(ok.all()), (
            f"The unmasked col idxs of rows {(~ok).nonzero().flatten().tolist()} "
            f"are not contiguous"
        )

    def _check_sub_range_valid(self, q_range: AttnRange, k_range: AttnRange) -> None:
        assert q_range.end <= self.total_seqlen_q, (
            f"The {q_range.end=} should be no greater than {self.total_seqlen_q=}"
        )
        assert k_range.end <= self.total_seqlen_k, (
            f"The {k_range.end=} should be no greater than {self.total_seqlen_k=}"
        )

    def calc_sub_area(self, q_range: AttnRange, k_range: AttnRange) -> int:
        self._check_sub_range_valid(q_range, k_range)
        return int(
            self.mask_flag_array[
                q_range.start : q_range.end, k_range.start : k_range.end
            ].sum()
        )

    def make_sub_mask(self, q_range: AttnRange, k_range: AttnRange) -> "AttnMask":
        self._check_sub_range_valid(q_range, k_range)
        sub = self.mask_tensor[
            q_range.start : q_range.end,
            k_range.start : k_range.end,
            self.__class__.mask_flag_dim_idx,
        ]
        return AttnMask.from_mask(sub)

    @property
    def area(self) -> int:
        return int(self.mask_flag_array.sum())

    def is_square(self) -> bool:
        return self.total_seqlen_q == self.total_seqlen_k

    def is_pure_full(self) -> bool:
        if self._is_pure_full is None:
            self._is_pure_full = (
                self.area == self.total_seqlen_q * self.total_seqlen_k
            )
        return self._is_pure_full

    def is_pure_causal(self) -> bool:
        if self._is_pure_causal is None:
            ref = self.make_causal_mask(
                self.total_seqlen_q, self.total_seqlen_k
            ).numpy()
            self._is_pure_causal = bool((self.mask_flag_array == ref).all())
        return self._is_pure_causal

    def is_varlen_full(self) -> bool:
        return (
            all(t == AttnMaskType.FULL for t in self.attn_mask_type)
            and self.q_ranges.is_cu_seqlens(self.total_seqlen_q)
            and self.k_ranges.is_cu_seqlens(self.total_seqlen_k)
        )

    def is_varlen_causal(self) -> bool:
        return (
            all(t == AttnMaskType.CAUSAL for t in self.attn_mask_type)
            and self.q_ranges.is_cu_seqlens(self.total_seqlen_q)
            and self.k_ranges.is_cu_seqlens(self.total_seqlen_k)
        )

    def is_empty(self) -> bool:
        if self._is_empty is None:
            self._is_empty = self.area == 0
        return self._is_empty

    @staticmethod
    def make_causal_mask(
        seqlen_q: int,
        seqlen_k: int,
        align: str = "bottom-right",
        dtype=torch.int32,
        device: str = "cpu",
    ) -> torch.Tensor:
        n = max(seqlen_q, seqlen_k)
        tri = torch.tril(torch.ones((n, n))).to(dtype=dtype, device=device)
        if align == "bottom-right":
            return tri[n - seqlen_q :, n - seqlen_k :]
        if align == "top-left":
            return tri[:seqlen_q, :seqlen_k]
        raise ValueError(f"Invalid alignment mode: {align}")

    def visualize(self, save_path: str | None = None) -> None:
        """Render the mask; needs matplotlib (optional in this image)."""
        try:
            import matplotlib.pyplot as plt  # noqa: F401
        except ImportError as e:  # pragma: no cover
            raise RuntimeError(
                "AttnMask.visualize requires matplotlib, which is not "
                "installed in this environment"
            ) from e
        fig, ax = plt.subplots()
        ax.imshow(self.mask_flag_array, cmap="gray_r", interpolation="nearest")
        ax.set_title("Attention Mask")
        ax.set_xlabel("Key/Value")
        ax.set_ylabel("Query")
        if save_path is not None:
            fig.savefig(save_path)
        else:  # pragma: no cover
            plt.show()
        plt.close(fig)

    def __repr__(self) -> str:  # pragma: no cover
        lines = [
            "",
            f"{self.total_seqlen_q=} | {self.total_seqlen_k=} | {self.area=}",
            f"{self.q_ranges=}",
            f"{self.k_ranges=}",
            f"{self.attn_mask_type=}",
            "attn_mask=",
            "\n".join(
                " ".join(str(int(v)) for v in row) for row in self.mask_flag_array
            ),
        ]
        return "\n".join(lines)

    def __eq__(self, other: Any) -> bool:
        if isinstance(other, AttnMask):
            return (
                torch.equal(self.mask_tensor, other.mask_tensor)
                and self.q_ranges == other.q_ranges
                and self.k_ranges == other.k_ranges
                and self.attn_mask_type == other.attn_mask_type
            )
        return False

    @classmethod
    @contextmanager
    def can_instantiate_ctx(cls):
        cls._can_instantiate = True
        yield
        cls._can_instantiate = False
