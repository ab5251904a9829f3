# Planner output containers (reference meta/collection/{calc_meta.py:62 AttnArg,
# :719 CalcMeta; comm_meta.py:572 CommMeta, :41 GroupCollectiveArg} — same roles,
# rebuilt for the RCCL a2av transport).
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Dict, List, Optional, Tuple

import torch


@dataclass
class AttnArg:
    """Kernel arguments of one FFA call: (q_range, k_range, type) triples in
    LOCAL coordinates (q: rank-local rows; k: host-local or stage-buffer rows).
    Device tensors are materialised once per device and cached
    (reference calc_meta.py:83-117 __post_init__)."""

    q_ranges: List[Tuple[int, int]]
    k_ranges: List[Tuple[int, int]]
    attn_type_map: List[int]
    max_seqlen_q: int = 0
    total_area: int = 0
    _cache: Dict[str, torch.Tensor] = field(default_factory=dict, repr=False)

    def is_empty(self) -> bool:
        return len(self.q_ranges) == 0

    def to_device(self, device) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
        key = str(device)
        if key not in self._cache:
            self._cache[key] = (
                torch.tensor(self.q_ranges, dtype=torch.int32, device=device).reshape(-1, 2),
                torch.tensor(self.k_ranges, dtype=torch.int32, device=device).reshape(-1, 2),
                torch.tensor(self.attn_type_map, dtype=torch.int32, device=device),
            )
        return self._cache[key]

    def to_device_merged(self, device):
        """Cached auto_range_merge tables for this arg: one (outer, inner,
        types, seg_starts) tuple per pass direction (q-merged for fwd/dq,
        k-merged for dkv) — built once per device, reused every step
        (MAGI_ATTENTION_AUTO_RANGE_MERGE, reference env/general.py:206)."""
        key = ("merged", str(device))
        if key not in self._cache:
            from ..functional.flex_flash_attn import _seg_starts, merge_ranges

            qr, kr, tm = self.to_device(device)
            n = qr.shape[0]
            mq, _, sk, st, inv_q, _ = merge_ranges(qr, kr, tm)
            mk, _, sq, st2, inv_k, _ = merge_ranges(kr, qr, tm)
            self._cache[key] = (
                (mq, sk, st, _seg_starts(inv_q, n)),
                (mk, sq, st2, _seg_starts(inv_k, n)),
            )
        return self._cache[key]


@dataclass
class RowChunkMap:
    """Row copy plan: (in_ranges over a source buffer, out_starts in a dest
    buffer) — the unpack/pack tables driving range_gather/range_reduce."""

    in_ranges: List[Tuple[int, int]]
    out_starts: List[int]
    total_rows: int
    _cache: Dict[str, Tuple[torch.Tensor, torch.Tensor]] = field(
        default_factory=dict, repr=False
    )

    def to_device(self, device) -> Tuple[torch.Tensor, torch.Tensor]:
        key = str(device)
        if key not in self._cache:
            self._cache[key] = (
                torch.tensor(self.in_ranges, dtype=torch.int32, device=device).reshape(-1, 2),
                torch.tensor(self.out_starts, dtype=torch.int32, device=device),
            )
        return self._cache[key]


@dataclass
class GroupCastArg:
    """One overlap stage's K/V multicast for ONE rank (a2av realisation;
    reference comm_meta.py:41 GroupCollectiveArg + grpcoll/utils.py:593
    calc_group_cast_a2a_args)."""

    # send: my host-local k rows packed per dst rank (rows of the K tensor;
    # V reuses the same tables at +total_local offset)
    send_pack: RowChunkMap            # local kv rows -> send buffer
    input_split_sizes: List[int]      # per dst rank (k rows)
    # recv: rows arrive ordered by src rank; unpack into the globally-sorted
    # stage buffer
    recv_unpack: RowChunkMap          # recv buffer rows -> stage buffer
    output_split_sizes: List[int]     # per src rank (k rows)
    stage_tokens: int                 # stage buffer rows (k)
    # tensor copies packed through the same row tables: 2 for (K,V) or
    # (q,do) / (lse,dpsum); 1 for single-tensor casts (QO-comm q)
    ncopies: int = 2


@dataclass
class GroupReduceArg:
    """Reverse path: partial dK/dV rows of the stage buffer sent back to owner
    ranks and sum-reduced into owner-local accumulators."""

    send_pack: RowChunkMap            # stage-buffer rows -> send buffer (per dst owner)
    input_split_sizes: List[int]
    recv_reduce: RowChunkMap          # recv rows -> local kv row positions (sum)
    output_split_sizes: List[int]
    total_recv: int


@dataclass
class HierGroupCastArg:
    """Hierarchical (2D-mesh) K/V multicast for one stage: pre-intra a2av for
    same-node destinations, inter a2av sending ONE deduplicated copy per remote
    node to the same-local-rank proxy, post-intra a2av forwarding proxy rows to
    final destinations (reference grpcoll/_group_collective_hier.py:49
    HierGroupCastMetaSolver — but built fully at PLAN time from the solver's
    global overlap table, no runtime all_gather_object)."""

    pre: GroupCastArg                 # intra hop (recv lands in stage buffer)
    inter_send_pack: RowChunkMap      # local kv rows -> inter send buffer
    inter_in_splits: List[int]        # per peer node (2*tok rows)
    inter_out_splits: List[int]
    inter_total_recv: int
    post_send_pack: RowChunkMap       # inter-recv rows -> post send buffer
    post_in_splits: List[int]
    post_recv_unpack: RowChunkMap     # post recv rows -> stage buffer
    post_out_splits: List[int]
    stage_tokens: int


@dataclass
class HierGroupReduceArg:
    """Mirror path for partial dK/dV: pre-intra a2av routes partials to the
    in-node proxy holding the owner's local rank, which SUM-reduces its node's
    contributions (the traffic saving), then one inter a2av delivers each
    node-sum to the owner."""

    pre_send_pack: RowChunkMap        # stage partial rows -> pre send buffer
    pre_in_splits: List[int]
    pre_recv_direct: RowChunkMap      # pre recv -> local dkv rows (sum)
    pre_recv_proxy: RowChunkMap       # pre recv -> proxy buffer rows (sum)
    pre_out_splits: List[int]
    pre_total_recv: int
    proxy_rows: int                   # proxy buffer rows (== inter send buffer)
    inter_in_splits: List[int]        # per peer node
    inter_recv_reduce: RowChunkMap    # inter recv -> local dkv rows (sum)
    inter_out_splits: List[int]
    inter_total_recv: int


@dataclass
class CommMeta:
    stages_cast: List[GroupCastArg]
    stages_reduce: List[GroupReduceArg]
    # hierarchical realisation of the same stages (set iff 2D mesh + env flag)
    stages_cast_hier: Optional[List[HierGroupCastArg]] = None
    stages_reduce_hier: Optional[List[HierGroupReduceArg]] = None
    # native HIP-IPC pull plan (set iff MAGI_ATTENTION_NATIVE_GRPCOLL)
    stages_native: Optional[List["NativeStageMeta"]] = None
    stage_tokens_all: Optional[List[List[int]]] = None  # [stage][rank] S_r

    @property
    def overlap_degree(self) -> int:
        return len(self.stages_cast)


@dataclass
class NativeStageMeta:
    """One overlap stage's PULL plan for the native (HIP-IPC over xGMI)
    grpcoll transport (reference csrc/comm/grpcoll intranode kernels;
    MI355X-first pull redesign — see csrc/grpcoll.hip header).
    Pieces are [peer, src_row, dst_row, n_rows] int32 quadruples."""

    cast_pieces: List[Tuple[int, int, int, int]]   # peer kv window -> stage
    reduce_pieces: List[Tuple[int, int, int, int]]  # peer partial win -> dkv
    stage_tokens: int            # S (stage buffer rows = 2*S)
    cast_consumers: int          # ranks that pull from MY kv window
    reduce_consumers: int        # ranks that pull from MY partial window
    _cache: Dict[str, Tuple[torch.Tensor, torch.Tensor]] = field(
        default_factory=dict, repr=False
    )

    def to_device(self, device):
        key = str(device)
        if key not in self._cache:
            self._cache[key] = (
                torch.tensor(self.cast_pieces or [[0, 0, 0, 0]],
                             dtype=torch.int32, device=device).reshape(-1, 4),
                torch.tensor(self.reduce_pieces or [[0, 0, 0, 0]],
                             dtype=torch.int32, device=device).reshape(-1, 4),
            )
        return self._cache[key]


@dataclass
class QoCommMeta:
    """QO-comm plan (MAGI_ATTENTION_QO_COMM=1; reference env/comm.py:72,
    dist_attn.py:1659 _fetch_remote_qo_do_lse): remote slices are computed at
    the rank hosting their K side. Built from the solver run on the
    TRANSPOSED mask, so the remote-K machinery emits remote-Q tables."""

    calc: "CalcMeta"                  # orientation-restored stage args
    stages_cast1: List[GroupCastArg]  # single-tensor cast (fwd q)
    stages_cast2: List[GroupCastArg]  # doubled cast (bwd q+do / lse+dpsum)
    stages_reduce1: List[GroupReduceArg]  # partial (out,lse) fwd / dq bwd


@dataclass
class CalcMeta:
    host_arg: AttnArg
    stage_args: List[AttnArg]
