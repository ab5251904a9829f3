# Public drop-in API (reference magi_attention/api/magi_attn_interface.py —
# signatures and semantics kept: magi_attn_varlen_key:160, magi_attn_flex_key:440,
# dispatch:887, undispatch:924, calc_attn:1041, get_position_ids:1112,
# get_most_recent_key:1136, clear_cache:1157, DistAttnRuntimeDictManager:64).
from __future__ import annotations

import warnings
from typing import List, Optional, Tuple, Union

import torch
import torch.distributed as dist

from .. import env
from ..common.enum import AttnMaskType
from ..common.range import AttnRange
from ..common.ranges import AttnRanges
from ..config import DistAttnConfig
from ..dist_attn_runtime_mgr import (
    DistAttnRuntimeDict,
    DistAttnRuntimeKey,
    DistAttnRuntimeMgr,
    init_dist_attn_runtime_key,
    init_dist_attn_runtime_mgr,
)
from .functools import compute_pad_size, pad_at_dim, unpad_at_dim

GeneralAttnMaskType = Union[str, AttnMaskType, List[Union[str, AttnMaskType]]]


class DistAttnRuntimeDictManager:
    """Per-cp-group LRU caches (reference magi_attn_interface.py:64-157)."""

    def __init__(self, max_size_per_group: int = 100):
        self.max_size = max_size_per_group
        self._per_group: dict = {}
        self._tag_stamp: dict = {}   # tag -> monotonic last-touch stamp
        self._stamp = 0

    def _touch(self, tag):
        self._stamp += 1
        self._tag_stamp[tag] = self._stamp

    def _cache_for(self, key: DistAttnRuntimeKey) -> DistAttnRuntimeDict:
        tag = key.cp_group_tag
        if tag not in self._per_group:
            self._per_group[tag] = DistAttnRuntimeDict(self.max_size)
        return self._per_group[tag]

    def get(self, key, default=None):
        return self._cache_for(key).get(key, default)

    def __contains__(self, key):
        return key in self._cache_for(key)

    def __setitem__(self, key, value):
        self._touch(key.cp_group_tag)
        self._cache_for(key)[key] = value

    def __getitem__(self, key):
        self._touch(key.cp_group_tag)
        return self._cache_for(key)[key]

    def get_most_recent_key(self, cp_group=None):
        """Most recent key of the REQUESTED cp_group; with cp_group=None the
        globally most recently touched group's key (reference
        magi_attn_interface.py:1136)."""
        if cp_group is not None:
            from ..dist_attn_runtime_mgr import _group_tag

            cache = self._per_group.get(_group_tag(cp_group))
            return cache.most_recent_key() if cache is not None else None
        for tag in sorted(self._tag_stamp, key=self._tag_stamp.get,
                          reverse=True):
            cache = self._per_group.get(tag)
            if cache is not None:
                k = cache.most_recent_key()
                if k is not None:
                    return k
        return None

    def clear(self, cp_group=None):
        if cp_group is None:
            self._per_group.clear()
            self._tag_stamp.clear()
            return
        from ..dist_attn_runtime_mgr import _group_tag

        tag = _group_tag(cp_group)
        self._per_group.pop(tag, None)
        self._tag_stamp.pop(tag, None)


dist_attn_runtime_dict_mgr = DistAttnRuntimeDictManager()


def _resolve_group(cp_group_or_mesh):
    """Returns (flat cp_group, mesh_groups). mesh_groups is
    (intra_group, inter_group, ws_intra, ws_inter) iff a 2D DeviceMesh was
    passed AND MAGI_ATTENTION_HIERARCHICAL_COMM is on (reference
    api:632 cp_mesh._flatten(), comm_meta.py:227-228 intra=dim1, inter=dim0).
    """
    if cp_group_or_mesh is None:
        return dist.group.WORLD, None
    if isinstance(cp_group_or_mesh, dist.ProcessGroup):
        return cp_group_or_mesh, None
    try:
        mesh = cp_group_or_mesh
        if getattr(mesh, "ndim", 1) == 2:
            flat = mesh._flatten().get_group()
            if env.is_hierarchical_comm_enable():
                inter = mesh.get_group(0)
                intra = mesh.get_group(1)
                return flat, (
                    intra, inter,
                    dist.get_world_size(intra), dist.get_world_size(inter),
                )
            return flat, None
        return mesh.get_group(), None
    except AttributeError as e:  # pragma: no cover
        raise ValueError(f"unsupported cp_group_or_mesh: {e}")


def magi_attn_flex_key(
    q_ranges: AttnRanges,
    k_ranges: AttnRanges,
    attn_mask_type: GeneralAttnMaskType,
    total_seqlen_q: int,
    total_seqlen_k: int,
    num_heads_q: int,
    num_heads_kv: int,
    head_dim: int,
    pad_size: int = 0,
    cp_group_or_mesh=None,
    dist_attn_config: DistAttnConfig = DistAttnConfig(),
    is_same_source: bool = True,
    is_q_permutable: bool = True,
    is_k_permutable: bool = True,
    chunk_size: Optional[int] = None,
) -> DistAttnRuntimeKey:
    """The most flexible key-creation interface (reference :440)."""
    # cross-attn settings: the reference raises NotImplementedError for every
    # is_same_source=False case (_make_dispatch_meta.py:203-214) — mirrored
    if not is_same_source:
        if is_q_permutable and not is_k_permutable:
            raise NotImplementedError(
                "A cross-attn setting for encoder-decoder transformer like T5."
            )
        if not is_q_permutable and is_k_permutable:
            raise NotImplementedError(
                "A cross-attn setting for multi-modal transformer "
                "with external encoders."
            )
        if is_q_permutable and is_k_permutable:
            raise NotImplementedError(
                "An unknown case as a pure cross-attn setting."
            )
        raise NotImplementedError("A trivial case with no need to dispatch.")
    assert total_seqlen_q == total_seqlen_k, "self-attn requires equal seqlens"
    if torch.cuda.is_available():  # kernel constraint; CPU test oracle is free
        assert head_dim in (64, 128), (
            "context-parallel runs support head_dim 64/128 in r1 (other dims "
            "run zero-padded through the single-GPU functional API; the "
            "padded CP path lands next round)"
        )
    group, mesh_groups = _resolve_group(cp_group_or_mesh)
    cp_size = dist.get_world_size(group)
    if chunk_size is not None:
        warnings.warn(
            "chunk_size arg is deprecated; pass DispatchConfig(chunk_size=...)",
            DeprecationWarning,
        )
        from dataclasses import replace

        dist_attn_config = replace(
            dist_attn_config,
            dispatch_config=type(dist_attn_config.dispatch_config)(
                chunk_size=chunk_size,
                alg=dist_attn_config.dispatch_config.alg,
            ),
        )
    ck = dist_attn_config.dispatch_config.chunk_size
    computed_pad = compute_pad_size(total_seqlen_q, cp_size, ck)
    if pad_size not in (0, computed_pad):
        warnings.warn(
            "pad_size arg is deprecated and recomputed internally",
            DeprecationWarning,
        )
    pad_size = computed_pad

    key = init_dist_attn_runtime_key(
        q_ranges, k_ranges, attn_mask_type, total_seqlen_q, total_seqlen_k,
        pad_size, ck, num_heads_q, num_heads_kv, head_dim, group,
        dist_attn_config,
    )
    if key not in dist_attn_runtime_dict_mgr:
        dist_attn_runtime_dict_mgr[key] = init_dist_attn_runtime_mgr(
            key, group, dist_attn_config, mesh_groups=mesh_groups
        )
    return key


def magi_attn_varlen_key(
    cu_seqlens_q: torch.Tensor,
    cu_seqlens_k: torch.Tensor,
    total_seqlen_q: int = None,
    total_seqlen_k: int = None,
    num_heads_q: int = 1,
    num_heads_kv: int = 1,
    head_dim: int = 128,
    pad_size: int = 0,
    cp_group_or_mesh=None,
    causal: bool = False,
    window_size: Tuple[int, int] = (-1, -1),
    dist_attn_config: DistAttnConfig = DistAttnConfig(),
) -> DistAttnRuntimeKey:
    """Varlen (cu_seqlens) front-end (reference :160)."""
    from .functools import infer_attn_mask_from_cu_seqlens

    q_ranges, k_ranges, types, tq, tk = infer_attn_mask_from_cu_seqlens(
        cu_seqlens_q, cu_seqlens_k, causal=causal, window_size=window_size
    )
    return magi_attn_flex_key(
        q_ranges, k_ranges, types,
        total_seqlen_q or tq, total_seqlen_k or tk,
        num_heads_q, num_heads_kv, head_dim, pad_size, cp_group_or_mesh,
        dist_attn_config,
    )


def dispatch(x: torch.Tensor, key: DistAttnRuntimeKey) -> torch.Tensor:
    """Pad + scatter x to this rank's permuted local shard (reference :887)."""
    mgr = dist_attn_runtime_dict_mgr[key]
    x = pad_at_dim(x, 0, key.pad_size)
    return mgr.dispatch_qo(x)


def undispatch(x_local: torch.Tensor, key: DistAttnRuntimeKey) -> torch.Tensor:
    """Gather local shards back to the global (unpadded) order (reference :924)."""
    mgr = dist_attn_runtime_dict_mgr[key]
    full = mgr.undispatch_qo(x_local)
    return unpad_at_dim(full, 0, key.total_seqlen_q)


def calc_attn(
    q: torch.Tensor,
    k: torch.Tensor,
    v: torch.Tensor,
    key: DistAttnRuntimeKey,
    sink: Optional[torch.Tensor] = None,
    softmax_scale: Optional[float] = None,
    softcap: float = 0.0,
    return_max_logits: bool = False,
):
    """Distributed flex-flash-attention on dispatched shards (reference :1041).
    Returns (out, AttnForwardMeta(lse, max_logits)). `sink` is the replicated
    [seqlen_sink, num_heads_q] sink tensor; `max_logits` is all-reduced MAX
    across the cp group (forward_meta.py:28)."""
    mgr = dist_attn_runtime_dict_mgr[key]
    return mgr.calc_attn(q, k, v, sink=sink, softmax_scale=softmax_scale,
                         softcap=softcap, return_max_logits=return_max_logits)


def magi_attn_flex_dispatch(x, *args, **kwargs):
    """Key creation + dispatch in one call (reference :725)."""
    key = magi_attn_flex_key(*args, **kwargs)
    return dispatch(x, key), key


def magi_attn_varlen_dispatch(x, *args, **kwargs):
    key = magi_attn_varlen_key(*args, **kwargs)
    return dispatch(x, key), key


def get_position_ids(key: DistAttnRuntimeKey) -> torch.Tensor:
    mgr = dist_attn_runtime_dict_mgr[key]
    return mgr.get_position_ids()


def get_most_recent_key(cp_group=None) -> Optional[DistAttnRuntimeKey]:
    return dist_attn_runtime_dict_mgr.get_most_recent_key(cp_group)


def clear_cache(cp_group=None) -> None:
    dist_attn_runtime_dict_mgr.clear(cp_group)


def roll(x: torch.Tensor, key: DistAttnRuntimeKey, shifts: int = 1) -> torch.Tensor:
    """Cyclic shift along the (padded) GLOBAL sequence of a dispatched tensor
    (reference :960; used by MTP), via the P2P roll — each row moves once,
    peak memory stays O(N/P) (functional/roll.py)."""
    from ..functional.roll import roll_p2p

    mgr = dist_attn_runtime_dict_mgr[key]
    return roll_p2p(x, shifts, mgr.dispatch_meta, mgr.cp_group)


def roll_simple(x: torch.Tensor, key: DistAttnRuntimeKey, shifts: int = 1):
    return roll(x, key, shifts)


def make_flex_key_for_new_mask_after_dispatch(
    q_ranges: AttnRanges,
    k_ranges: AttnRanges,
    attn_mask_type: GeneralAttnMaskType,
    total_seqlen_q: int,
    total_seqlen_k: int,
    key_for_dispatch: DistAttnRuntimeKey,
    dist_attn_config: Optional[DistAttnConfig] = None,
) -> DistAttnRuntimeKey:
    """New mask, SAME dispatch solution (reference api:1315) — for hybrid-attn
    models applying several masks within one training pass."""
    base_mgr = dist_attn_runtime_dict_mgr[key_for_dispatch]
    cfg = dist_attn_config or base_mgr.config
    from dataclasses import replace

    key = init_dist_attn_runtime_key(
        q_ranges, k_ranges, attn_mask_type, total_seqlen_q, total_seqlen_k,
        key_for_dispatch.pad_size, key_for_dispatch.chunk_size,
        key_for_dispatch.num_heads_q, key_for_dispatch.num_heads_kv,
        key_for_dispatch.head_dim, base_mgr.cp_group, cfg,
    )
    key = replace(key, dispatch_from=hash(key_for_dispatch))
    if key not in dist_attn_runtime_dict_mgr:
        dist_attn_runtime_dict_mgr[key] = DistAttnRuntimeMgr(
            key, base_mgr.cp_group, cfg, reuse_dispatch_from=base_mgr,
            mesh_groups=base_mgr.mesh_groups,
        )
    return key


def make_varlen_key_for_new_mask_after_dispatch(
    cu_seqlens_q: torch.Tensor,
    cu_seqlens_k: torch.Tensor,
    key_for_dispatch: DistAttnRuntimeKey,
    causal: bool = False,
    window_size: Tuple[int, int] = (-1, -1),
    dist_attn_config: Optional[DistAttnConfig] = None,
) -> DistAttnRuntimeKey:
    """Varlen front-end of make_flex_key_for_new_mask_after_dispatch
    (reference api:1167)."""
    from .functools import infer_attn_mask_from_cu_seqlens

    q_ranges, k_ranges, types, tq, tk = infer_attn_mask_from_cu_seqlens(
        cu_seqlens_q, cu_seqlens_k, causal=causal, window_size=window_size
    )
    return make_flex_key_for_new_mask_after_dispatch(
        q_ranges, k_ranges, types, tq, tk, key_for_dispatch, dist_attn_config
    )
