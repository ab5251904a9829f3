"""Mask-IR containers (reference surface: meta/container/ — AttnSlice /
MultiKAttnSlice / AttnChunk / AttnBucket): the chunked representation of a
varlen flex mask that the dispatch layer hands to load balancing.

The reference's HostRankEntry/RemoteRankEntry/transfer_table live here too
in the reference tree; they are internals of ITS solver pipeline — this
rebuild's solver keeps its own stage tables (meta/containers.py CalcMeta /
CommMeta / NativeStageMeta), so those names are not replicated.
"""
from .bucket import AttnBucket
from .chunk import AttnChunk
from .slice import AttnSlice, MultiKAttnSlice

__all__ = ["AttnBucket", "AttnChunk", "AttnSlice", "MultiKAttnSlice"]
