# Reference surface: magi_attention/common/forward_meta.py (AttnForwardMeta).
from __future__ import annotations

from dataclasses import dataclass

import torch


@dataclass
class AttnForwardMeta:
    """Meta info of an attention forward pass (reference-compatible)."""

    lse: torch.Tensor | None = None
    max_logits: torch.Tensor | None = None
