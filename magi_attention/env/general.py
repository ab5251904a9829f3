"""Alias of the flat env flags (reference env/general.py)."""
from . import *  # noqa: F401,F403
from . import (  # noqa: F401
    is_auto_range_merge_enable,
    is_deterministic_mode_enable,
    kernel_backend,
)
