"""Alias of the flat env flags (reference env/comm.py)."""
from . import *  # noqa: F401,F403
from . import (  # noqa: F401
    is_hierarchical_comm_enable,
    is_native_grpcoll_enable,
    is_qo_comm_enable,
)
