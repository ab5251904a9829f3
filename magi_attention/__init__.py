# magi_attention — MI355X-native distributed flex-flash-attention engine.
#
# A from-scratch CDNA4/gfx950 rebuild with the API surface of
# SandAI-org/MagiAttention (the reference). Compute path: hand-written HIP
# kernels behind a C-ABI (include/magi_ffa.h) + RCCL-over-xGMI collectives.
__version__ = "0.1.0"
# reference surface: __init__.py:51-55 exposes the install-stamped version
# under both names (an AOT in-tree build always has one)
git_version = __version__
version = __version__

from . import comm, common, config, env, functional  # noqa: F401
from . import magi_attn_comm, magi_attn_ext  # noqa: F401
from .dist_attn_runtime_mgr import (  # noqa: F401
    init_dist_attn_runtime_key,
    init_dist_attn_runtime_mgr,
)
