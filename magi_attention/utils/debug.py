"""Remote-debug attach helper (reference surface: utils/debug.py
debugpy_listen, driven by MAGI_ATTENTION_ENABLE_REMOTE_DEBUG =
false | true (rank 0) | all | comma-separated rank list)."""
from __future__ import annotations

import os

import torch.distributed as dist


def debugpy_listen() -> None:  # pragma: no cover
    flag = os.environ.get("MAGI_ATTENTION_ENABLE_REMOTE_DEBUG", "false").lower()
    if flag == "false":
        return
    try:
        import debugpy
    except ImportError as e:
        raise RuntimeError(
            "MAGI_ATTENTION_ENABLE_REMOTE_DEBUG is set but debugpy is not "
            "installed in this environment"
        ) from e

    rank = dist.get_rank() if dist.is_initialized() else 0
    world = dist.get_world_size() if dist.is_initialized() else 1
    if flag == "true":
        ranks = [0]
    elif flag == "all":
        ranks = list(range(world))
    else:
        ranks = [int(i) for i in flag.split(",")]
    if rank in ranks:
        port = 37777 + rank
        print(f"[rank {rank}] Starting remote debug on port {port}")
        debugpy.listen(("127.0.0.1", port))
        debugpy.wait_for_client()
