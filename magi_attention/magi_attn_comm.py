"""Reference magi_attn_comm = the native (NVSHMEM-style) grpcoll comm
extension. This engine's transport is RCCL a2av over xGMI
(comm/primitive.py); a HIP-IPC analogue is a later-round item."""


def __getattr__(name):  # pragma: no cover
    raise NotImplementedError(
        f"magi_attn_comm.{name}: native grpcoll is replaced by the RCCL "
        "a2av transport (comm/primitive.py); a HIP-IPC analogue lands "
        "in a later round"
    )
