"""Timing probe for the index-attention forward kernel (not a pytest test).

DiT-shaped workload: ratio-128 PackGQA tokens gathering topk of a long KV.
Prints achieved TFLOP/s (2*2*tokens*hq*topk*d flops) and effective gather
bandwidth (K+V bytes actually read per launch).
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from magi_attention.functional.flex_flash_attn import (
    _flex_flash_attn_forward_index,
)
from magi_attention.utils import build_index_attn_indices


def run(tokens=4096, hq=128, d=128, s_kv=8192, topk=2048, iters=20):
    dev = "cuda"
    torch.manual_seed(0)
    q = torch.randn(tokens, hq, d, dtype=torch.bfloat16, device=dev)
    k = torch.randn(s_kv, 1, d, dtype=torch.bfloat16, device=dev)
    v = torch.randn_like(k)
    if s_kv <= 65536:
        idx = build_index_attn_indices(
            1, 1, tokens, s_kv, topk, topk, device=dev
        ).view(tokens, topk)
    else:
        # HBM-resident regime probe: sample WITH replacement (perf-only —
        # the without-replacement builder would need tokens x s_kv scores)
        idx = (
            torch.randint(0, s_kv, (tokens, topk), device=dev)
            .sort(dim=-1).values.int()
        )
    scale = d ** (-0.5)

    for _ in range(3):
        _flex_flash_attn_forward_index(q, k, v, idx, scale, 0.0)
    torch.cuda.synchronize()
    ev0, ev1 = torch.cuda.Event(True), torch.cuda.Event(True)
    ev0.record()
    for _ in range(iters):
        _flex_flash_attn_forward_index(q, k, v, idx, scale, 0.0)
    ev1.record()
    torch.cuda.synchronize()
    ms = ev0.elapsed_time(ev1) / iters
    flops = 4.0 * tokens * hq * topk * d
    gather_bytes = 2.0 * tokens * topk * d * 2  # K+V rows touched per token
    print(
        f"tokens={tokens} hq={hq} d={d} topk={topk}: {ms:.3f} ms  "
        f"{flops / ms / 1e9:.1f} TF  gather {gather_bytes / ms / 1e6:.0f} GB/s"
    )


if __name__ == "__main__":
    args = [int(a) for a in sys.argv[1:]]
    run(*args) if args else None
    if not args:
        run(4096, 128, 128, 8192, 2048)     # DiT ratio-128 dense-ish
        run(16384, 128, 128, 16384, 2048)   # bigger grid
        run(8192, 32, 128, 16384, 1024)     # ratio-32 shape (W1 head path)
        run(8192, 64, 64, 16384, 2048)      # d=64
        # HBM-resident gathers (KV pool 512 MB >> 32 MB aggregate L2)
        run(8192, 128, 128, 1048576, 2048)  # ratio-128, cold gather
        run(8192, 32, 128, 1048576, 1024)   # ratio-32, cold gather
