#!/usr/bin/env python3
"""Minimal drop-in usage demo (mirrors the reference's torch_native example
shape, examples/torch_native/main.py): a toy attention layer training step on
dispatched shards with context parallelism.

Run (single GPU):     python examples/torch_native_demo.py
Run (cp=N, one node): torchrun --nproc-per-node N --master-addr 127.0.0.1 \
                          examples/torch_native_demo.py
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.distributed as dist

from magi_attention.api import (
    calc_attn,
    dispatch,
    magi_attn_varlen_key,
    undispatch,
)
from magi_attention.config import DispatchConfig, DistAttnConfig


def main():
    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29541")
    use_gpu = torch.cuda.is_available()
    if use_gpu:
        torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", rank)))
    dist.init_process_group("nccl" if use_gpu else "gloo",
                            rank=rank, world_size=world)
    device = "cuda" if use_gpu else "cpu"

    # a varlen batch of 3 causal documents, total 6144 tokens
    cu = torch.tensor([0, 2048, 4096, 6144], dtype=torch.int32)
    hq, hkv, hd = 8, 2, 128
    key = magi_attn_varlen_key(
        cu, cu, num_heads_q=hq, num_heads_kv=hkv, head_dim=hd,
        cp_group_or_mesh=dist.group.WORLD, causal=True,
        dist_attn_config=DistAttnConfig(
            dispatch_config=DispatchConfig(chunk_size=512)
        ),
    )

    total = int(cu[-1])
    torch.manual_seed(7)
    wq = torch.randn(hq * hd, hq * hd, device=device, dtype=torch.bfloat16,
                     requires_grad=True)
    x = torch.randn(total, hq, hd, device=device, dtype=torch.bfloat16)

    xl = dispatch(x, key)                       # local permuted shard
    q = (xl.reshape(-1, hq * hd) @ wq).reshape(-1, hq, hd)
    kv = xl.reshape(-1, hq, hd)[:, :hkv]
    out, meta = calc_attn(q, kv.contiguous(), kv.contiguous(), key)
    loss = out.float().square().mean()
    loss.backward()
    full_out = undispatch(out, key)

    if rank == 0:
        print(f"cp={world} loss={loss.item():.5f} "
              f"out={tuple(full_out.shape)} grad_norm={wq.grad.norm():.3f}")
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
