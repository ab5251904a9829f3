"""P2P roll of dispatched tensors (reference functional/roll.py:448 roll_p2p):
cyclic shift of the padded global sequence without materialising it — here as
one a2av permutation. gloo ws=2/4 vs dispatch(torch.roll(global))."""
import os
import socket

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp


def _free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return p


def _worker(rank, ws, port, shift):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=ws)
    try:
        from magi_attention.api import dispatch, magi_attn_flex_key, roll
        from magi_attention.api.magi_attn_interface import (
            dist_attn_runtime_dict_mgr,
        )
        from magi_attention.common.ranges import AttnRanges
        from magi_attention.config import DispatchConfig, DistAttnConfig
        from magi_attention.functional import roll_func

        total = 704  # needs pad with chunk 64 at ws 4 -> exercises pad rows
        key = magi_attn_flex_key(
            AttnRanges.from_ranges([[0, total]]),
            AttnRanges.from_ranges([[0, total]]),
            [1], total, total, 2, 2, 16,
            cp_group_or_mesh=dist.group.WORLD,
            dist_attn_config=DistAttnConfig(
                dispatch_config=DispatchConfig(chunk_size=64)
            ),
        )
        mgr = dist_attn_runtime_dict_mgr[key]
        g = torch.Generator().manual_seed(55)
        x = torch.randn(total, 2, 16, generator=g, dtype=torch.float64)
        xl = dispatch(x, key)

        got = roll(xl, key, shifts=shift)
        # reference result: roll the PADDED global tensor, re-dispatch
        padded = torch.cat(
            [x, torch.zeros(key.pad_size, 2, 16, dtype=x.dtype)]
        )
        want = mgr.dispatch_qo(torch.roll(padded, shifts=shift, dims=0))
        assert torch.equal(got, want), "roll mismatch"

        # functional form + autograd (backward = roll by -shift)
        xl2 = xl.clone().requires_grad_(True)
        out = roll_func(xl2, shift, mgr.dispatch_meta, mgr.cp_group)
        out.sum().backward()
        assert torch.equal(
            xl2.grad, torch.ones_like(xl2)
        ), "roll backward must be a permutation of ones"
    finally:
        dist.destroy_process_group()


@pytest.mark.parametrize("ws", [2, 4])
@pytest.mark.parametrize("shift", [1, 100, -37])
def test_roll_p2p(ws, shift):
    port = _free_port()
    mp.spawn(_worker, args=(ws, port, shift), nprocs=ws, join=True)
