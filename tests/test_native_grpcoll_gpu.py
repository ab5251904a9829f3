"""Native grpcoll (HIP-IPC pull transport, MAGI_ATTENTION_NATIVE_GRPCOLL=1)
end-to-end on ONE GPU with TWO processes: gloo exchanges the IPC handles and
both ranks' kernels share cuda:0 — the same dmabuf-IPC mechanics as the
production single-node multi-GPU case (reference
csrc/comm/grpcoll intranode path).

The full product path runs: key -> dispatch -> calc_attn (native casts) ->
backward (native pull-sum reduces); each rank checks its local rows/grads
against the single-GPU kernel run on the global tensors."""
import os
import socket

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

requires_gpu = pytest.mark.gpu


def _free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return p


def _worker(rank, ws, port):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["MAGI_ATTENTION_NATIVE_GRPCOLL"] = "1"
    dist.init_process_group("gloo", rank=rank, world_size=ws)
    try:
        torch.cuda.set_device(0)
        from magi_attention.api import (
            calc_attn, dispatch, get_position_ids, magi_attn_flex_key,
        )
        from magi_attention.common.ranges import AttnRanges
        from magi_attention.config import (
            DispatchConfig, DistAttnConfig, OverlapConfig,
        )
        from magi_attention.functional import flex_flash_attn_func

        total, hq, hk, d = 2048, 4, 2, 128
        g = torch.Generator().manual_seed(11)
        q = (torch.randn(total, hq, d, generator=g) * 0.5).bfloat16().cuda()
        k = (torch.randn(total, hk, d, generator=g) * 0.5).bfloat16().cuda()
        v = (torch.randn(total, hk, d, generator=g) * 0.5).bfloat16().cuda()
        dout = (torch.randn(total, hq, d, generator=g) * 0.5).bfloat16().cuda()
        key = magi_attn_flex_key(
            AttnRanges.from_ranges([[0, total]]),
            AttnRanges.from_ranges([[0, total]]),
            "causal", total, total, hq, hk, d,
            cp_group_or_mesh=dist.group.WORLD,
            dist_attn_config=DistAttnConfig(
                dispatch_config=DispatchConfig(chunk_size=256),
                overlap_config=OverlapConfig(degree=2, min_chunk_size=128),
            ),
        )
        from magi_attention.api.magi_attn_interface import (
            dist_attn_runtime_dict_mgr,
        )
        rt = dist_attn_runtime_dict_mgr[key].runtime
        assert rt.comm_meta.stages_native is not None, "native plan missing"

        ql = dispatch(q, key).requires_grad_(True)
        kl = dispatch(k, key).requires_grad_(True)
        vl = dispatch(v, key).requires_grad_(True)
        out_l, _ = calc_attn(ql, kl, vl, key)
        dout_l = dispatch(dout, key)
        (out_l.float() * dout_l.float()).sum().backward()
        torch.cuda.synchronize()
        assert rt._native is not None, "native transport was not used"

        # single-GPU reference on the same global tensors
        qr = torch.tensor([[0, total]], dtype=torch.int32, device="cuda")
        tm = torch.tensor([1], dtype=torch.int32, device="cuda")
        q2 = q.clone().requires_grad_(True)
        k2 = k.clone().requires_grad_(True)
        v2 = v.clone().requires_grad_(True)
        o2, _ = flex_flash_attn_func(q2, k2, v2, qr, qr.clone(), tm)
        (o2.float() * dout.float()).sum().backward()
        torch.cuda.synchronize()

        pos = get_position_ids(key)
        tol = dict(atol=3e-2, rtol=3e-2)
        torch.testing.assert_close(out_l.float(), o2[pos].float(), **tol)
        torch.testing.assert_close(ql.grad.float(), q2.grad[pos].float(), **tol)
        torch.testing.assert_close(kl.grad.float(), k2.grad[pos].float(), **tol)
        torch.testing.assert_close(vl.grad.float(), v2.grad[pos].float(), **tol)

        # a second step reuses the windows (exercises the ack backpressure)
        out_l2, _ = calc_attn(ql.detach(), kl.detach(), vl.detach(), key)
        torch.cuda.synchronize()
        torch.testing.assert_close(out_l2.float(), o2[pos].float(), **tol)
        dist.barrier()
    finally:
        dist.destroy_process_group()


@requires_gpu
def test_native_grpcoll_two_procs_one_gpu():
    port = _free_port()
    mp.spawn(_worker, args=(2, port), nprocs=2, join=True)
