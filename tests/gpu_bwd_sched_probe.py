"""Backward pass scheduling probe (not a pytest test): times dq / dv / dk
solo, then serial / 2-stream / 3-stream arrangements, on the 64k causal
bench shape. Run on a GPU box: python tests/gpu_bwd_sched_probe.py"""
import ctypes
import sys

import torch

sys.path.insert(0, ".")
from magi_attention import _ffa_lib
from magi_attention._ffa_lib import (
    MagiFfaBwdArgs,
    check,
    current_stream_ptr,
    ptr,
)

T, HQ, HK, D = 65536, 32, 32, 128
dev = "cuda"
torch.manual_seed(0)
q = torch.randn(T, HQ, D, device=dev).bfloat16()
k = torch.randn(T, HK, D, device=dev).bfloat16()
v = torch.randn(T, HK, D, device=dev).bfloat16()
do = torch.randn(T, HQ, D, device=dev).bfloat16()
out = torch.randn(T, HQ, D, device=dev).bfloat16()
lse = torch.randn(T, HQ, device=dev).float().abs() + 5
dps = torch.randn(T, HQ, device=dev).float()
dq = torch.zeros(T, HQ, D, device=dev).float()
dk = torch.zeros(T, HK, D, device=dev).float()
dv = torch.zeros(T, HK, D, device=dev).float()
qr = torch.tensor([[0, T]], dtype=torch.int32, device=dev)
kr = torch.tensor([[0, T]], dtype=torch.int32, device=dev)
tm = torch.tensor([1], dtype=torch.int32, device=dev)

lib = _ffa_lib.lib()
args = MagiFfaBwdArgs(
    dout=ptr(do), q=ptr(q), k=ptr(k), v=ptr(v), out=ptr(out), lse=ptr(lse),
    dq=ptr(dq), dk=ptr(dk), dv=ptr(dv), dpsum=ptr(dps),
    q_ranges=ptr(qr), k_ranges=ptr(kr), attn_type_map=ptr(tm),
    n_ranges=1, total_q=T, total_k=T, hq=HQ, hk=HK, d=D,
    max_seqlen_k=T, out_is_fp32=0, softmax_scale=D ** -0.5, softcap=0.0,
    cu_margin=0, stream=current_stream_ptr(),
)

FL = {"dq": 1.5, "dv": 1.0, "dk": 1.5, "dkv": 2.0}
FWD_FLOPS = 4 * (T * T // 2) * HQ * D  # 2 matmuls over the causal area


def fns():
    return {"dq": lib.magi_ffa_bwd_dq, "dv": lib.magi_ffa_bwd_dv,
            "dk": lib.magi_ffa_bwd_dk, "dkv": lib.magi_ffa_bwd_dkv}


def time_arrangement(name, launcher, reps=3):
    torch.cuda.synchronize()
    launcher()  # warmup
    torch.cuda.synchronize()
    e0, e1 = torch.cuda.Event(True), torch.cuda.Event(True)
    e0.record()
    for _ in range(reps):
        launcher()
    e1.record()
    torch.cuda.synchronize()
    ms = e0.elapsed_time(e1) / reps
    print(f"{name:28s} {ms:8.2f} ms")
    return ms


main = torch.cuda.current_stream()
s1 = torch.cuda.Stream()
s2 = torch.cuda.Stream()

solo = {}
for nm in ("dq", "dv", "dk", "dkv"):
    fn = fns()[nm]
    args.stream = current_stream_ptr()
    solo[nm] = time_arrangement(
        f"solo {nm}", lambda fn=fn: check(fn(args), "x"))
    eff = FL[nm] * FWD_FLOPS / (solo[nm] / 1e3) / 1e12
    print(f"{'':28s} -> {eff:6.1f} TF effective")

print(f"sum solo dq+dv+dk = {solo['dq']+solo['dv']+solo['dk']:.1f} ms")


def serial():
    args.stream = current_stream_ptr()
    check(lib.magi_ffa_bwd_dq(args), "dq")
    check(lib.magi_ffa_bwd_dv(args), "dv")
    check(lib.magi_ffa_bwd_dk(args), "dk")


def two_stream():  # current production arrangement
    ev = torch.cuda.Event()
    ev.record(main)
    s1.wait_event(ev)
    args.stream = ctypes.c_void_p(s1.cuda_stream)
    check(lib.magi_ffa_bwd_dq(args), "dq")
    args.stream = ctypes.c_void_p(main.cuda_stream)
    check(lib.magi_ffa_bwd_dv(args), "dv")
    check(lib.magi_ffa_bwd_dk(args), "dk")
    ev2 = torch.cuda.Event()
    ev2.record(s1)
    main.wait_event(ev2)


def three_stream():
    ev = torch.cuda.Event()
    ev.record(main)
    s1.wait_event(ev)
    s2.wait_event(ev)
    args.stream = ctypes.c_void_p(s1.cuda_stream)
    check(lib.magi_ffa_bwd_dq(args), "dq")
    args.stream = ctypes.c_void_p(s2.cuda_stream)
    check(lib.magi_ffa_bwd_dv(args), "dv")
    args.stream = ctypes.c_void_p(main.cuda_stream)
    check(lib.magi_ffa_bwd_dk(args), "dk")
    for s in (s1, s2):
        e = torch.cuda.Event()
        e.record(s)
        main.wait_event(e)


def two_stream_fused():
    ev = torch.cuda.Event()
    ev.record(main)
    s1.wait_event(ev)
    args.stream = ctypes.c_void_p(s1.cuda_stream)
    check(lib.magi_ffa_bwd_dq(args), "dq")
    args.stream = ctypes.c_void_p(main.cuda_stream)
    check(lib.magi_ffa_bwd_dkv(args), "dkv")
    ev2 = torch.cuda.Event()
    ev2.record(s1)
    main.wait_event(ev2)


time_arrangement("serial dq;dv;dk", serial)
time_arrangement("2-stream dq | dv;dk (prod)", two_stream)
time_arrangement("3-stream dq|dv|dk", three_stream)
time_arrangement("2-stream dq | fused dkv", two_stream_fused)
