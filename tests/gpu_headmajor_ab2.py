"""Head-major gate decision points, part 2: 128k GQA dkv (fp8-config bwd
shape) and the varlen-16k product path."""
import os
import sys
import time

sys.path.insert(0, ".")
import torch

from magi_attention.functional import flex_flash_attn_func


def bench_bwd(n, hq, d, steps=2):
    from magi_attention import _ffa_lib
    from magi_attention._ffa_lib import check
    from tests.gpu_bwd_exp1 import bwd_args, make_case, time_pass

    lib = _ffa_lib.lib()
    case = make_case(n=n, hq=hq, hk=hq, d=d)
    dq = torch.zeros(n, hq, d, dtype=torch.float32, device="cuda")
    dk = torch.zeros(n, hq, d, dtype=torch.float32, device="cuda")
    dv = torch.zeros(n, hq, d, dtype=torch.float32, device="cuda")
    dpsum = torch.empty(n, hq, dtype=torch.float32, device="cuda")
    a = bwd_args(case, dq, dk, dv, dpsum)
    check(lib.magi_ffa_bwd_preprocess(a), "pre")
    torch.cuda.synchronize()
    for entry in ("magi_ffa_bwd_dq", "magi_ffa_bwd_dkv"):
        best = {}
        for _ in range(2):
            for hm in ("1", "0"):
                os.environ["MAGI_BWD_HEAD_MAJOR"] = hm
                ms = time_pass(lambda: check(getattr(lib, entry)(a), entry),
                               steps=steps, warm=1)
                best[hm] = min(best.get(hm, 1e9), ms)
        os.environ.pop("MAGI_BWD_HEAD_MAJOR", None)
        print(f"bwd n={n:6d} h{hq} {entry.split('_')[-1]}: hm1 {best['1']:.2f} ms  "
              f"hm0 {best['0']:.2f} ms  ratio {best['0']/best['1']:.3f}x")


def bench_varlen(hm_env_vals, docs=8, dl=2048, hq=16, d=128, steps=10):
    n = docs * dl
    q = (torch.randn(n, hq, d) * 0.5).bfloat16().cuda().requires_grad_(True)
    k = (torch.randn(n, hq, d) * 0.5).bfloat16().cuda().requires_grad_(True)
    v = (torch.randn(n, hq, d) * 0.5).bfloat16().cuda().requires_grad_(True)
    qr = torch.tensor([[i * dl, (i + 1) * dl] for i in range(docs)],
                      dtype=torch.int32, device="cuda")
    tm = torch.tensor([1] * docs, dtype=torch.int32, device="cuda")
    do = torch.randn_like(q)
    area = docs * dl * (dl + 1) // 2
    fl = 4 * area * hq * d * 3.5

    def step():
        o, _ = flex_flash_attn_func(q, k, v, qr, qr.clone(), tm,
                                    max_seqlen_q=dl, max_seqlen_k=dl)
        o.backward(do)
        q.grad = k.grad = v.grad = None

    for _ in range(4):
        step()
    torch.cuda.synchronize()
    best = {}
    for _ in range(3):
        for hm in hm_env_vals:
            for var, val in hm.items():
                os.environ[var] = val
            step()
            torch.cuda.synchronize()
            t0 = time.perf_counter()
            for _ in range(steps):
                step()
            torch.cuda.synchronize()
            dt = (time.perf_counter() - t0) / steps
            key = tuple(sorted(hm.items()))
            best[key] = min(best.get(key, 1e9), dt)
            for var in hm:
                os.environ.pop(var, None)
    for key, dt in best.items():
        print(f"varlen8x2k {dict(key)}: {dt*1e3:7.3f} ms  {fl/dt/1e12:6.1f} TF")


bench_varlen([
    {"MAGI_BWD_HEAD_MAJOR": "1", "MAGI_FWD_HEADMAJOR": "1"},
    {"MAGI_BWD_HEAD_MAJOR": "0", "MAGI_FWD_HEADMAJOR": "0"},
])
bench_bwd(131072, 16, 128, steps=2)
