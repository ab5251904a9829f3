"""A/B of the fwd grid's head-major XCD-affinity gate at the judged 8k
config (exactly at the 4 MB L2 boundary) and neighbors."""
import os
import sys
import time

sys.path.insert(0, ".")
import torch

from magi_attention.functional import flex_flash_attn_func


def bench(n, hq, d, steps=10):
    q = (torch.randn(n, hq, d) * 0.5).bfloat16().cuda()
    k = (torch.randn(n, hq, d) * 0.5).bfloat16().cuda()
    v = (torch.randn(n, hq, d) * 0.5).bfloat16().cuda()
    qr = torch.tensor([[0, n]], dtype=torch.int32, device="cuda")
    tm = torch.tensor([1], dtype=torch.int32, device="cuda")
    fl = 4 * (n * (n + 1) // 2) * hq * d

    def fwd():
        with torch.no_grad():
            flex_flash_attn_func(q, k, v, qr, qr.clone(), tm, max_seqlen_q=n)

    for _ in range(6):
        fwd()
    torch.cuda.synchronize()
    best = {}
    for _ in range(3):
        for hm in ("1", "0"):
            os.environ["MAGI_FWD_HEADMAJOR"] = hm
            fwd()
            torch.cuda.synchronize()
            t0 = time.perf_counter()
            for _ in range(steps):
                fwd()
            torch.cuda.synchronize()
            dt = (time.perf_counter() - t0) / steps
            best[hm] = min(best.get(hm, 1e9), dt)
    os.environ.pop("MAGI_FWD_HEADMAJOR", None)
    for hm in ("1", "0"):
        print(f"n={n:6d} h{hq} headmajor={hm}: {best[hm]*1e3:7.3f} ms "
              f"{fl/best[hm]/1e12:6.1f} TF")
    print(f"n={n:6d} ratio hm1/hm0: {best['0']/best['1']:.3f}x")


bench(8192, 32, 128)
bench(4096, 32, 128)
bench(16384, 32, 128)

bench(32768, 32, 128, steps=5)
bench(65536, 32, 128, steps=3)


def bench_bwd(n, hq, d, steps=3):
    import ctypes

    from magi_attention import _ffa_lib
    from magi_attention._ffa_lib import check
    from tests.gpu_bwd_exp1 import bwd_args, make_case, time_pass

    lib = _ffa_lib.lib()
    case = make_case(n=n, hq=hq, hk=hq, d=d)
    q, k, v, do, out, lse, qr, tm, *_ = case
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
        print(f"bwd n={n:6d} {entry.split('_')[-1]}: hm1 {best['1']:.2f} ms  "
              f"hm0 {best['0']:.2f} ms  ratio {best['0']/best['1']:.3f}x")


bench_bwd(16384, 32, 128)
bench_bwd(65536, 32, 128)
