"""A/B: fused dkv MODE0 (8 waves, 32-row cadence) vs MODE3 "fat" (4 waves x
2 column tiles, 1 wave/SIMD, 512-reg budget). 64k dense causal + numerics."""
import ctypes
import os
import sys

sys.path.insert(0, ".")
import torch

from magi_attention import _ffa_lib
from magi_attention._ffa_lib import check, ptr
from tests.gpu_bwd_exp1 import bwd_args, make_case, time_pass


def stream_ptr():
    return ctypes.c_void_p(torch.cuda.current_stream().cuda_stream)


def main():
    lib = _ffa_lib.lib()
    case = make_case()
    q, k, v, do, out, lse, qr, tm, n, hq, hk, d = case
    dq = torch.zeros(n, hq, d, dtype=torch.float32, device="cuda")
    dk = torch.zeros(n, hk, d, dtype=torch.float32, device="cuda")
    dv = torch.zeros(n, hk, d, dtype=torch.float32, device="cuda")
    dpsum = torch.empty(n, hq, dtype=torch.float32, device="cuda")
    a = bwd_args(case, dq, dk, dv, dpsum)
    check(lib.magi_ffa_bwd_preprocess(a), "pre")
    torch.cuda.synchronize()

    area = n * (n + 1) // 2
    unit = 2 * area * hq * d

    def run(tag, env):
        for kk, vv in env.items():
            os.environ[kk] = vv
        ms = time_pass(lambda: check(lib.magi_ffa_bwd_dkv(a), "dkv"), steps=3)
        for kk in env:
            os.environ.pop(kk, None)
        tf = 4 * unit / (ms / 1e3) / 1e12
        print(f"{tag:24s} {ms:8.2f} ms  issued {tf:7.1f} TF ({tf/25:.1f}% peak)")
        return ms

    m0 = run("fused v2 (MODE0 W8)", {})
    m3 = run("fused fat (MODE3 W4)", {"MAGI_BWD_FAT": "1"})
    m6 = run("fused W6 64-row ring", {"MAGI_BWD_W6": "1"})
    run("fused v2 again", {})
    print(f"fat speedup: {m0 / m3:.3f}x   w6 speedup: {m0 / m6:.3f}x")

    # numerics: fat vs v2 grads must agree to fp32-atomic-order noise
    def grads(env):
        for kk, vv in env.items():
            os.environ[kk] = vv
        dk.zero_(); dv.zero_()
        a.stream = stream_ptr()
        check(lib.magi_ffa_bwd_dkv(a), "dkv")
        torch.cuda.synchronize()
        for kk in env:
            os.environ.pop(kk, None)
        return dk.clone(), dv.clone()

    g0 = grads({})
    for tag, env in (("fat", {"MAGI_BWD_FAT": "1"}),
                     ("w6", {"MAGI_BWD_W6": "1"})):
        g = grads(env)
        for name, x, y in zip(("dk", "dv"), g, g0):
            rel = (x - y).abs().max() / y.abs().max()
            print(f"{tag}-vs-v2 {name}: max rel diff {rel:.2e}")
            assert rel < 1e-5, (tag, name)


if __name__ == "__main__":
    main()
