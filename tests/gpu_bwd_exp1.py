"""r2 experiment 1: solo backward-pass timings at 64k dense causal across
kernel variants (split NBUF2 = r1 baseline, split NBUF3 ring, fused-dkv r2),
plus fwd 8k/64k reference points. Prints ms per launch + effective TFLOPS of
the ISSUED matmul work (dq pass issues 3/5 of bwd FLOPs, dv 2/5... wait:
dq = S,dP,dQ = 3 units; dv = S,dV = 2; dk = S,dP,dK = 3; fused = S,dP,dV,dK = 4;
one 'unit' = fwd_flops/2)."""
import os, sys, time
import torch
sys.path.insert(0, ".")
from magi_attention import _ffa_lib
from magi_attention._ffa_lib import MagiFfaBwdArgs, MagiFfaFwdArgs, check, ptr
import ctypes


def stream_ptr():
    return ctypes.c_void_p(torch.cuda.current_stream().cuda_stream)


def make_case(n=65536, hq=32, hk=32, d=128):
    torch.manual_seed(7)
    q = (torch.randn(n, hq, d) * 0.5).bfloat16().cuda()
    k = (torch.randn(n, hk, d) * 0.5).bfloat16().cuda()
    v = (torch.randn(n, hk, d) * 0.5).bfloat16().cuda()
    do = torch.randn_like(q)
    qr = torch.tensor([[0, n]], dtype=torch.int32, device="cuda")
    tm = torch.tensor([1], dtype=torch.int32, device="cuda")
    from magi_attention.functional import flex_flash_attn_func
    with torch.no_grad():
        out, meta = flex_flash_attn_func(q, k, v, qr, qr.clone(), tm,
                                         max_seqlen_q=n)
    return q, k, v, do, out, meta.lse, qr, tm, n, hq, hk, d


def bwd_args(case, dq, dk, dv, dpsum):
    q, k, v, do, out, lse, qr, tm, n, hq, hk, d = case
    return MagiFfaBwdArgs(
        dout=ptr(do), q=ptr(q), k=ptr(k), v=ptr(v), out=ptr(out),
        lse=ptr(lse), dq=ptr(dq), dk=ptr(dk), dv=ptr(dv), dpsum=ptr(dpsum),
        q_ranges=ptr(qr), k_ranges=ptr(qr), attn_type_map=ptr(tm),
        n_ranges=1, total_q=n, total_k=n, hq=hq, hk=hk, d=d,
        max_seqlen_k=n, out_is_fp32=0, softmax_scale=d ** -0.5,
        softcap=0.0, cu_margin=0, stream=stream_ptr(),
    )


def time_pass(fn, steps=3, warm=1):
    for _ in range(warm):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(steps):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / steps * 1e3


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
    unit = 2 * area * hq * d  # one matmul's FLOPs
    results = {}

    def run(tag, entry, units, env=None):
        env = env or {}
        for kk, vv in env.items():
            os.environ[kk] = vv
        ms = time_pass(lambda: check(getattr(lib, entry)(a), entry))
        for kk in env:
            os.environ.pop(kk, None)
        tf = units * unit / (ms / 1e3) / 1e12
        eff = tf / 2500 * 100
        results[tag] = (ms, tf)
        print(f"{tag:28s} {ms:8.2f} ms   issued {tf:7.1f} TF/s ({eff:4.1f}% peak)")

    print(f"== 64k causal h{hq} d{d}: one matmul unit = {unit/1e12:.2f} TFLOP ==")
    run("dq64 (1 wave/SIMD)", "magi_ffa_bwd_dq", 3)
    run("dq32 (2 waves)", "magi_ffa_bwd_dq", 3, {"MAGI_BWD_DQ64": "0"})
    run("dv  W8", "magi_ffa_bwd_dv", 2)
    run("dk", "magi_ffa_bwd_dk", 3)
    run("dkv fused", "magi_ffa_bwd_dkv", 4)

    # best-combo whole-backward estimate
    best_dq = min(results["dq64 (1 wave/SIMD)"][0], results["dq32 (2 waves)"][0])
    split = results["dv  W8"][0] + results["dk"][0]
    fused = results["dkv fused"][0]
    bwd_flops = 2.5 * 2 * unit
    for tag, tot in (("split best", best_dq + split),
                     ("fused best", best_dq + fused)):
        print(f"bwd serialized {tag}: {tot:.1f} ms -> frac "
              f"{bwd_flops/(tot/1e3)/2.5e15:.3f} of MFMA peak")

    # fused parity vs split (both fp32-atomic; order differs)
    def grads(env):
        for kk, vv in env.items():
            os.environ[kk] = vv
        dq.zero_(); dk.zero_(); dv.zero_()
        from magi_attention.functional.flex_flash_attn import run_bwd_passes
        a.stream = stream_ptr()
        run_bwd_passes(a, torch.device("cuda"))
        torch.cuda.synchronize()
        for kk in env:
            os.environ.pop(kk, None)
        return dq.clone(), dk.clone(), dv.clone()

    g1 = grads({"MAGI_BWD_FUSED_DKV": "1"})
    g2 = grads({})
    for name, x, y in zip("dq dk dv".split(), g1, g2):
        rel = (x - y).abs().max() / y.abs().max()
        print(f"fused-vs-split {name}: max rel diff {rel:.2e}")

    # fwd reference points
    from magi_attention.functional import flex_flash_attn_func
    for nn in (8192, 65536):
        qq = q[:nn].contiguous(); kk2 = k[:nn].contiguous(); vv2 = v[:nn].contiguous()
        qrr = torch.tensor([[0, nn]], dtype=torch.int32, device="cuda")
        def fwd():
            with torch.no_grad():
                flex_flash_attn_func(qq, kk2, vv2, qrr, qrr.clone(), tm,
                                     max_seqlen_q=nn)
        ms = time_pass(fwd, steps=5, warm=2)
        fl = 4 * (nn * (nn + 1) // 2) * hq * d
        print(f"fwd {nn//1024}k: {ms:.2f} ms  {fl/(ms/1e3)/1e12:.1f} TF/s "
              f"({fl/(ms/1e3)/2.5e15*100:.1f}% peak)")


if __name__ == "__main__":
    main()
