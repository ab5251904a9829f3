"""Quick kernel-level perf probe (not the contract bench). Times FFA fwd and
fwd+bwd on BASELINE-like shapes, prints TFLOPS (convention: fwd=4*area*hq*hd,
bwd=2.5x fwd)."""
import sys, time
import torch
sys.path.insert(0, ".")
from magi_attention.functional import flex_flash_attn_func

def area_causal(n): return n*(n+1)//2

def bench(name, tq, tk, hq, hk, d, q_ranges, k_ranges, types, area, steps=10, warm=3, bwd=True):
    q = (torch.randn(tq, hq, d)*0.5).bfloat16().cuda().requires_grad_(bwd)
    k = (torch.randn(tk, hk, d)*0.5).bfloat16().cuda().requires_grad_(bwd)
    v = (torch.randn(tk, hk, d)*0.5).bfloat16().cuda().requires_grad_(bwd)
    qr = torch.tensor(q_ranges, dtype=torch.int32, device="cuda")
    kr = torch.tensor(k_ranges, dtype=torch.int32, device="cuda")
    tm = torch.tensor(types, dtype=torch.int32, device="cuda")
    do = torch.randn_like(q)
    ms = max(r[1]-r[0] for r in q_ranges)
    msk = max(r[1]-r[0] for r in k_ranges)
    def fwd_only():
        with torch.no_grad():
            flex_flash_attn_func(q, k, v, qr, kr, tm, max_seqlen_q=ms, max_seqlen_k=msk)
    def fwd_bwd():
        out, _ = flex_flash_attn_func(q, k, v, qr, kr, tm, max_seqlen_q=ms, max_seqlen_k=msk)
        out.backward(do)
        q.grad = k.grad = v.grad = None
    for fn, tag, mult in ((fwd_only, "fwd", 1.0),) + (((fwd_bwd, "fwd+bwd", 3.5),) if bwd else ()):
        for _ in range(warm): fn()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(steps): fn()
        torch.cuda.synchronize()
        dt = (time.perf_counter()-t0)/steps
        fl = 4*area*hq*d*mult
        print(f"{name} {tag}: {dt*1e3:.2f} ms  {fl/dt/1e12:.1f} TFLOPS")

# dense causal 8k (north_star roofline target shape)
bench("dense8k_h32_d128", 8192, 8192, 32, 32, 128, [[0,8192]], [[0,8192]], [1], area_causal(8192))
# varlen 16k: 8 causal docs of 2k, hq16 d128 (BASELINE config 2)
rs = [[i*2048, (i+1)*2048] for i in range(8)]
bench("varlen16k_h16_d128", 16384, 16384, 16, 16, 128, rs, rs, [1]*8, 8*area_causal(2048))
# dense causal 64k h32 d128 (config 3 whole-job at cp=1)
bench("dense64k_h32_d128", 65536, 65536, 32, 32, 128, [[0,65536]], [[0,65536]], [1], area_causal(65536), steps=3, warm=1)

# fp8 extension (fwd only)
def bench_fp8(name, n, hq, hk, d, steps=10):
    q=(torch.randn(n,hq,d)*0.5).to(torch.float8_e4m3fn).cuda()
    k=(torch.randn(n,hk,d)*0.5).to(torch.float8_e4m3fn).cuda()
    v=(torch.randn(n,hk,d)*0.5).to(torch.float8_e4m3fn).cuda()
    qr=torch.tensor([[0,n]],dtype=torch.int32,device="cuda"); kr=qr.clone()
    tm=torch.tensor([1],dtype=torch.int32,device="cuda")
    with torch.no_grad():
        for _ in range(3): flex_flash_attn_func(q,k,v,qr,kr,tm,max_seqlen_q=n)
        torch.cuda.synchronize(); t0=time.perf_counter()
        for _ in range(steps): flex_flash_attn_func(q,k,v,qr,kr,tm,max_seqlen_q=n)
        torch.cuda.synchronize()
    dt=(time.perf_counter()-t0)/steps
    fl=4*area_causal(n)*hq*d
    print(f"{name} fp8-fwd: {dt*1e3:.2f} ms  {fl/dt/1e12:.1f} TFLOPS")

bench_fp8("dense8k_h32_d128", 8192, 32, 32, 128)
