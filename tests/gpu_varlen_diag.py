"""Varlen (config 2) per-kernel diagnosis: 8x2k causal docs, h16 d128."""
import sys, time, ctypes
import torch
sys.path.insert(0, ".")
from magi_attention.functional import flex_flash_attn_func
from magi_attention.functional.flex_flash_attn import run_bwd_passes
from magi_attention import _ffa_lib
from magi_attention._ffa_lib import MagiFfaBwdArgs, check, ptr

n, hq, hk, d = 16384, 16, 16, 128
torch.manual_seed(7)
q = (torch.randn(n, hq, d)*0.5).bfloat16().cuda()
k = (torch.randn(n, hk, d)*0.5).bfloat16().cuda()
v = (torch.randn(n, hk, d)*0.5).bfloat16().cuda()
rs = [[i*2048, (i+1)*2048] for i in range(8)]
qr = torch.tensor(rs, dtype=torch.int32, device="cuda")
tm = torch.tensor([1]*8, dtype=torch.int32, device="cuda")
do = torch.randn_like(q)
area = 8*(2048*2049//2)
unit = 2*area*hq*d

with torch.no_grad():
    out, meta = flex_flash_attn_func(q,k,v,qr,qr.clone(),tm,max_seqlen_q=2048,max_seqlen_k=2048)
torch.cuda.synchronize()

def t(fn, steps=20, warm=5):
    for _ in range(warm): fn()
    torch.cuda.synchronize(); t0=time.perf_counter()
    for _ in range(steps): fn()
    torch.cuda.synchronize()
    return (time.perf_counter()-t0)/steps*1e3

def fwd():
    with torch.no_grad():
        flex_flash_attn_func(q,k,v,qr,qr.clone(),tm,max_seqlen_q=2048,max_seqlen_k=2048)
ms = t(fwd)
print(f"fwd: {ms:.3f} ms  {2*unit/(ms/1e3)/1e12:.1f} TF ({2*unit/(ms/1e3)/2.5e15*100:.1f}%)")

dq = torch.zeros(n, hq, d, dtype=torch.float32, device="cuda")
dk = torch.zeros(n, hk, d, dtype=torch.float32, device="cuda")
dv = torch.zeros(n, hk, d, dtype=torch.float32, device="cuda")
dpsum = torch.empty(n, hq, dtype=torch.float32, device="cuda")
def stream_ptr():
    return ctypes.c_void_p(torch.cuda.current_stream().cuda_stream)
a = MagiFfaBwdArgs(
    dout=ptr(do), q=ptr(q), k=ptr(k), v=ptr(v), out=ptr(out.float()),
    lse=ptr(meta.lse), dq=ptr(dq), dk=ptr(dk), dv=ptr(dv), dpsum=ptr(dpsum),
    q_ranges=ptr(qr), k_ranges=ptr(qr), attn_type_map=ptr(tm),
    n_ranges=8, total_q=n, total_k=n, hq=hq, hk=hk, d=d,
    max_seqlen_k=2048, out_is_fp32=1, softmax_scale=d**-0.5,
    softcap=0.0, cu_margin=0, stream=stream_ptr())
lib = _ffa_lib.lib()
check(lib.magi_ffa_bwd_preprocess(a), "pre")
import os
for name, entry, units in [("dq", "magi_ffa_bwd_dq", 3), ("dv", "magi_ffa_bwd_dv", 2),
                           ("dk", "magi_ffa_bwd_dk", 3), ("fused", "magi_ffa_bwd_dkv", 4)]:
    def run():
        a.stream = stream_ptr()
        check(getattr(lib, entry)(a), entry)
    ms = t(run)
    tf = units*unit/(ms/1e3)/1e12
    print(f"{name}: {ms:.3f} ms  issued {tf:.1f} TF ({tf/2500*100:.1f}%)")
# W8 variants for comparison despite short ranges (launcher picks W4 <8192)
