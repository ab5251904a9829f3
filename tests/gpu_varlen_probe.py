import sys, time
import torch
sys.path.insert(0, ".")
from magi_attention.functional import flex_flash_attn_func

docs = 8
DL = 2048
T = docs * DL
hq, hk, d = 16, 16, 128
qr = [[i*DL, (i+1)*DL] for i in range(docs)]
tt = [1]*docs
area = docs * DL*(DL+1)//2
q = (torch.randn(T, hq, d)*0.5).bfloat16().cuda().requires_grad_(True)
k = (torch.randn(T, hk, d)*0.5).bfloat16().cuda().requires_grad_(True)
v = (torch.randn(T, hk, d)*0.5).bfloat16().cuda().requires_grad_(True)
qrt = torch.tensor(qr, dtype=torch.int32, device="cuda")
tmt = torch.tensor(tt, dtype=torch.int32, device="cuda")
do = torch.randn_like(q)

import ctypes
from magi_attention import _ffa_lib
from magi_attention._ffa_lib import MagiFfaBwdArgs, check, current_stream_ptr, ptr
lse = torch.randn(T, hq, device="cuda").float().abs() + 5
dps = torch.randn(T, hq, device="cuda").float()
out = torch.randn_like(q)
dq = torch.zeros(T, hq, d, device="cuda").float()
dk = torch.zeros(T, hk, d, device="cuda").float()
dv = torch.zeros(T, hk, d, device="cuda").float()
lib = _ffa_lib.lib()
args = MagiFfaBwdArgs(
    dout=ptr(do), q=ptr(q), k=ptr(k), v=ptr(v), out=ptr(out), lse=ptr(lse),
    dq=ptr(dq), dk=ptr(dk), dv=ptr(dv), dpsum=ptr(dps),
    q_ranges=ptr(qrt), k_ranges=ptr(qrt), attn_type_map=ptr(tmt),
    n_ranges=docs, total_q=T, total_k=T, hq=hq, hk=hk, d=d,
    max_seqlen_k=DL, out_is_fp32=0, softmax_scale=d**-0.5, softcap=0.0,
    cu_margin=0, stream=current_stream_ptr(),
)
BWD_UNIT = 2*area*hq*d*2  # one GEMM-equiv = 2*area*h*d flops; per kernel xN
for nm, fn, gemms in (("dq", lib.magi_ffa_bwd_dq, 3), ("dv", lib.magi_ffa_bwd_dv, 2),
                      ("dk", lib.magi_ffa_bwd_dk, 3), ("dkv", lib.magi_ffa_bwd_dkv, 4),
                      ("fwd(autograd)", None, 2)):
    if fn is None:
        def call():
            with torch.no_grad():
                flex_flash_attn_func(q, k, v, qrt, qrt, tmt, max_seqlen_q=DL, max_seqlen_k=DL)
    else:
        def call(fn=fn):
            check(fn(args), "x")
    for _ in range(5): call()
    torch.cuda.synchronize(); t0=time.perf_counter()
    for _ in range(20): call()
    torch.cuda.synchronize(); dt=(time.perf_counter()-t0)/20
    print(f"{nm:14s} {dt*1e3:7.3f} ms  {gemms*2*area*hq*d/dt/1e12:6.1f} TF-eff")
