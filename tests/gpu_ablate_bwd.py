import os, sys, time, torch
sys.path.insert(0, ".")
from magi_attention.functional import flex_flash_attn_func
from magi_attention.functional.flex_flash_attn import _flex_flash_attn_backward

def run(n=8192, hq=32, hk=32, d=128, steps=5):
    q=(torch.randn(n,hq,d)*0.5).bfloat16().cuda()
    k=(torch.randn(n,hk,d)*0.5).bfloat16().cuda()
    v=(torch.randn(n,hk,d)*0.5).bfloat16().cuda()
    qr=torch.tensor([[0,n]],dtype=torch.int32,device="cuda"); kr=qr.clone()
    tm=torch.tensor([1],dtype=torch.int32,device="cuda")
    do=torch.randn_like(q)
    with torch.no_grad():
        out, meta = flex_flash_attn_func(q,k,v,qr,kr,tm,max_seqlen_q=n)
    def bwd():
        _flex_flash_attn_backward(
            dout=do, q=q, k=k, v=v, sink=None, sink_layout="sh", out=out,
            lse=meta.lse, dq=None, dk=None, dv=None, dsink=None,
            q_ranges=qr, k_ranges=kr, attn_type_map=tm,
            softmax_scale=d**-0.5, softcap=0.0, dq_type=None, dk_type=None,
            dv_type=None, disable_bwd_dkv_atomic_reduction=False,
            deterministic=False, sm_margin=0, max_seqlen_k=n)
    for _ in range(2): bwd()
    torch.cuda.synchronize(); t0=time.perf_counter()
    for _ in range(steps): bwd()
    torch.cuda.synchronize()
    return (time.perf_counter()-t0)/steps*1e3

import ctypes
for mode, name in [(0,"full"),(1,"no_dq_atomics"),(2,"no_dkv_stores"),(3,"no_stores")]:
    os.environ["MAGI_BWD_ABLATE"]=str(mode)
    print(f"{name}: {run():.2f} ms")
