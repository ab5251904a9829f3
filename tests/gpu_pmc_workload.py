"""Minimal workload for rocprofv3 PMC passes: fwd, dq, dv, dk, fused-dkv at
64k dense causal h32 d128, 2 launches each (keep the pass short)."""
import sys, ctypes
import torch
sys.path.insert(0, ".")
from magi_attention import _ffa_lib
from tests.gpu_bwd_exp1 import make_case, bwd_args, stream_ptr

lib = _ffa_lib.lib()
case = make_case()
q, k, v, do, out, lse, qr, tm, n, hq, hk, d = case
dq = torch.zeros(n, hq, d, dtype=torch.float32, device="cuda")
dk = torch.zeros(n, hk, d, dtype=torch.float32, device="cuda")
dv = torch.zeros(n, hk, d, dtype=torch.float32, device="cuda")
dpsum = torch.empty(n, hq, dtype=torch.float32, device="cuda")
a = bwd_args(case, dq, dk, dv, dpsum)
lib.magi_ffa_bwd_preprocess(a)
for _ in range(2):
    lib.magi_ffa_bwd_dq(a)
    lib.magi_ffa_bwd_dv(a)
    lib.magi_ffa_bwd_dk(a)
    lib.magi_ffa_bwd_dkv(a)
from magi_attention.functional import flex_flash_attn_func
with torch.no_grad():
    for _ in range(2):
        flex_flash_attn_func(q, k, v, qr, qr.clone(), tm, max_seqlen_q=n)
torch.cuda.synchronize()
print("PMC_WORKLOAD_DONE")
