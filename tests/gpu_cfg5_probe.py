import sys, time, torch
sys.path.insert(0, ".")
from magi_attention.functional import flex_flash_attn_func
T, HQ, HKV, D = 131072, 64, 8, 128
q8 = (torch.randn(T, HQ, D)*0.3).bfloat16().to(torch.float8_e4m3fn).cuda()
k8 = (torch.randn(T, HKV, D)*0.3).bfloat16().to(torch.float8_e4m3fn).cuda()
v8 = (torch.randn(T, HKV, D)*0.3).bfloat16().to(torch.float8_e4m3fn).cuda()
qr = torch.tensor([[0, T]], dtype=torch.int32, device="cuda")
tm = torch.tensor([1], dtype=torch.int32, device="cuda")
area = T*(T+1)//2
def f():
    with torch.no_grad():
        flex_flash_attn_func(q8, k8, v8, qr, qr, tm, max_seqlen_q=T, max_seqlen_k=T)
for _ in range(2): f()
torch.cuda.synchronize(); t0=time.perf_counter()
for _ in range(4): f()
torch.cuda.synchronize(); dt=(time.perf_counter()-t0)/4
print(f"cfg5 fp8 fwd 128k GQA64/8: {dt*1e3:.1f} ms  {4*area*HQ*D/dt/1e12:.1f} TF")
# deterministic overhead at 64k bf16 dense
Tb = 65536
q = (torch.randn(Tb, 32, D)*0.5).bfloat16().cuda().requires_grad_(True)
k = (torch.randn(Tb, 32, D)*0.5).bfloat16().cuda().requires_grad_(True)
v = (torch.randn(Tb, 32, D)*0.5).bfloat16().cuda().requires_grad_(True)
qrb = torch.tensor([[0, Tb]], dtype=torch.int32, device="cuda")
do = torch.randn_like(q)
areab = Tb*(Tb+1)//2
for det in (False, True):
    def g(det=det):
        out, _ = flex_flash_attn_func(q, k, v, qrb, qrb, tm, deterministic=det,
                                      max_seqlen_q=Tb, max_seqlen_k=Tb)
        out.backward(do); q.grad=k.grad=v.grad=None
    for _ in range(2): g()
    torch.cuda.synchronize(); t0=time.perf_counter()
    for _ in range(4): g()
    torch.cuda.synchronize(); dt=(time.perf_counter()-t0)/4
    print(f"64k dense fwd+bwd det={det}: {dt*1e3:.1f} ms  {4*areab*32*D*3.5/dt/1e12:.1f} TF")
