import sys, itertools
import torch
sys.path.insert(0, ".")
from oracle import make_attn_mask, ref_attn
from tests.util import make_flex_case
from magi_attention.functional import flex_flash_attn_func

total = 3072
QR = [[0, 1024], [512, 2048], [1536, 3072], [2048, 3072]]
KR = [[0, 1024], [0, 2048], [1024, 2560], [0, 512]]
TY = [1, 1, 1, 0]

def run(idx, tag):
    qr_l = [QR[i] for i in idx]; kr_l = [KR[i] for i in idx]; ty_l = [TY[i] for i in idx]
    q, k, v, dout, qr, kr, tm = make_flex_case(total, total, 8, 4, 128, qr_l, kr_l, ty_l, seed=5)
    out, meta = flex_flash_attn_func(q, k, v, qr, kr, tm)
    torch.cuda.synchronize()
    mask = make_attn_mask(total, total, qr_l, kr_l, ty_l)
    o_hi, lse_hi = ref_attn(q.cpu(), k.cpu(), v.cpu(), mask)
    err = (out.cpu().double() - o_hi.double()).norm() / o_hi.double().norm().clamp_min(1e-9)
    # worst rows
    rowerr = (out.cpu().double() - o_hi.double()).norm(dim=(1,2))
    bad = rowerr.argsort(descending=True)[:5]
    print(f"{tag}: rel {err:.3e}  worst rows {bad.tolist()} errs {[f'{rowerr[b]:.2e}' for b in bad]}")

for i in range(4):
    run([i], f"slice{i}")
for pair in itertools.combinations(range(4), 2):
    run(list(pair), f"pair{pair}")
run([0,1,2,3], "all")
