import ctypes, sys
import torch
sys.path.insert(0, ".")
from magi_attention import _ffa_lib
lib = _ffa_lib.lib()
lib.magi_probe_tr16.argtypes = [ctypes.c_void_p, ctypes.c_void_p, ctypes.c_int, ctypes.c_void_p]
lib.magi_probe_tr16.restype = ctypes.c_int
inp = torch.arange(2048, dtype=torch.int16).cuda()
for mode in (0, 1, 2):
    out = torch.zeros(256, dtype=torch.int16).cuda()
    rc = lib.magi_probe_tr16(ctypes.c_void_p(inp.data_ptr()), ctypes.c_void_p(out.data_ptr()),
                             mode, _ffa_lib.current_stream_ptr())
    torch.cuda.synchronize()
    o = out.cpu().to(torch.int32).tolist()
    print(f"mode {mode}: rc={rc}")
    for l in range(0, 64, 4):
        print("  ", [o[x*4:(x+1)*4] for x in range(l, l+4)])
