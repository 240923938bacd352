import sys, os
sys.path.insert(0, ".")
import torch
from xotorch_amd.ops import _hip_ops as hip
from xotorch_amd.ops import pack_decode_weight
from tools.gpu_microbench import timeit

tgt = os.getenv("XOT_SKINNY_TARGET", "1024")
M = int(sys.argv[1]) if len(sys.argv) > 1 else 128
res = []
for (k, n, tag) in ((8192, 10240, "qkv"), (8192, 57344, "gate_up"), (28672, 8192, "down"), (8192, 8192, "o")):
    a = torch.randn(M, k, device="cuda").to(torch.bfloat16)
    w = (torch.randn(n, k, device="cuda") * 0.02).to(torch.bfloat16)
    wp = pack_decode_weight(w)
    us = timeit(lambda: hip.skinny_gemm_packed(a, wp, n, None), iters=50)
    res.append(f"{tag} {us:6.1f}us {n*k*2/us/1e3:4.2f}TB/s")
    del a, w, wp
print(f"target={tgt} M={M}: " + " | ".join(res))
