import sys
from pathlib import Path
sys.path.insert(0, str(Path(__file__).resolve().parent.parent))
import torch
from xotorch_amd.ops import _hip_ops as hip
from xotorch_amd.ops import pack_decode_weight
for M in (64, 128):
    a = torch.randn(M, 28672, device="cuda").to(torch.bfloat16)
    w = (torch.randn(8192, 28672, device="cuda") * 0.02).to(torch.bfloat16)
    wp = pack_decode_weight(w)
    for _ in range(30):
        hip.skinny_gemm_packed(a, wp, 8192, None)
    torch.cuda.synchronize()
