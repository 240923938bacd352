"""PMC microbench target: the headline decode-attention shape (70B B=128)."""
import sys
from pathlib import Path
sys.path.insert(0, str(Path(__file__).resolve().parent.parent))
import torch
from xotorch_amd.ops import _hip_ops as hip

B, H, KVH, hd, T, sl = 128, 64, 8, 128, 672, 650
t32 = T
q = (torch.randn(B, 1, H, hd, device="cuda") * 0.2).to(torch.bfloat16)
kp = torch.randn(B, KVH, t32 // 16, 4, 64, 8, device="cuda").to(torch.bfloat16)
vp = torch.randn(B, KVH, 8, t32 // 32, 64, 8, device="cuda").to(torch.bfloat16)
sl_t = torch.full((B,), sl, dtype=torch.int32, device="cuda")
for _ in range(40):
  hip.attn_decode_mfma(q, kp, vp, sl_t, T)
torch.cuda.synchronize()
import time
t0 = time.perf_counter()
for _ in range(40):
  hip.attn_decode_mfma(q, kp, vp, sl_t, T)
torch.cuda.synchronize()
dt = (time.perf_counter() - t0) / 40
bytes_ = B * KVH * sl * hd * 2 * 2  # K + V packed streams
print(f"attn_decode_mfma B={B} sl={sl}: {dt*1e6:.1f} us, {bytes_/dt/1e12:.2f} TB/s")
