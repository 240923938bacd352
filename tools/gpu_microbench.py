"""Microbenchmarks for the custom kernels at decode shapes (B=64, 70B dims)."""
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))
import torch  # noqa: E402

from xotorch_amd.ops import _hip_ops as hip  # noqa: E402


def timeit(fn, iters=200):
  for _ in range(10):
    fn()
  torch.cuda.synchronize()
  start = torch.cuda.Event(enable_timing=True)
  end = torch.cuda.Event(enable_timing=True)
  start.record()
  for _ in range(iters):
    fn()
  end.record()
  torch.cuda.synchronize()
  return start.elapsed_time(end) / iters * 1000  # us


def main():
  B, D, I, H, KVH, hd, T = 64, 8192, 28672, 64, 8, 128, 640
  gu = torch.randn(B, 1, 2 * I, device="cuda").to(torch.bfloat16)
  x = torch.randn(B, 1, D, device="cuda").to(torch.bfloat16)
  r = torch.randn_like(x)
  w = torch.randn(D, device="cuda").to(torch.bfloat16)
  us = timeit(lambda: hip.swiglu_packed(gu))
  bytes_io = gu.numel() * 2 * 1.5
  print(f"swiglu_packed [{B},2x{I}]: {us:.1f} us  {bytes_io/us/1e3:.2f} TB/s")
  us = timeit(lambda: (torch.nn.functional.silu(gu[..., :I]) * gu[..., I:]))
  print(f"torch silu*mul          : {us:.1f} us")
  us = timeit(lambda: hip.rmsnorm(x, w, 1e-5))
  print(f"rmsnorm [{B},{D}]       : {us:.1f} us  {x.numel()*2*2/us/1e3:.2f} TB/s")
  us = timeit(lambda: hip.rmsnorm_residual(x, r, w, 1e-5))
  print(f"rmsnorm_residual        : {us:.1f} us")
  qkv = torch.randn(B, 1, (H + 2 * KVH) * hd, device="cuda").to(torch.bfloat16)
  cos = torch.randn(T, hd // 2, device="cuda")
  sin = torch.randn(T, hd // 2, device="cuda")
  kc = torch.zeros(B, KVH, T, hd, dtype=torch.bfloat16, device="cuda")
  vc = torch.zeros_like(kc)
  pos = torch.full((1,), 500, dtype=torch.int32, device="cuda")
  us = timeit(lambda: hip.rope_qkv_append(qkv, cos, sin, pos, kc, vc, H, KVH, hd))
  print(f"rope_qkv_append         : {us:.1f} us")
  q = qkv[:, :, : H * hd].view(B, 1, H, hd)
  sl = torch.full((B,), 576, dtype=torch.int32, device="cuda")
  us = timeit(lambda: hip.attn_decode(q, kc, vc, sl))
  kv_bytes = B * KVH * 576 * hd * 2 * 2
  print(f"attn_decode B{B} sl576  : {us:.1f} us  {kv_bytes/us/1e3:.2f} TB/s")
  # GEMM A/B: hipBLASLt via torch vs the skinny weight-streaming MFMA kernel
  for M in (64, 128, 256):
    print(f"-- decode GEMMs at M={M} (70B dims) --")
    for (m, k, n, tag) in ((M, D, (H + 2 * KVH) * hd, "qkv"), (M, D, 2 * I, "gate_up"),
                           (M, I, D, "down"), (M, H * hd, D, "o"), (M, D, 128256, "lm_head")):
      a = torch.randn(m, k, device="cuda").to(torch.bfloat16)
      wt = torch.randn(n, k, device="cuda").to(torch.bfloat16) * 0.02
      us = timeit(lambda: torch.nn.functional.linear(a, wt), iters=50)
      wbytes = n * k * 2
      us2 = timeit(lambda: hip.skinny_gemm(a, wt, None), iters=50)
      from xotorch_amd.ops import pack_decode_weight
      wp = pack_decode_weight(wt)
      us3 = timeit(lambda: hip.skinny_gemm_packed(a, wp, n, None), iters=50)
      ref = torch.nn.functional.linear(a.float(), wt.float())
      got = hip.skinny_gemm(a, wt, None).float()
      err = (got - ref).abs().max().item() / (ref.abs().max().item() + 1e-9)
      gotp = hip.skinny_gemm_packed(a, wp, n, None).float()
      errp = (gotp - ref).abs().max().item() / (ref.abs().max().item() + 1e-9)
      del wp
      print(f"linear {tag:8s} [{m},{k}]x[{k},{n}]: blaslt {us:7.1f} us {wbytes/us/1e3:5.2f} TB/s"
            f" | skinny {us2:7.1f} us {wbytes/us2/1e3:5.2f} TB/s e={err:.1e}"
            f" | packed {us3:7.1f} us {wbytes/us3/1e3:5.2f} TB/s e={errp:.1e}")


def attn_mfma_bench():
  import sys
  sys.path.insert(0, ".")
  from tests.test_kernels_gpu import _pack_k, _pack_v
  from xotorch_amd.ops import torch_ref
  for (B, H, KVH, T, sl) in ((64, 64, 8, 640, 576), (256, 32, 8, 640, 576)):
    hd = 128
    q = torch.randn(B, 1, H, hd, device="cuda").to(torch.bfloat16)
    k = torch.randn(B, KVH, T, hd, device="cuda").to(torch.bfloat16)
    v = torch.randn(B, KVH, T, hd, device="cuda").to(torch.bfloat16)
    t32 = (T + 31) // 32 * 32
    kp, vp = _pack_k(k, t32), _pack_v(v, t32)
    sl_t = torch.full((B,), sl, dtype=torch.int32, device="cuda")
    us_old = timeit(lambda: hip.attn_decode(q, k, v, sl_t))
    us_new = timeit(lambda: hip.attn_decode_mfma(q, kp, vp, sl_t, T))
    kv_bytes = B * KVH * sl * hd * 2 * 2
    err = (hip.attn_decode_mfma(q, kp, vp, sl_t, T).float() - hip.attn_decode(q, k, v, sl_t).float()).abs().max().item()
    print(f"attn B{B} H{H} sl{sl}: valu {us_old:6.1f} us {kv_bytes/us_old/1e3:5.2f} TB/s"
          f" | mfma {us_new:6.1f} us {kv_bytes/us_new/1e3:5.2f} TB/s  maxdiff {err:.3e}")


if __name__ == "__main__":
  main()
  attn_mfma_bench()
  prefill_attn_bench()


def prefill_attn_bench():
  import sys
  from pathlib import Path
  sys.path.insert(0, str(Path(__file__).resolve().parent.parent))
  from tests.test_kernels_gpu import _pack_k, _pack_v
  import torch.nn.functional as F
  B, S, H, KVH, hd = 64, 512, 64, 8, 128
  t32 = (S + 31) // 32 * 32
  q = torch.randn(B, S, H, hd, device="cuda").to(torch.bfloat16)
  k = torch.randn(B, KVH, S, hd, device="cuda").to(torch.bfloat16)
  v = torch.randn(B, KVH, S, hd, device="cuda").to(torch.bfloat16)
  kp, vp = _pack_k(k, t32), _pack_v(v, t32)
  def sdpa():
    return F.scaled_dot_product_attention(q.transpose(1, 2), k, v, is_causal=True,
                                          enable_gqa=True).transpose(1, 2).contiguous()
  us_sdpa = timeit(sdpa, iters=20)
  from xotorch_amd.ops import _hip_ops
  us_mfma = timeit(lambda: _hip_ops.attn_prefill_mfma(q, kp, vp, 0), iters=20)
  flops = B * H * S * (S / 2) * 4 * hd
  print(f"prefill attn B{B} S{S}: sdpa {us_sdpa:7.1f} us {flops/us_sdpa/1e6:6.0f} TF"
        f" | mfma {us_mfma:7.1f} us {flops/us_mfma/1e6:6.0f} TF")
