#!/usr/bin/env python3
"""Decode-GEMM shape race: hipBLASLt (TunableOp) vs packed-LDS (v2) vs
packed-xreg (v3) on the 70B/8B decode shapes at M=64/128/256.

Run on a GPU box:
  python tools/gemm_decode_bench.py [--check-only]
"""
import argparse
import os
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch

from xotorch_amd import ops
from xotorch_amd.ops import _load_hip

SHAPES = [
  # (name, N, K)  — 70B decode shapes + lm_head; 8B qkv/gate_up for reference
  ("qkv70",     10240, 8192),
  ("o70",        8192, 8192),
  ("gate_up70", 57344, 8192),
  ("down70",     8192, 28672),
  ("lm_head",  128256, 8192),
  ("qkv8",       6144, 4096),
  ("gate_up8",  28672, 4096),
]


def time_us(fn, reps=20, rounds=5):
  fn()
  torch.cuda.synchronize()
  best = float("inf")
  for _ in range(rounds):
    s = torch.cuda.Event(enable_timing=True)
    e = torch.cuda.Event(enable_timing=True)
    s.record()
    for _ in range(reps):
      fn()
    e.record()
    e.synchronize()
    best = min(best, s.elapsed_time(e) * 1000.0 / reps)
  return best


def main():
  p = argparse.ArgumentParser()
  p.add_argument("--check-only", action="store_true")
  p.add_argument("--ms", type=str, default="64,128,256")
  args = p.parse_args()
  hip = _load_hip()
  assert hip is not None, "HIP extension missing"
  dev = "cuda"
  torch.manual_seed(7)
  print(f"{'shape':<10} {'M':>4} {'blaslt':>9} {'packed':>9} {'pk-bk64':>9} {'xreg':>9}  best (TB/s of weight stream)")
  for name, N, K in SHAPES:
    w = (torch.randn(N, K, device=dev) / K**0.5).to(torch.bfloat16)
    wp = ops.pack_decode_weight(w)
    wbytes = N * K * 2
    for M in [int(m) for m in args.ms.split(",")]:
      x = (torch.randn(M, K, device=dev) / 4).to(torch.bfloat16)
      ref = torch.nn.functional.linear(x, w)
      yp = hip.skinny_gemm_packed(x, wp, N, None)
      yx = hip.skinny_gemm_packed_xreg(x, wp, N, None)
      for tag, y in (("packed", yp), ("xreg", yx)):
        err = (y.float() - ref.float()).abs().max().item()
        scale = ref.float().abs().max().item()
        assert err < 0.02 * max(scale, 1.0) + 0.05, f"{name} M={M} {tag}: err {err} scale {scale}"
      if args.check_only:
        print(f"{name:<10} {M:>4} ok")
        continue
      t_bl = time_us(lambda: torch.nn.functional.linear(x, w))
      t_pk = time_us(lambda: hip.skinny_gemm_packed(x, wp, N, None))
      os.environ["XOT_SKINNY_BK64"] = "1"
      t_p64 = time_us(lambda: hip.skinny_gemm_packed(x, wp, N, None))
      del os.environ["XOT_SKINNY_BK64"]
      t_xr = time_us(lambda: hip.skinny_gemm_packed_xreg(x, wp, N, None))
      best = min(t_bl, t_pk, t_p64, t_xr)
      tbps = wbytes / best / 1e6
      win = {t_bl: "blaslt", t_pk: "packed", t_p64: "pk-bk64", t_xr: "xreg"}[best]
      print(f"{name:<10} {M:>4} {t_bl:>9.1f} {t_pk:>9.1f} {t_p64:>9.1f} {t_xr:>9.1f}  {win} {tbps:.2f} TB/s")
    del w, wp
    torch.cuda.empty_cache()


if __name__ == "__main__":
  main()
