#!/usr/bin/env python3
"""Speculative decoding demo/bench: target + draft (random-init — accept
rate is near zero without real weights; the tool reports mechanism overhead
and verifies the exact-output invariant on silicon)."""
import argparse
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))
import torch

from xotorch_amd.engine.spec import SpeculativeDecoder


def main():
  p = argparse.ArgumentParser()
  p.add_argument("--target", default="llama-3-8b")
  p.add_argument("--draft", default="llama-3.2-1b")
  p.add_argument("--gamma", type=int, default=4)
  p.add_argument("--max-new", type=int, default=64)
  p.add_argument("--dtype", default=None, choices=["bf16", "fp32"])
  args = p.parse_args()
  dev = "cuda" if torch.cuda.is_available() else "cpu"
  dt = torch.bfloat16 if dev == "cuda" else torch.float32
  if args.dtype:
    dt = torch.bfloat16 if args.dtype == "bf16" else torch.float32
  sd = SpeculativeDecoder.from_model_ids(args.target, args.draft, device=dev, dtype=dt,
                                         gamma=args.gamma)
  g = torch.Generator().manual_seed(5)
  prompt = torch.randint(0, 32000, (1, 64), generator=g)
  # warm + invariant check
  toks, stats = sd.generate(prompt, max_new=16)
  sd.reset()
  ref = sd.generate_plain(prompt, max_new=16)
  assert toks == ref, "spec output diverged from target-only greedy"
  sd.reset()
  t0 = time.perf_counter()
  toks, stats = sd.generate(prompt, max_new=args.max_new)
  if dev == "cuda":
    torch.cuda.synchronize()
  dt_spec = time.perf_counter() - t0
  sd.reset()
  t0 = time.perf_counter()
  ref = sd.generate_plain(prompt, max_new=args.max_new)
  if dev == "cuda":
    torch.cuda.synchronize()
  dt_plain = time.perf_counter() - t0
  assert toks == ref
  print(f"spec({args.target}+{args.draft} gamma={args.gamma}): "
        f"{len(toks)/dt_spec:.1f} tok/s, accept {stats.accept_rate:.2f} "
        f"({stats.accepted}/{stats.proposed} over {stats.rounds} rounds); "
        f"plain {len(ref)/dt_plain:.1f} tok/s; outputs identical "
        f"(random-init draft -> near-zero acceptance; real weights needed for speedup)")


if __name__ == "__main__":
  main()
