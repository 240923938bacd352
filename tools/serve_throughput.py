#!/usr/bin/env python3
"""Serving-path throughput: N requests through the slot ring (graphs on),
aggregate tok/s + TTFT distribution."""
import argparse
import queue
import statistics
import sys
import threading
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))
import numpy as np
import torch

from xotorch_amd.serve_ring import AdmitMsg, RingSlotWorker


def main():
  p = argparse.ArgumentParser()
  p.add_argument("--model", default="llama-3-8b")
  p.add_argument("--slots", type=int, default=8)
  p.add_argument("--requests", type=int, default=32)
  p.add_argument("--max-new", type=int, default=64)
  p.add_argument("--prompt-len", type=int, default=128)
  p.add_argument("--no-graphs", action="store_true")
  args = p.parse_args()
  dev = "cuda" if torch.cuda.is_available() else "cpu"
  dt = torch.bfloat16 if dev == "cuda" else torch.float32
  max_seq = max(2048, args.prompt_len + args.max_new + 2)
  w = RingSlotWorker(args.model, 0, 1, device=dev, dtype=dt,
                     slots=args.slots, max_seq=max_seq, use_graphs=not args.no_graphs)
  if dev == "cuda" and not args.no_graphs:
    w._build_graph()
  rng = np.random.default_rng(3)
  q = queue.Queue()
  got, ttfts = {}, {}
  remaining = set()
  done = threading.Event()

  gaps = []
  last_emit = {}

  def emit(rid, tok, fin, meta):
    now = time.perf_counter()
    if rid in last_emit:
      gaps.append(now - last_emit[rid])
    last_emit[rid] = now
    got.setdefault(rid, []).append(tok)
    if "ttft_s" in meta:
      ttfts[rid] = meta["ttft_s"]
    if fin:
      remaining.discard(rid)
      if not remaining:
        done.set()

  for i in range(args.requests):
    rid = f"r{i}"
    remaining.add(rid)
    ids = [int(v) for v in rng.integers(0, 32000, args.prompt_len)]
    q.put(AdmitMsg(rid, torch.tensor([ids], dtype=torch.int64), args.max_new, 0.0))
  t = threading.Thread(target=w.serve_forever, args=(q, emit), daemon=True)
  t0 = time.perf_counter()
  t.start()
  ok = done.wait(900)
  dt_all = time.perf_counter() - t0
  q.put(AdmitMsg("stop", None, 0, 0.0))
  t.join(timeout=60)
  assert ok, f"unfinished: {remaining}"
  total = sum(len(v) for v in got.values())
  tl = sorted(ttfts.values())
  print(f"serve throughput {args.model}: {args.requests} reqs x {args.max_new} tok, "
        f"{args.slots} slots, graphs={w._graph is not None}: "
        f"{total} tokens in {dt_all:.2f}s = {total/dt_all:.0f} tok/s aggregate; "
        f"TTFT p50 {statistics.median(tl)*1000:.0f} ms / p95 {tl[int(len(tl)*0.95)-1]*1000:.0f} ms")
  if gaps:
    gs = sorted(gaps)
    print(f"inter-token gap: p50 {statistics.median(gs)*1000:.1f} ms / "
          f"p95 {gs[int(len(gs)*0.95)-1]*1000:.1f} ms / max {gs[-1]*1000:.1f} ms")


if __name__ == "__main__":
  main()
