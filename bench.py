#!/usr/bin/env python3
"""Driver benchmark: ring-pipeline decode tokens/sec (whole ring) + p50 TTFT.

Measures BASELINE.json's headline metric — Llama-3-70B sharded across N GPUs
of one node as ring pipeline stages (RCCL send/recv over xGMI), greedy decode
on synthetic prompts with random-init weights (no network for datasets or
checkpoints).

Launch (driver contract):
  python bench.py --gpus 1 --steps K --warmup W                 # single GPU
  python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
      --master-addr 127.0.0.1 bench.py --gpus N --steps K --warmup W

One "step" advances every in-flight micro-batch by one token; value =
whole-job decode tokens/sec = global_batch * steps / elapsed (max over ranks).
Scaling is weak: micro-batch count = N pipeline stages with fixed per-mb
batch, so per-GPU work (its layer shard's weight stream per step) is constant
as N grows.
"""
import argparse
import json
import os
import statistics
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent))

# hipBLASLt/rocBLAS algorithm selection pre-tuned on MI355X (gfx950) for the
# bench GEMM shapes; read-only (tuning off) so TTFT stays clean. Must be set
# before torch initializes.
_tuned = Path(__file__).resolve().parent / "profiles" / "tunableop_gfx950.csv"
if _tuned.exists() and os.getenv("XOT_TUNABLEOP", "1") == "1":
  # TunableOp inserts the device ordinal before ".csv" when resolving the
  # filename -- stage per-ordinal copies it will actually find.
  import shutil
  base = Path("/tmp/xot_tunableop.csv")
  for i in list(range(8)) + [""]:
    dst = Path(f"/tmp/xot_tunableop{i}.csv")
    if not dst.exists():
      shutil.copyfile(_tuned, dst)
  os.environ.setdefault("PYTORCH_TUNABLEOP_ENABLED", "1")
  os.environ.setdefault("PYTORCH_TUNABLEOP_TUNING", "0")
  os.environ.setdefault("PYTORCH_TUNABLEOP_FILENAME", str(base))

import torch  # noqa: E402


def main():
  p = argparse.ArgumentParser()
  p.add_argument("--gpus", type=int, default=1)
  p.add_argument("--steps", type=int, default=32)
  p.add_argument("--warmup", type=int, default=8)
  p.add_argument("--model", type=str, default="llama-3-70b")
  p.add_argument("--mb-batch", type=int, default=128, help="sequences per micro-batch (per pipeline slot)")
  p.add_argument("--prompt-len", type=int, default=512)
  p.add_argument("--no-graphs", action="store_true")
  p.add_argument("--device", type=str, default=None)
  p.add_argument("--dtype", type=str, default="bf16", choices=["bf16", "fp32"])
  args = p.parse_args()

  from xotorch_amd.parallel.comm import init_distributed, max_over_ranks
  import torch.distributed as dist

  rank, world = init_distributed()
  if world != args.gpus and int(os.getenv("WORLD_SIZE", "1")) != args.gpus:
    # --gpus describes the intended N; trust the actual world size
    pass
  device = args.device or ("cuda" if torch.cuda.is_available() else "cpu")
  dtype = torch.bfloat16 if args.dtype == "bf16" else torch.float32
  cfg_override = None
  if device == "cpu" and args.model == "llama-3-70b":
    # CPU smoke path (no GPU in dev container): tiny model, same code path;
    # give it >= world layers so every rank gets a shard
    from xotorch_amd.models.registry import builtin_config
    args.model = "dummy"
    args.prompt_len = min(args.prompt_len, 128)  # dummy max_seq_len is 256
    world_hint = int(os.getenv("WORLD_SIZE", "1"))
    if world_hint > 4:
      cfg_override = dict(builtin_config("dummy"))
      cfg_override["num_hidden_layers"] = world_hint

  from xotorch_amd.parallel.ring import RingPipeline

  max_gen = args.warmup + args.steps + 4
  ring = RingPipeline(
    model_id=args.model, rank=rank, world=world, device=device, dtype=dtype,
    mb_batch=args.mb_batch, prompt_len=args.prompt_len, max_gen=max_gen,
    use_graphs=not args.no_graphs, cfg_override=cfg_override,
  )
  global_batch = args.mb_batch * ring.M

  # ---- prefill (TTFT measured at the last stage) ----
  stats = ring.prefill()
  ttfts = torch.tensor(stats.ttft_ms, dtype=torch.float64, device="cpu")
  if world > 1:
    # TTFT lives on the last stage; ship it to rank 0 (gloo groups move it
    # on host, nccl on device)
    if device == "cuda" and dist.get_backend() != "gloo":
      ttfts = ttfts.to(device)
    dist.broadcast(ttfts, world - 1)
    ttfts = ttfts.cpu()
  p50_ttft = float(statistics.median(ttfts.tolist()))

  # ---- warmup ----
  for _ in range(args.warmup):
    ring.decode_step()

  # ---- timed region ----
  if device == "cuda":
    torch.cuda.synchronize()
  if dist.is_initialized():
    dist.barrier()
  if device == "cuda":
    torch.cuda.synchronize()
  t0 = time.perf_counter()
  for _ in range(args.steps):
    ring.decode_step()
  if device == "cuda":
    torch.cuda.synchronize()
  if dist.is_initialized():
    dist.barrier()
  if device == "cuda":
    torch.cuda.synchronize()
  elapsed = time.perf_counter() - t0
  elapsed = max_over_ranks(elapsed, device if device == "cuda" else "cpu")
  ring.finish()

  tokens = global_batch * args.steps
  toks_per_s = tokens / elapsed
  ms_per_step = elapsed / args.steps * 1000.0

  if rank == 0:
    result = {
      "metric": "decode tokens/sec (whole ring)",
      "value": round(toks_per_s, 2),
      "unit": "tokens/s",
      "n_gpus": world,
      "steps": args.steps,
      "warmup": args.warmup,
      "ms_per_step": round(ms_per_step, 3),
      "higher_is_better": True,
      "scaling": "weak",
      "vs_baseline": None,
      "dtype": (args.dtype
                + ("+fp8-w8a8" if os.getenv("XOT_FP8_GEMM", "0") == "1" else "")
                + ("+fp8-kv" if os.getenv("XOT_FP8_KV", "0") == "1" else "")),
      "data": "synthetic",
      "config": {
        "model": args.model,
        "global_batch": global_batch,
        "seq_len": args.prompt_len,
        "parallelism": f"pp{world}",
        "micro_batches": ring.M,
        "mb_batch": args.mb_batch,
        "p50_ttft_ms": round(p50_ttft, 1),
        "ttft_semantics": "whole-batch pipelined prefill (request-level TTFT: serve path)",
        "sampling": "greedy",
        "hip_graphs": not args.no_graphs and device == "cuda",
      },
    }
    print(json.dumps(result))
  if dist.is_initialized():
    dist.destroy_process_group()


if __name__ == "__main__":
  main()
