"""torch.distributed (RCCL on GPU / gloo on CPU) process-group helpers.

This replaces the reference's gRPC-per-hop transport: stage
hops are point-to-point send/recv of resident bf16 hidden states over a
single xGMI link; token wrap-around is a tiny int send; gradient all-reduce
for the DP train path uses the communicator's topology-aware algorithms.
"""
from __future__ import annotations

import datetime
import os
from typing import Optional

import torch
import torch.distributed as dist


def init_distributed(backend: Optional[str] = None) -> tuple[int, int]:
  """Initialize from torchrun env vars; no-op single-process if absent.

  Returns (rank, world_size).
  """
  if dist.is_initialized():
    return dist.get_rank(), dist.get_world_size()
  world = int(os.getenv("WORLD_SIZE", "1"))
  if world <= 1:
    return 0, 1
  rank = int(os.getenv("RANK", "0"))
  if backend is None:
    backend = "nccl" if torch.cuda.is_available() else "gloo"
  os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
  os.environ.setdefault("MASTER_PORT", "29521")
  dist.init_process_group(backend=backend, rank=rank, world_size=world,
                          timeout=datetime.timedelta(seconds=300))
  if backend == "nccl":
    local = int(os.getenv("LOCAL_RANK", str(rank)))
    torch.cuda.set_device(local % max(1, torch.cuda.device_count()))
  return rank, world


def barrier():
  if dist.is_initialized():
    if dist.get_backend() == "nccl":
      dist.barrier(device_ids=[torch.cuda.current_device()])
    else:
      dist.barrier()


def send(t: torch.Tensor, dst: int):
  dist.send(t, dst)


def recv(buf: torch.Tensor, src: int):
  dist.recv(buf, src)
  return buf


def max_over_ranks(value: float, device) -> float:
  if not dist.is_initialized():
    return value
  t = torch.tensor([value], dtype=torch.float64, device=device)
  dist.all_reduce(t, op=dist.ReduceOp.MAX)
  return float(t.item())


def broadcast_from(t: torch.Tensor, src: int):
  if dist.is_initialized():
    dist.broadcast(t, src)
  return t
