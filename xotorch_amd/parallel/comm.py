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

  Backend selection: RCCL ("nccl") whenever every rank can own its own GPU.
  RCCL refuses two ranks on one device ("Duplicate GPU detected" — that is
  what killed the round-1 2-ranks/1-GPU attempt, gpurun_out/b8_2rank.log),
  so when WORLD_SIZE exceeds the visible GPU count we fall back to gloo for
  the process group while compute stays on the GPU (tensors are staged
  through pinned CPU buffers for the hops — see RingComm). That makes
  multi-rank verification runnable in a 1-GPU lease; on the driver's 8-GPU
  node each rank gets its own device and the path is pure RCCL over xGMI.

  Returns (rank, world_size).
  """
  if dist.is_initialized():
    return dist.get_rank(), dist.get_world_size()
  world = int(os.getenv("WORLD_SIZE", "1"))
  if world <= 1:
    return 0, 1
  rank = int(os.getenv("RANK", "0"))
  if backend is None:
    backend = os.getenv("XOT_RING_BACKEND")
  if backend is None:
    n_gpus = torch.cuda.device_count() if torch.cuda.is_available() else 0
    backend = "nccl" if n_gpus >= world else "gloo"
  os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
  os.environ.setdefault("MASTER_PORT", "29521")
  dist.init_process_group(backend=backend, rank=rank, world_size=world,
                          timeout=datetime.timedelta(seconds=300))
  if torch.cuda.is_available():
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
  if dist.get_backend() == "gloo":
    device = "cpu"  # gloo collectives run on host tensors
  t = torch.tensor([value], dtype=torch.float64, device=device)
  dist.all_reduce(t, op=dist.ReduceOp.MAX)
  return float(t.item())


class _StagedWork:
  """Completion handle for a p2p op staged through a host buffer."""

  def __init__(self, req, cpu: Optional[torch.Tensor] = None, dst: Optional[torch.Tensor] = None):
    self._req, self._cpu, self._dst = req, cpu, dst

  def wait(self):
    self._req.wait()
    if self._dst is not None:
      self._dst.copy_(self._cpu.view(self._dst.dtype), non_blocking=False)


class RingComm:
  """Point-to-point hop transport for the ring pipeline.

  On the product path (one rank per GPU) this is a thin veneer over RCCL
  send/recv: resident bf16 tensors move directly over one xGMI link per hop.
  When the process group runs on gloo with CUDA compute (multi-rank
  verification on fewer GPUs than ranks), tensors are staged through pinned
  host buffers; gloo has no bf16 wire type, so staged buffers travel as
  int16 views of the same bytes.
  """

  def __init__(self, device: str):
    self.device = device
    self.staged = (
      dist.is_initialized() and dist.get_backend() == "gloo" and device == "cuda"
    )
    self._bufs: dict = {}

  def _host(self, t: torch.Tensor) -> torch.Tensor:
    # Keyed by identity: async ops on different persistent buffers of the
    # same shape (e.g. the M per-micro-batch token buffers) must not share
    # a staging buffer. Callers pass long-lived tensors for async ops.
    wire_dtype = torch.int16 if t.dtype == torch.bfloat16 else t.dtype
    key = (id(t), tuple(t.shape), t.dtype)
    buf = self._bufs.get(key)
    if buf is None:
      buf = torch.empty(t.shape, dtype=wire_dtype, device="cpu", pin_memory=True)
      self._bufs[key] = buf
    return buf

  def send(self, t: torch.Tensor, dst: int):
    if self.staged:
      # blocking: a transient host copy is fine (and avoids caching buffers
      # for temporaries like prefill activations)
      c = t.contiguous()
      c = c.view(torch.int16) if c.dtype == torch.bfloat16 else c
      dist.send(c.cpu(), dst)
    else:
      dist.send(t.contiguous(), dst)

  def recv(self, buf: torch.Tensor, src: int) -> torch.Tensor:
    if self.staged:
      c = self._host(buf)
      dist.recv(c, src)
      buf.copy_(c.view(buf.dtype), non_blocking=False)
    else:
      dist.recv(buf, src)
    return buf

  def isend(self, t: torch.Tensor, dst: int):
    if self.staged:
      c = self._host(t)
      c.copy_(t.view(c.dtype) if t.dtype == torch.bfloat16 else t, non_blocking=False)
      return _StagedWork(dist.isend(c, dst))
    return dist.isend(t, dst)

  def irecv(self, buf: torch.Tensor, src: int):
    if self.staged:
      c = self._host(buf)
      return _StagedWork(dist.irecv(c, src), cpu=c, dst=buf)
    return dist.irecv(buf, src)

  def broadcast(self, t: torch.Tensor, src: int) -> torch.Tensor:
    if not dist.is_initialized():
      return t
    if self.staged:
      c = self._host(t)
      c.copy_(t.view(c.dtype) if t.dtype == torch.bfloat16 else t, non_blocking=False)
      dist.broadcast(c, src)
      t.copy_(c.view(t.dtype), non_blocking=False)
    else:
      dist.broadcast(t, src)
    return t


def broadcast_from(t: torch.Tensor, src: int):
  if dist.is_initialized():
    dist.broadcast(t, src)
  return t
