"""Data-parallel trainer: bucketed gradient all-reduce over RCCL/xGMI.

BASELINE.json config "Llama-3 8B torchtune LoRA fine-tune, DP=8 with RCCL
all-reduce". Design (MI355X-first): one process per GPU; backward hooks feed
gradients into fixed-size buckets that all-reduce asynchronously as soon as
they fill, overlapping communication with the rest of backward; bucket size
defaults to 32 MB — sized for per-link xGMI bandwidth (ring all-reduce is
per-link bound at ~153 GB/s, so buckets must be large enough to amortize
latency but small enough to pipeline). Falls back to gloo on CPU (tests).
"""
from __future__ import annotations

import os
from typing import Dict, List, Optional

import torch
import torch.distributed as dist


class GradBucketAllReducer:
  """Bucketed async all-reduce of gradients during backward."""

  def __init__(self, params: List[torch.nn.Parameter], bucket_bytes: int = 32 * 1024 * 1024,
               world_size: Optional[int] = None):
    self.params = [p for p in params if p.requires_grad]
    self.world = world_size or (dist.get_world_size() if dist.is_initialized() else 1)
    self.bucket_bytes = bucket_bytes
    self._pending: List[torch.Tensor] = []
    self._pending_bytes = 0
    self._works = []
    self._hooks = []
    if self.world > 1:
      for p in self.params:
        h = p.register_post_accumulate_grad_hook(self._on_grad)
        self._hooks.append(h)

  def _on_grad(self, p: torch.nn.Parameter):
    self._pending.append(p.grad)
    self._pending_bytes += p.grad.numel() * p.grad.element_size()
    if self._pending_bytes >= self.bucket_bytes:
      self._flush()

  def _flush(self):
    if not self._pending:
      return
    flat = torch._utils._flatten_dense_tensors(self._pending)
    if dist.get_backend() == "gloo" and flat.is_cuda:
      # verification mode (more ranks than GPUs): gloo reduces on host
      cpu = flat.detach().to("cpu")
      dist.all_reduce(cpu, op=dist.ReduceOp.SUM)
      flat.copy_(cpu)
      self._works.append((None, flat, list(self._pending)))
    else:
      work = dist.all_reduce(flat, op=dist.ReduceOp.SUM, async_op=True)
      self._works.append((work, flat, list(self._pending)))
    self._pending = []
    self._pending_bytes = 0

  def finalize(self):
    """Wait for outstanding reductions and scatter averaged grads back."""
    if self.world <= 1:
      return
    self._flush()
    for work, flat, grads in self._works:
      if work is not None:
        work.wait()
      flat.div_(self.world)
      for g, synced in zip(grads, torch._utils._unflatten_dense_tensors(flat, grads)):
        g.copy_(synced)
    self._works = []

  def remove(self):
    for h in self._hooks:
      h.remove()
    self._hooks = []


class DPTrainer:
  """Simple DP training loop: CE loss, bucketed all-reduce, AdamW."""

  def __init__(self, model: torch.nn.Module, lr: float = 1e-4, bucket_bytes: int = 32 * 1024 * 1024,
               trainable_params: Optional[List[torch.nn.Parameter]] = None):
    self.model = model
    params = trainable_params if trainable_params is not None else [p for p in model.parameters() if p.requires_grad]
    self.params = params
    self.opt = torch.optim.AdamW(params, lr=lr)
    self.reducer = GradBucketAllReducer(params, bucket_bytes)
    self.world = self.reducer.world

  def sync_initial_state(self):
    """Broadcast rank 0's trainable params so every replica starts identical."""
    if self.world > 1:
      staged = dist.get_backend() == "gloo" and any(p.is_cuda for p in self.params)
      for p in self.params:
        if staged and p.is_cuda:
          cpu = p.data.detach().to("cpu")
          dist.broadcast(cpu, src=0)
          p.data.copy_(cpu)
        else:
          dist.broadcast(p.data, src=0)

  def step(self, forward_fn, inputs: torch.Tensor, targets: torch.Tensor,
           lengths: Optional[torch.Tensor] = None) -> float:
    """One DP step: forward_fn(inputs) -> logits [B,S,V]; masked CE; returns loss."""
    self.opt.zero_grad(set_to_none=False)
    logits = forward_fn(inputs)
    V = logits.shape[-1]
    ce = torch.nn.functional.cross_entropy(
      logits.float().reshape(-1, V), targets.reshape(-1).long(), reduction="none"
    ).reshape(targets.shape)
    if lengths is not None:
      mask = torch.arange(targets.shape[1], device=targets.device)[None, :] < lengths[:, None]
      loss = (ce * mask).sum() / mask.sum().clamp(min=1)
    else:
      loss = ce.mean()
    loss.backward()
    self.reducer.finalize()
    self.opt.step()
    return float(loss.detach())
