"""Device-resident KV caches, laid out for the CDNA4 decode-attention kernel.

Layout [B, KVH, T, hd]: each (b, kv_head) has T contiguous rows of hd
elements (256 B at hd=128 bf16) — the decode kernel streams rows with
coalesced 16 B/lane loads. Capacity is sized per request (prompt + max gen),
allocated from the 288 GB HBM pool; the reference's equivalent is torchtune's
per-layer cache setup (/root/reference/xotorch/inference/torch/sharded_inference_engine.py:71-82).
"""
from __future__ import annotations

from typing import List, Tuple

import torch


class ShardKVCache:
  """One (k, v) cache pair per local layer of a shard."""

  def __init__(self, n_layers: int, batch: int, n_kv_heads: int, capacity: int, head_dim: int,
               dtype: torch.dtype = torch.bfloat16, device: str = "cpu"):
    self.capacity = capacity
    self.batch = batch
    self.caches: List[Tuple[torch.Tensor, torch.Tensor]] = []
    for _ in range(n_layers):
      k = torch.zeros(batch, n_kv_heads, capacity, head_dim, dtype=dtype, device=device)
      v = torch.zeros(batch, n_kv_heads, capacity, head_dim, dtype=dtype, device=device)
      self.caches.append((k, v))

  def __getitem__(self, i):
    return self.caches[i]

  def __len__(self):
    return len(self.caches)

  def reset(self):
    for k, v in self.caches:
      k.zero_()
      v.zero_()

  def nbytes(self) -> int:
    return sum(k.numel() * k.element_size() + v.numel() * v.element_size() for k, v in self.caches)
