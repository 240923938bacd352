"""Device-resident KV caches, laid out for the CDNA4 decode-attention kernels.

Standard layout [B, KVH, T, hd]: each (b, kv_head) has T contiguous rows of
hd elements (256 B at hd=128 bf16) — prefill attention (sdpa) and the VALU
decode kernel stream rows with coalesced 16 B/lane loads.

MFMA layout (GPU, hd=128): a second copy of K and V pre-shuffled into
v_mfma_f32_16x16x32_bf16 B-fragment order so the MFMA flash-decoding kernel
streams the cache with fully-coalesced 1 KB wave loads:
  K_packed [B, KVH, T32/16, 4, 64, 8]   (16-position x 32-hd-chunk tiles)
  V_packed [B, KVH, 8, T32/32, 64, 8]   (32-position x 16-hd-column tiles)
(T32 = capacity rounded up to 32; both copies are appended by the fused
rope_qkv_append kernel.)  The reference's equivalent is torchtune's per-layer
cache setup (/root/reference/xotorch/inference/torch/sharded_inference_engine.py:71-82).
"""
from __future__ import annotations

import os
from typing import List, NamedTuple, Optional

import torch


class LayerKV(NamedTuple):
  k: torch.Tensor
  v: torch.Tensor
  kp: Optional[torch.Tensor] = None  # MFMA-packed K (cuda, hd=128; bf16 or e4m3 bytes)
  vp: Optional[torch.Tensor] = None  # MFMA-packed V
  ksc: Optional[torch.Tensor] = None  # fp8 mode: per-row K scales [B, KVH, T32]
  vsc: Optional[torch.Tensor] = None  # fp8 mode: per-row V scales


def _want_packed(device: str, head_dim: int, dtype: torch.dtype) -> bool:
  return (
    str(device).startswith("cuda")
    and head_dim == 128
    and dtype == torch.bfloat16
    and os.getenv("XOT_MFMA_ATTN", "1") == "1"
  )


class ShardKVCache:
  """One LayerKV per local layer of a shard."""

  def __init__(self, n_layers: int, batch: int, n_kv_heads: int, capacity: int, head_dim: int,
               dtype: torch.dtype = torch.bfloat16, device: str = "cpu",
               v_dim: Optional[int] = None):
    """v_dim: per-position width of the v tensor when it differs from
    head_dim (MLA latent caches: k stores the kv_lora latent, v the shared
    roped key — see ModelConfig.kv_cache_dims())."""
    self.capacity = capacity
    self.batch = batch
    self.caches: List[LayerKV] = []
    vd = v_dim if v_dim is not None else head_dim
    packed = _want_packed(device, head_dim, dtype) and vd == head_dim
    fp8_kv = packed and os.getenv("XOT_FP8_KV", "0") == "1"
    # MLA latent cache (k = 512-dim latent, v = 64-dim roped shared key,
    # one kv "head"): fragment-packed copies for the absorbed-MQA MFMA
    # decode kernel (18 qk chunks / 32 pv groups — hip_ops.hip MLA section)
    mla_packed = (
      str(device).startswith("cuda") and dtype == torch.bfloat16
      and n_kv_heads == 1 and head_dim == 512 and vd == 64
      and os.getenv("XOT_MFMA_ATTN", "1") == "1"
    )
    t32 = (capacity + 31) // 32 * 32
    for _ in range(n_layers):
      k = torch.zeros(batch, n_kv_heads, capacity, head_dim, dtype=dtype, device=device)
      v = torch.zeros(batch, n_kv_heads, capacity, vd, dtype=dtype, device=device)
      if fp8_kv:
        # e4m3 packed copies (half the decode stream) + per-row scales;
        # plain bf16 cache stays for prefill
        kp = torch.zeros(batch, n_kv_heads, t32 // 16, 4, 64, 8, dtype=torch.uint8, device=device)
        vp = torch.zeros(batch, n_kv_heads, 8, t32 // 32, 64, 8, dtype=torch.uint8, device=device)
        ksc = torch.ones(batch, n_kv_heads, t32, dtype=torch.float32, device=device)
        vsc = torch.ones(batch, n_kv_heads, t32, dtype=torch.float32, device=device)
        self.caches.append(LayerKV(k, v, kp, vp, ksc, vsc))
      elif packed:
        kp = torch.zeros(batch, n_kv_heads, t32 // 16, 4, 64, 8, dtype=dtype, device=device)
        vp = torch.zeros(batch, n_kv_heads, 8, t32 // 32, 64, 8, dtype=dtype, device=device)
        self.caches.append(LayerKV(k, v, kp, vp))
      elif mla_packed and os.getenv("XOT_FP8_KV", "0") == "1":
        kp = torch.zeros(batch, t32 // 16, 18, 64, 8, dtype=torch.uint8, device=device)
        vp = torch.zeros(batch, 32, t32 // 32, 64, 8, dtype=torch.uint8, device=device)
        ksc = torch.ones(batch, t32, dtype=torch.float32, device=device)
        self.caches.append(LayerKV(k, v, kp, vp, ksc))
      elif mla_packed:
        kp = torch.zeros(batch, t32 // 16, 18, 64, 8, dtype=dtype, device=device)
        vp = torch.zeros(batch, 32, t32 // 32, 64, 8, dtype=dtype, device=device)
        self.caches.append(LayerKV(k, v, kp, vp))
      else:
        self.caches.append(LayerKV(k, v))

  def __getitem__(self, i):
    return self.caches[i]

  def __len__(self):
    return len(self.caches)

  def reset(self):
    for c in self.caches:
      c.k.zero_()
      c.v.zero_()
      if c.kp is not None:
        c.kp.zero_()
        c.vp.zero_()

  def nbytes(self) -> int:
    total = 0
    for c in self.caches:
      total += c.k.numel() * c.k.element_size() + c.v.numel() * c.v.element_size()
      if c.kp is not None:
        total += c.kp.numel() * c.kp.element_size() + c.vp.numel() * c.vp.element_size()
    return total
