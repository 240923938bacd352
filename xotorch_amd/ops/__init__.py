"""Op dispatch: hand-written CDNA4 HIP kernels on GPU, torch reference on CPU.

The HIP extension (`xotorch_amd.ops.hip`, built in-tree by
`xotorch_amd/ops/build.py` / `__graft_entry__.build()`) owns the per-token
hot path on MI355X. On a GPU box the HIP path is MANDATORY: if a CUDA tensor
reaches an op and the extension is missing, we raise rather than silently
falling back to eager PyTorch (set XOT_ALLOW_EAGER=1 to override for
debugging/ablation only).
"""
from __future__ import annotations

import os
from typing import Optional, Tuple

import torch

from xotorch_amd.ops import torch_ref

_hip = None
_hip_load_error: Optional[Exception] = None


def _load_hip():
  global _hip, _hip_load_error
  if _hip is not None or _hip_load_error is not None:
    return _hip
  try:
    import importlib
    _hip = importlib.import_module("xotorch_amd.ops._hip_ops")
  except Exception as e:  # extension not built
    _hip_load_error = e
    _hip = None
  return _hip


def hip_available() -> bool:
  return _load_hip() is not None


def _use_hip(t: torch.Tensor) -> bool:
  if not t.is_cuda:
    return False
  if torch.is_grad_enabled():
    # training path: HIP kernels are inference-only (no autograd); the torch
    # ops are differentiable. Inference always runs under inference_mode.
    return False
  if _load_hip() is not None:
    return True
  if os.getenv("XOT_ALLOW_EAGER", "0") == "1":
    return False
  raise RuntimeError(
    "xotorch_amd HIP extension is not built but a CUDA tensor reached the op "
    f"dispatch (load error: {_hip_load_error}). Build it with "
    "`python -c \"import __graft_entry__; __graft_entry__.build()\"` or set "
    "XOT_ALLOW_EAGER=1 to explicitly allow the eager fallback."
  )


def rmsnorm(x: torch.Tensor, weight: torch.Tensor, eps: float) -> torch.Tensor:
  if _use_hip(x) and x.dtype == torch.bfloat16:
    return _hip.rmsnorm(x, weight, eps)
  return torch_ref.rmsnorm(x, weight, eps)


def rmsnorm_residual(
  x: torch.Tensor, residual: torch.Tensor, weight: torch.Tensor, eps: float
) -> Tuple[torch.Tensor, torch.Tensor]:
  if _use_hip(x) and x.dtype == torch.bfloat16:
    return _hip.rmsnorm_residual(x, residual, weight, eps)
  return torch_ref.rmsnorm_residual(x, residual, weight, eps)


def rope_apply(q, k, cos, sin, positions):
  if _use_hip(q) and q.dtype == torch.bfloat16:
    return _hip.rope_apply(q, k, cos, sin, positions)
  return torch_ref.rope_apply(q, k, cos, sin, positions)


def rope_kv_append(q, k, v, cos, sin, positions, k_cache, v_cache, start_pos: int):
  """Fused: RoPE on q,k + append (rotated k, v) into the cache at start_pos.

  Returns rotated q. k/v: [B, S, KVH, hd]; caches [B, KVH, T, hd].
  """
  if _use_hip(q) and q.dtype == torch.bfloat16:
    return _hip.rope_kv_append(q, k, v, cos, sin, positions, k_cache, v_cache, start_pos)
  q_r, k_r = torch_ref.rope_apply(q, k, cos, sin, positions)
  torch_ref.kv_append(k_cache, v_cache, k_r, v, start_pos)
  return q_r


def attn_prefill(q, k_cache, v_cache, start_pos: int, s_len: int):
  if q.is_cuda:
    # Prefill goes through sdpa (rocm flash/mem-efficient backends) in model
    # dtype with native GQA — a hand-written MFMA flash-prefill kernel is a
    # planned replacement; decode (the headline metric) is the HIP kernel.
    import torch.nn.functional as F
    B, S, H, hd = q.shape
    total = start_pos + s_len
    qh = q.transpose(1, 2)
    k = k_cache[:, :, :total]
    v = v_cache[:, :, :total]
    if start_pos == 0:
      out = F.scaled_dot_product_attention(qh, k, v, is_causal=True, enable_gqa=True)
    else:
      mask = torch.ones(S, total, dtype=torch.bool, device=q.device).tril(diagonal=start_pos)
      out = F.scaled_dot_product_attention(qh, k, v, attn_mask=mask, enable_gqa=True)
    return out.transpose(1, 2).contiguous()
  return torch_ref.attn_prefill(q, k_cache, v_cache, start_pos, s_len)


def attn_decode(q, k_cache, v_cache, seq_len):
  """seq_len: int or int32 device tensor [B] (per-sequence lengths)."""
  if _use_hip(q) and q.dtype == torch.bfloat16:
    return _hip.attn_decode(q, k_cache, v_cache, seq_len)
  if isinstance(seq_len, torch.Tensor):
    seq_len = int(seq_len.max().item())
  return torch_ref.attn_decode(q, k_cache, v_cache, seq_len)


def swiglu(gate, up):
  if _use_hip(gate) and gate.dtype == torch.bfloat16:
    return _hip.swiglu(gate, up)
  return torch_ref.swiglu(gate, up)


def softmax_sample(logits, temperature: float = 0.0, top_k: int = 0, generator=None):
  # sampling is tiny; HIP path exists to keep the decode step graph-capturable
  if _use_hip(logits) and temperature > 0.0 and top_k > 0:
    q = torch.empty_like(logits, dtype=torch.float32).exponential_(1, generator=generator)
    return _hip.topk_sample(logits, q, temperature, top_k)
  return torch_ref.softmax_sample(logits, temperature, top_k, generator)
