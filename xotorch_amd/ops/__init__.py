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


def _use_hip(*tensors: torch.Tensor) -> bool:
  t = tensors[0]
  if not t.is_cuda:
    return False
  if torch.is_grad_enabled() and any(x.requires_grad for x in tensors):
    # training path: HIP kernels are inference-only (no autograd); the torch
    # ops are differentiable. Inference runs on the HIP kernels.
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


def rmsnorm(x: torch.Tensor, weight: torch.Tensor, eps: float, w_bias: float = 0.0) -> torch.Tensor:
  """w_bias=1.0 gives the gemma convention (scale by 1 + w, fp32 math)."""
  if _use_hip(x, weight) and x.dtype == torch.bfloat16:
    return _hip.rmsnorm(x, weight, eps, w_bias)
  return torch_ref.rmsnorm(x, weight, eps, w_bias)


def rmsnorm_residual(
  x: torch.Tensor, residual: torch.Tensor, weight: torch.Tensor, eps: float,
  w_bias: float = 0.0
) -> Tuple[torch.Tensor, torch.Tensor]:
  if _use_hip(x, residual, weight) and x.dtype == torch.bfloat16:
    return _hip.rmsnorm_residual(x, residual, weight, eps, w_bias)
  return torch_ref.rmsnorm_residual(x, residual, weight, eps, w_bias)


def rope_apply(q, k, cos, sin, positions):
  """Standalone RoPE (training / oracle paths). On the GPU inference hot
  path RoPE is fused into rope_qkv_append — there is deliberately no
  separate HIP kernel for this op."""
  return torch_ref.rope_apply(q, k, cos, sin, positions)


def rope_qkv_append(qkv, cos, sin, positions, k_cache, v_cache, n_heads: int, n_kv_heads: int,
                    head_dim: int, kp=None, vp=None, q_norm=None, k_norm=None,
                    norm_eps: float = 1e-6, k_scale=None, v_scale=None):
  """Fused on the packed qkv GEMM output [B,S,(H+2KVH)*hd]: (optionally
  per-head-RMSNorm q/k — qwen3), RoPE-rotate the q heads in place, rotate k
  heads into the cache, copy v heads into the cache. When the MFMA-packed
  cache copies (kp, vp) exist they are appended too.
  """
  if _use_hip(qkv) and qkv.dtype == torch.bfloat16:
    _hip.rope_qkv_append(qkv, cos, sin, positions, k_cache, v_cache, n_heads, n_kv_heads, head_dim,
                         kp, vp, q_norm, k_norm, norm_eps, k_scale, v_scale)
    return
  torch_ref.rope_qkv_append(qkv, cos, sin, positions, k_cache, v_cache, n_heads, n_kv_heads,
                            head_dim, q_norm, k_norm, norm_eps)


def attn_prefill(q, k_cache, v_cache, start_pos: int, s_len: int, kp=None, vp=None,
                 scale=None, softcap: float = 0.0, window: int = 0):
  if q.is_cuda and q.dtype == torch.bfloat16 and kp is not None and q.shape[3] == 128 \
     and kp.dtype == torch.bfloat16 \
     and os.getenv("XOT_MFMA_ATTN", "1") == "1" and _use_hip(q):
    # causal flash-forward on matrix cores, streaming the MFMA-packed cache;
    # softcap/window run gemma2 semantics inside the same kernel (fp8-KV
    # mode keeps prefill on the plain bf16 cache via the sdpa path below)
    return _hip.attn_prefill_mfma(q, kp, vp, start_pos, scale or 0.0, softcap, window)
  if q.is_cuda and not softcap and not window:
    # fallback: sdpa (rocm flash/mem-efficient backends) in model dtype with
    # native GQA — hd != 128 or unpacked-cache path
    import torch.nn.functional as F
    B, S, H, hd = q.shape
    total = start_pos + s_len
    qh = q.transpose(1, 2)
    k = k_cache[:, :, :total]
    v = v_cache[:, :, :total]
    if start_pos == 0:
      out = F.scaled_dot_product_attention(qh, k, v, is_causal=True, enable_gqa=True, scale=scale)
    else:
      mask = torch.ones(S, total, dtype=torch.bool, device=q.device).tril(diagonal=start_pos)
      out = F.scaled_dot_product_attention(qh, k, v, attn_mask=mask, enable_gqa=True, scale=scale)
    return out.transpose(1, 2).contiguous()
  return torch_ref.attn_prefill(q, k_cache, v_cache, start_pos, s_len, scale, softcap, window)


def attn_decode(q, k_cache, v_cache, seq_len, kp=None, vp=None,
                scale=None, softcap: float = 0.0, window: int = 0,
                k_scale=None, v_scale=None):
  """seq_len: int or int32 device tensor [B] (per-sequence lengths).
  With MFMA-packed cache copies (kp, vp) the flash-decoding kernel scores on
  matrix cores (v_mfma_f32_16x16x32_bf16, coalesced 1 KB cache streams).
  softcap/window select gemma2 semantics (MFMA or torch_ref path only)."""
  if _use_hip(q) and q.dtype == torch.bfloat16:
    if not isinstance(seq_len, torch.Tensor):
      seq_len = torch.full((q.shape[0],), int(seq_len), dtype=torch.int32, device=q.device)
    rep = q.shape[2] // k_cache.shape[1]
    if kp is not None and rep <= 16 and os.getenv("XOT_MFMA_ATTN", "1") == "1":
      return _hip.attn_decode_mfma(q, kp, vp, seq_len, k_cache.shape[2],
                                   scale or 0.0, softcap, window, k_scale, v_scale)
    if not softcap and not window and scale is None:
      return _hip.attn_decode(q, k_cache, v_cache, seq_len)
  return torch_ref.attn_decode(q, k_cache, v_cache, seq_len, scale, softcap, window)


def swiglu(gate, up):
  if _use_hip(gate, up) and gate.dtype == torch.bfloat16:
    return _hip.swiglu(gate, up)
  return torch_ref.swiglu(gate, up)


def swiglu_packed(gu):
  """silu(gate)*up on the packed [.., 2I] fused gate_up GEMM output."""
  if _use_hip(gu) and gu.dtype == torch.bfloat16:
    return _hip.swiglu_packed(gu)
  return torch_ref.swiglu_packed(gu)


def geglu_packed(gu):
  """gelu_tanh(gate)*up on the packed [.., 2I] fused gate_up output (gemma2)."""
  if _use_hip(gu) and gu.dtype == torch.bfloat16:
    return _hip.geglu_packed(gu)
  return torch_ref.geglu_packed(gu)


_SKINNY = os.getenv("XOT_SKINNY", "1") == "1"


def pack_decode_weight(w: torch.Tensor) -> torch.Tensor:
  """Pre-shuffle a [N, K] bf16 weight into MFMA A-fragment order for the
  packed skinny decode GEMM: [N/32, K/16, 64 lanes, 8 bf16] with lane =
  (k-half)*32 + n-row. One-time at weight load; coalesced 1 KB wave streams
  at decode."""
  N, K = w.shape
  assert N % 32 == 0 and K % 64 == 0, (N, K)
  return w.view(N // 32, 32, K // 16, 2, 8).permute(0, 2, 3, 1, 4).contiguous()


def pack_decode_weight_fp8(w: torch.Tensor):
  """Quantize a [N, K] weight to OCP e4m3 with per-output-channel scales and
  pre-shuffle into the MFMA fragment order (bytes): returns (packed_uint8,
  s_w fp32 [N]). Half the stream bytes of the bf16 prepack — the W8A8 decode
  GEMM path (XOT_FP8_GEMM=1)."""
  N, K = w.shape
  assert N % 32 == 0 and K % 64 == 0, (N, K)
  s_w = (w.float().abs().amax(dim=1).clamp(min=1e-12) / 448.0)
  w8 = (w.float() / s_w[:, None]).clamp(-448.0, 448.0).to(torch.float8_e4m3fn)
  packed = w8.view(torch.uint8).view(N // 32, 32, K // 16, 2, 8).permute(0, 2, 3, 1, 4).contiguous()
  return packed, s_w.contiguous()


def fp8_gemm_enabled() -> bool:
  return os.getenv("XOT_FP8_GEMM", "0") == "1"


def linear(x: torch.Tensor, weight: torch.Tensor, bias: Optional[torch.Tensor] = None) -> torch.Tensor:
  """y = x @ weight^T (+bias) on hipBLASLt/aten.

  The decode hot path uses XotLinear.pack_decode() + skinny_gemm_packed (the
  hand-written CDNA4 weight-streaming MFMA kernel) instead; this is the
  library-GEMM path for prefill/training/CPU and unpacked weights.
  """
  return torch.nn.functional.linear(x, weight, bias)


def softmax_sample(logits, temperature: float = 0.0, top_k: int = 0, generator=None,
                   top_p: float = 0.0):
  # torch ops here are graph-capturable (argmax / exponential+argmax)
  return torch_ref.softmax_sample(logits, temperature, top_k, generator, top_p)
