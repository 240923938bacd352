"""Pure-PyTorch reference implementations of every hot op.

These are the numerics oracle for the HIP kernels (tests compare the HIP path
against these in fp32) and the CPU execution path for the plumbing tests.
They implement the same ops the reference delegates to torchtune
(SURVEY.md §2.2 table), with HF "rotate-half" RoPE convention so HF
safetensors load without the q/k permute torchtune needs
(/root/reference/xotorch/inference/torch/llm_utils.py:175-183).
"""
from __future__ import annotations

import math
from typing import Optional, Tuple

import torch
import torch.nn.functional as F


def rmsnorm(x: torch.Tensor, weight: torch.Tensor, eps: float, w_bias: float = 0.0) -> torch.Tensor:
  """RMSNorm in fp32 math, cast back to input dtype. w_bias=1 gives the
  gemma convention (scale by 1 + w)."""
  xf = x.float()
  norm = xf * torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + eps)
  return (norm * (weight.float() + w_bias)).to(x.dtype)


def rmsnorm_residual(
  x: torch.Tensor, residual: torch.Tensor, weight: torch.Tensor, eps: float,
  w_bias: float = 0.0
) -> Tuple[torch.Tensor, torch.Tensor]:
  """Fused residual-add + RMSNorm: returns (norm(x+residual), x+residual)."""
  s = (x.float() + residual.float())
  norm = s * torch.rsqrt(s.pow(2).mean(-1, keepdim=True) + eps)
  return (norm * (weight.float() + w_bias)).to(x.dtype), s.to(x.dtype)


def rope_cos_sin(
  head_dim: int,
  max_seq_len: int,
  theta: float,
  scaling=None,
  dtype: torch.dtype = torch.float32,
  device="cpu",
) -> Tuple[torch.Tensor, torch.Tensor]:
  """Precompute RoPE cos/sin tables [max_seq_len, head_dim//2] (fp32).

  `scaling` is an optional llama3-style RopeScaling (models/config.py).
  """
  inv_freq = 1.0 / (theta ** (torch.arange(0, head_dim, 2, dtype=torch.float32, device=device) / head_dim))
  if scaling is not None and getattr(scaling, "rope_type", "default") == "llama3":
    # llama3 frequency scaling: scale low-frequency components, smooth the band between.
    low_freq_wavelen = scaling.original_max_position_embeddings / scaling.low_freq_factor
    high_freq_wavelen = scaling.original_max_position_embeddings / scaling.high_freq_factor
    wavelen = 2 * math.pi / inv_freq
    scaled = torch.where(wavelen > low_freq_wavelen, inv_freq / scaling.factor, inv_freq)
    smooth = (scaling.original_max_position_embeddings / wavelen - scaling.low_freq_factor) / (
      scaling.high_freq_factor - scaling.low_freq_factor
    )
    # the medium band interpolates from the UNSCALED frequency
    smoothed = (1 - smooth) * inv_freq / scaling.factor + smooth * inv_freq
    is_medium = (wavelen >= high_freq_wavelen) & (wavelen <= low_freq_wavelen)
    inv_freq = torch.where(is_medium, smoothed, scaled)
  attn_factor = 1.0
  if scaling is not None and getattr(scaling, "rope_type", "default") == "yarn":
    # YaRN (deepseek long-context): NTK-by-parts blend of interpolated and
    # extrapolated frequencies + a log-factor magnitude correction baked
    # into cos/sin (HF _compute_yarn_parameters semantics, validated
    # against transformers' DeepseekV3 in tests/test_deepseek_cpu.py)
    dim, base, factor = head_dim, theta, scaling.factor
    orig = scaling.original_max_position_embeddings

    def corr_dim(n_rot):
      return (dim * math.log(orig / (n_rot * 2 * math.pi))) / (2 * math.log(base))

    low, high = corr_dim(scaling.beta_fast), corr_dim(scaling.beta_slow)
    if scaling.truncate:
      low, high = math.floor(low), math.ceil(high)
    low, high = max(low, 0), min(high, dim - 1)
    if low == high:
      high += 0.001
    ramp = torch.clamp(
      (torch.arange(head_dim // 2, dtype=torch.float32, device=device) - low) / (high - low), 0, 1)
    extrap_w = 1.0 - ramp
    inv_freq = (inv_freq / factor) * (1 - extrap_w) + inv_freq * extrap_w

    def get_mscale(scale, m=1.0):
      return 1.0 if scale <= 1 else 0.1 * m * math.log(scale) + 1.0

    if scaling.mscale and scaling.mscale_all_dim:
      attn_factor = get_mscale(factor, scaling.mscale) / get_mscale(factor, scaling.mscale_all_dim)
    else:
      attn_factor = get_mscale(factor)
  t = torch.arange(max_seq_len, dtype=torch.float32, device=device)
  freqs = torch.outer(t, inv_freq)
  return (freqs.cos() * attn_factor).to(dtype), (freqs.sin() * attn_factor).to(dtype)


def _rotate_half(x: torch.Tensor) -> torch.Tensor:
  half = x.shape[-1] // 2
  x1, x2 = x[..., :half], x[..., half:]
  return torch.cat((-x2, x1), dim=-1)


def rope_apply(
  q: torch.Tensor,
  k: torch.Tensor,
  cos: torch.Tensor,
  sin: torch.Tensor,
  positions: torch.Tensor,
) -> Tuple[torch.Tensor, torch.Tensor]:
  """Apply HF rotate-half RoPE.

  q: [B, S, H, hd], k: [B, S, KVH, hd]; cos/sin: [max_seq, hd//2];
  positions: [S] or [B, S] absolute position ids.
  """
  if isinstance(positions, torch.Tensor) and positions.dtype != torch.long:
    positions = positions.long()
  c = cos[positions].float()  # [S, hd//2] or [B,S,hd//2]
  s = sin[positions].float()
  if c.dim() == 2:
    c = c.unsqueeze(0)
    s = s.unsqueeze(0)
  c = torch.cat([c, c], dim=-1).unsqueeze(2)  # [B,S,1,hd]
  s = torch.cat([s, s], dim=-1).unsqueeze(2)
  qf, kf = q.float(), k.float()
  q_out = qf * c + _rotate_half(qf) * s
  k_out = kf * c + _rotate_half(kf) * s
  return q_out.to(q.dtype), k_out.to(k.dtype)


def kv_append(
  k_cache: torch.Tensor,
  v_cache: torch.Tensor,
  k: torch.Tensor,
  v: torch.Tensor,
  start_pos: int,
) -> None:
  """Append k,v ([B, S, KVH, hd]) into caches laid out [B, KVH, T, hd] at start_pos."""
  S = k.shape[1]
  k_cache[:, :, start_pos:start_pos + S, :] = k.transpose(1, 2).to(k_cache.dtype)
  v_cache[:, :, start_pos:start_pos + S, :] = v.transpose(1, 2).to(v_cache.dtype)


def attn_prefill(
  q: torch.Tensor,
  k_cache: torch.Tensor,
  v_cache: torch.Tensor,
  start_pos: int,
  s_len: int,
  scale: Optional[float] = None,
  softcap: float = 0.0,
  window: int = 0,
) -> torch.Tensor:
  """Causal GQA attention for a prefill chunk.

  q: [B, S, H, hd] (post-RoPE). Caches [B, KVH, T, hd] already contain keys
  through start_pos + s_len. Query position i attends to cache[0..start_pos+i].
  Returns [B, S, H, hd].
  """
  B, S, H, hd = q.shape
  KVH = k_cache.shape[1]
  total = start_pos + s_len
  qh = q.transpose(1, 2)  # [B,H,S,hd]
  keys = k_cache[:, :, :total].repeat_interleave(H // KVH, dim=1)
  vals = v_cache[:, :, :total].repeat_interleave(H // KVH, dim=1)
  # mask: query i (absolute pos start_pos+i) sees keys 0..start_pos+i
  mask = torch.ones(S, total, dtype=torch.bool, device=q.device).tril(diagonal=start_pos)
  if window and window > 0:
    mask &= ~torch.ones(S, total, dtype=torch.bool, device=q.device).tril(diagonal=start_pos - window)
  if scale is None:
    scale = hd ** -0.5
  if softcap and softcap > 0.0:
    scores = torch.einsum("bhsd,bhtd->bhst", qh.float(), keys.float()) * scale
    scores = torch.tanh(scores / softcap) * softcap
    scores = scores.masked_fill(~mask[None, None], float("-inf"))
    out = torch.softmax(scores, dim=-1) @ vals.float()
  else:
    out = F.scaled_dot_product_attention(
      qh.float(), keys.float(), vals.float(), attn_mask=mask, scale=scale
    )
  return out.transpose(1, 2).to(q.dtype)


def attn_decode(
  q: torch.Tensor,
  k_cache: torch.Tensor,
  v_cache: torch.Tensor,
  seq_len,
  scale: Optional[float] = None,
  softcap: float = 0.0,
  window: int = 0,
) -> torch.Tensor:
  """Single-position GQA attention against the KV cache.

  q: [B, 1, H, hd]; caches [B, KVH, T, hd] valid through seq_len.
  seq_len: int, or an int tensor [B] of per-row lengths (ragged slots).
  Returns [B, 1, H, hd].
  """
  if isinstance(seq_len, torch.Tensor) and seq_len.numel() > 1:
    lens = [int(v) for v in seq_len.reshape(-1)]
    if len(set(lens)) > 1:
      outs = [
        attn_decode(q[b:b + 1], k_cache[b:b + 1], v_cache[b:b + 1], lens[b],
                    scale, softcap, window)
        for b in range(q.shape[0])
      ]
      return torch.cat(outs, dim=0)
    seq_len = lens[0]
  elif isinstance(seq_len, torch.Tensor):
    seq_len = int(seq_len.reshape(-1)[0])
  B, _, H, hd = q.shape
  KVH = k_cache.shape[1]
  lo = max(0, seq_len - window) if window and window > 0 else 0
  qh = q.transpose(1, 2).float()  # [B,H,1,hd]
  keys = k_cache[:, :, lo:seq_len].repeat_interleave(H // KVH, dim=1).float()
  vals = v_cache[:, :, lo:seq_len].repeat_interleave(H // KVH, dim=1).float()
  if scale is None:
    scale = hd ** -0.5
  if softcap and softcap > 0.0:
    scores = (qh @ keys.transpose(-1, -2)) * scale
    scores = torch.tanh(scores / softcap) * softcap
    out = torch.softmax(scores, dim=-1) @ vals
  else:
    out = F.scaled_dot_product_attention(qh, keys, vals, scale=scale)
  return out.transpose(1, 2).to(q.dtype)


def rope_qkv_append(
  qkv: torch.Tensor,
  cos: torch.Tensor,
  sin: torch.Tensor,
  positions: torch.Tensor,
  k_cache: torch.Tensor,
  v_cache: torch.Tensor,
  n_heads: int,
  n_kv_heads: int,
  head_dim: int,
  q_norm: torch.Tensor = None,
  k_norm: torch.Tensor = None,
  norm_eps: float = 1e-6,
) -> None:
  """Reference for the fused packed-qkv kernel: (optionally per-head-RMSNorm
  q/k — qwen3), rotate q in place inside qkv, rotate k and append (k, v)
  into the caches at `positions`."""
  B, S, _ = qkv.shape
  H, KVH, hd = n_heads, n_kv_heads, head_dim
  q = qkv[:, :, : H * hd].view(B, S, H, hd)
  k = qkv[:, :, H * hd: (H + KVH) * hd].view(B, S, KVH, hd)
  v = qkv[:, :, (H + KVH) * hd:].view(B, S, KVH, hd)
  qq, kk = q, k
  if q_norm is not None:
    qq = rmsnorm(q.float(), q_norm.float(), norm_eps).to(q.dtype)
  if k_norm is not None:
    kk = rmsnorm(k.float(), k_norm.float(), norm_eps).to(k.dtype)
  if positions.numel() == B * S and positions.numel() != S:
    # per-row positions (continuous batching): rope + append per batch row
    pos_bs = positions.reshape(B, S)
    q_r, k_r = rope_apply(qq, kk, cos, sin, pos_bs)
    q.copy_(q_r)
    for b in range(B):
      p0 = int(pos_bs[b, 0])
      k_cache[b:b + 1, :, p0:p0 + S] = k_r[b:b + 1].transpose(1, 2).to(k_cache.dtype)
      v_cache[b:b + 1, :, p0:p0 + S] = v[b:b + 1].transpose(1, 2).to(v_cache.dtype)
    return
  q_r, k_r = rope_apply(qq, kk, cos, sin, positions)
  q.copy_(q_r)
  start_pos = int(positions.reshape(-1)[0])
  kv_append(k_cache, v_cache, k_r, v, start_pos)


def swiglu_packed(gu: torch.Tensor) -> torch.Tensor:
  """silu(gate) * up on the packed [.., 2I] fused gate_up GEMM output."""
  I = gu.shape[-1] // 2
  return swiglu(gu[..., :I], gu[..., I:])


def swiglu(gate: torch.Tensor, up: torch.Tensor) -> torch.Tensor:
  """silu(gate) * up, fp32 internally."""
  return (F.silu(gate.float()) * up.float()).to(gate.dtype)


def geglu_packed(gu: torch.Tensor) -> torch.Tensor:
  """gelu_tanh(gate) * up on the packed [.., 2I] fused gate_up output (gemma2)."""
  I = gu.shape[-1] // 2
  g, u = gu[..., :I].float(), gu[..., I:].float()
  return (F.gelu(g, approximate="tanh") * u).to(gu.dtype)


def softmax_sample(
  logits: torch.Tensor,
  temperature: float = 0.0,
  top_k: int = 0,
  generator: Optional[torch.Generator] = None,
  top_p: float = 0.0,
) -> torch.Tensor:
  """Temperature + top-k + nucleus (top-p) sampling via the exponential
  trick; temp 0 → argmax. (The reference parses but ignores top_p.)

  logits: [B, V] → token ids [B].
  """
  if temperature <= 1e-4:
    # sub-epsilon temperature: logits/T overflows to inf -> NaN softmax;
    # the distribution is argmax to numerical precision anyway
    return logits.argmax(dim=-1)
  logits = logits.float() / temperature
  if top_k and top_k > 0 and top_k < logits.shape[-1]:
    kth = torch.topk(logits, top_k, dim=-1).values[..., -1, None]
    logits = torch.where(logits < kth, torch.full_like(logits, float("-inf")), logits)
  if top_p and 0.0 < top_p < 1.0:
    sorted_logits, sorted_idx = torch.sort(logits, descending=True, dim=-1)
    cum = torch.softmax(sorted_logits, dim=-1).cumsum(dim=-1)
    # keep tokens until cumulative prob exceeds top_p; STRICT inequality so
    # the argmax (prior cumulative mass 0) survives any top_p > 0
    cut = cum - torch.softmax(sorted_logits, dim=-1) > top_p
    sorted_logits = sorted_logits.masked_fill(cut, float("-inf"))
    logits = torch.full_like(logits, float("-inf")).scatter(-1, sorted_idx, sorted_logits)
  probs = torch.softmax(logits, dim=-1)
  q = torch.empty_like(probs).exponential_(1, generator=generator)
  return (probs / q).argmax(dim=-1)
