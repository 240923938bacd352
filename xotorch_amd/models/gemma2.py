"""Sharded Gemma2 decoder (gemma2-9b / gemma2-27b).

A SEPARATE decoder from ShardedModel (models/llama.py): gemma2 differs in
too many places to share the hot path — scaled embeddings, gemma-style
RMSNorm (x_norm * (1 + w), fp32), FOUR norms per block (post-attn and
post-ffn sandwich norms), attention logit soft-capping, alternating
sliding-window/global layers, query_pre_attn_scalar scaling, GeGLU MLP and
final-logit soft-capping. The reference lists gemma2 model cards but its
torchtune GQA assembly implements none of this
(/root/reference/xotorch/inference/torch/models/general_mha.py:23-254 — no
softcap/sliding-window/post-norms), so those cards cannot produce correct
outputs there.

On MI355X the family runs the SAME CDNA4 kernel path as llama: fused qkv
XotLinear GEMMs (packed skinny kernels), the fused RoPE+KV-append kernel
into the MFMA-packed cache, MFMA flash attention with softcap/window flags
(hip_ops.hip attn_*_mfma), gemma RMSNorm via the HIP rmsnorm kernel's
w_bias=1 mode, and a packed GeGLU kernel. head_dim must be 128 for the
MFMA attention (gemma2-27b); other dims fall back to the eager oracle path
(also the CPU path, validated against transformers' Gemma2ForCausalLM).

Shard semantics match ShardedModel: first shard owns the (scaled) embedding,
last owns the final norm + tied head; caches are the engine's LayerKV pairs.
"""
from __future__ import annotations

import math
import os
from typing import List, Optional

import torch
import torch.nn as nn

from xotorch_amd import ops
from xotorch_amd.models.config import ModelConfig
from xotorch_amd.models.llama import XotLinear
from xotorch_amd.ops.torch_ref import rope_cos_sin
from xotorch_amd.shard import Shard


def _rms(x: torch.Tensor, w: torch.Tensor, eps: float) -> torch.Tensor:
  # gemma convention: fp32 norm scaled by (1 + w); the HIP rmsnorm kernel
  # runs this as its w_bias=1 mode on GPU
  return ops.rmsnorm(x.contiguous(), w, eps, w_bias=1.0)


def _softcap(x: torch.Tensor, cap: float) -> torch.Tensor:
  return torch.tanh(x / cap) * cap if cap else x


def _use_mfma(x: torch.Tensor, hd: int, kv) -> bool:
  return (
    x.is_cuda and x.dtype == torch.bfloat16 and hd == 128
    and len(kv) > 2 and kv[2] is not None and ops.hip_available()
  )


class Gemma2Attention(nn.Module):
  def __init__(self, cfg: ModelConfig, layer_idx: int):
    super().__init__()
    self.cfg = cfg
    H, KVH, hd, D = cfg.n_heads, cfg.n_kv_heads, cfg.head_dim, cfg.dim
    # fused qkv: one [D, (H+2KVH)*hd] GEMM feeding the RoPE+append kernel
    self.qkv_proj = XotLinear(D, (H + 2 * KVH) * hd, bias=False)
    self.o_proj = XotLinear(H * hd, D, bias=False)
    self.scale = (cfg.query_pre_attn_scalar or cfg.head_dim) ** -0.5
    # even layers use the sliding window, odd layers are global (transformers
    # Gemma2DecoderLayer: is_sliding = not bool(layer_idx % 2))
    self.window = cfg.sliding_window if (layer_idx % 2 == 0) else 0

  def forward(self, x, cos, sin, positions, kv, start_pos: int,
              is_decode: bool = False, seq_lens=None):
    cfg = self.cfg
    B, S, _ = x.shape
    H, KVH, hd = cfg.n_heads, cfg.n_kv_heads, cfg.head_dim
    cap = cfg.attn_logit_softcapping or 0.0
    qkv = self.qkv_proj(x)  # [B, S, (H+2KVH)*hd]
    if kv is None:
      return self._forward_nocache(qkv, cos, sin, positions, cap)
    k_cache, v_cache = kv[0], kv[1]
    if _use_mfma(qkv, hd, kv):
      kp, vp = kv[2], kv[3]
      ksc = kv[4] if len(kv) > 4 else None
      vsc = kv[5] if len(kv) > 5 else None
      ops.rope_qkv_append(qkv, cos, sin, positions, k_cache, v_cache, H, KVH, hd, kp, vp,
                          k_scale=ksc, v_scale=vsc)
      q = qkv[:, :, : H * hd].view(B, S, H, hd)
      if is_decode:
        sl = seq_lens if seq_lens is not None else start_pos + 1
        out = ops.attn_decode(q, k_cache, v_cache, sl, kp, vp,
                              scale=self.scale, softcap=cap, window=self.window,
                              k_scale=ksc, v_scale=vsc)
      else:
        out = ops.attn_prefill(q, k_cache, v_cache, start_pos, S, kp, vp,
                               scale=self.scale, softcap=cap, window=self.window)
      return self.o_proj(out.reshape(B, S, H * hd))
    # eager oracle path (CPU / hd != 128): split the fused projection
    if start_pos < 0:  # ring/serve decode contract: derive from positions
      prow = positions.reshape(-1)
      if prow.numel() == B * S and S == 1 and B > 1 and int(prow.min()) != int(prow.max()):
        # ragged per-row slot positions: correctness-first row loop
        outs = []
        for b in range(B):
          outs.append(self.forward(x[b:b + 1], cos, sin, prow[b:b + 1],
                                   tuple(t[b:b + 1] if t is not None else None for t in kv),
                                   int(prow[b]), is_decode, None))
        return torch.cat(outs, dim=0)
      start_pos = int(prow[0])
    from xotorch_amd.ops import torch_ref as tr
    q, k, v = torch.split(qkv, [H * hd, KVH * hd, KVH * hd], dim=-1)
    q = q.view(B, S, H, hd)
    k = k.view(B, S, KVH, hd)
    v = v.view(B, S, KVH, hd)
    q, k = tr.rope_apply(q, k, cos, sin, positions)
    tr.kv_append(k_cache, v_cache, k, v, start_pos)
    total = start_pos + S
    rep = H // KVH
    kk = k_cache[:, :, :total].repeat_interleave(rep, dim=1)  # [B, H, T, hd]
    vv = v_cache[:, :, :total].repeat_interleave(rep, dim=1)
    scores = torch.einsum("bshd,bhtd->bhst", q.float(), kk.float()) * self.scale
    scores = _softcap(scores, cap)
    qpos = torch.arange(start_pos, total, device=x.device)[:, None]
    kpos = torch.arange(0, total, device=x.device)[None, :]
    mask = kpos <= qpos
    if self.window:
      mask &= kpos > qpos - self.window
    scores = scores.masked_fill(~mask[None, None], float("-inf"))
    probs = torch.softmax(scores, dim=-1).to(vv.dtype)
    out = torch.einsum("bhst,bhtd->bshd", probs, vv).reshape(B, S, H * hd)
    return self.o_proj(out)


  def _forward_nocache(self, qkv, cos, sin, positions, cap):
    """Cache-free causal attention (training/eval) on the same projections."""
    from xotorch_amd.ops import torch_ref as tr
    cfg = self.cfg
    B, S = qkv.shape[0], qkv.shape[1]
    H, KVH, hd = cfg.n_heads, cfg.n_kv_heads, cfg.head_dim
    q, k, v = torch.split(qkv, [H * hd, KVH * hd, KVH * hd], dim=-1)
    q = q.reshape(B, S, H, hd)
    k = k.reshape(B, S, KVH, hd)
    v = v.reshape(B, S, KVH, hd)
    q, k = tr.rope_apply(q, k, cos, sin, positions)
    rep = H // KVH
    kk = k.transpose(1, 2).repeat_interleave(rep, dim=1)  # [B,H,S,hd]
    vv = v.transpose(1, 2).repeat_interleave(rep, dim=1)
    scores = torch.einsum("bshd,bhtd->bhst", q.float(), kk.float()) * self.scale
    scores = _softcap(scores, cap)
    pos = torch.arange(S, device=qkv.device)
    mask = pos[None, :] <= pos[:, None]
    if self.window:
      mask &= pos[None, :] > pos[:, None] - self.window
    scores = scores.masked_fill(~mask[None, None], float("-inf"))
    probs = torch.softmax(scores, dim=-1).to(vv.dtype)
    out = torch.einsum("bhst,bhtd->bshd", probs.float(), vv.float()).to(qkv.dtype)
    return self.o_proj(out.reshape(B, S, H * hd))


class Gemma2MLP(nn.Module):
  def __init__(self, cfg: ModelConfig):
    super().__init__()
    I, D = cfg.intermediate_dim, cfg.dim
    # fused [gate | up] GEMM + packed GeGLU kernel on GPU
    self.gate_up_proj = XotLinear(D, 2 * I, bias=False)
    self.down_proj = XotLinear(I, D, bias=False)
    self.intermediate = I

  def forward(self, x):
    gu = self.gate_up_proj(x)
    if gu.is_cuda and gu.dtype == torch.bfloat16 and ops.hip_available():
      h = ops.geglu_packed(gu)
    else:
      I = self.intermediate
      h = (nn.functional.gelu(gu[..., :I].float(), approximate="tanh")
           * gu[..., I:].float()).to(gu.dtype)
    return self.down_proj(h)


class Gemma2Layer(nn.Module):
  def __init__(self, cfg: ModelConfig, layer_idx: int):
    super().__init__()
    self.eps = cfg.norm_eps
    self.self_attn = Gemma2Attention(cfg, layer_idx)
    self.mlp = Gemma2MLP(cfg)
    D = cfg.dim
    self.input_layernorm = nn.Parameter(torch.zeros(D))
    self.post_attention_layernorm = nn.Parameter(torch.zeros(D))
    self.pre_feedforward_layernorm = nn.Parameter(torch.zeros(D))
    self.post_feedforward_layernorm = nn.Parameter(torch.zeros(D))

  def forward(self, h, cos, sin, positions, kv, start_pos, is_decode=False, seq_lens=None):
    res = h
    hs = _rms(h, self.input_layernorm, self.eps)
    hs = self.self_attn(hs, cos, sin, positions, kv, start_pos, is_decode, seq_lens)
    h = res + _rms(hs, self.post_attention_layernorm, self.eps)
    res = h
    hs = _rms(h, self.pre_feedforward_layernorm, self.eps)
    hs = self.mlp(hs)
    return res + _rms(hs, self.post_feedforward_layernorm, self.eps)


class Gemma2Model(nn.Module):
  """Layer range [shard.start_layer .. shard.end_layer] of a Gemma2 model.
  Same forward contract as ShardedModel (engine-compatible)."""

  def __init__(self, cfg: ModelConfig, shard: Shard):
    super().__init__()
    self.cfg = cfg
    self.shard = shard
    if shard.is_first_layer:
      self.embed_tokens = nn.Embedding(cfg.vocab_size, cfg.dim)
    self.layers = nn.ModuleDict(
      {str(i): Gemma2Layer(cfg, i) for i in range(shard.start_layer, shard.end_layer + 1)}
    )
    if shard.is_last_layer:
      self.norm = nn.Parameter(torch.zeros(cfg.dim))
      if not shard.is_first_layer:
        # tied embeddings, but the owner shard is elsewhere: local head copy
        self.lm_head = nn.Linear(cfg.dim, cfg.vocab_size, bias=False)
    cos, sin = rope_cos_sin(cfg.head_dim, cfg.max_seq_len, cfg.rope_theta, cfg.rope_scaling)
    self.register_buffer("rope_cos", cos, persistent=False)
    self.register_buffer("rope_sin", sin, persistent=False)

  def reset_rope(self):
    cos, sin = rope_cos_sin(self.cfg.head_dim, self.cfg.max_seq_len, self.cfg.rope_theta,
                            self.cfg.rope_scaling, device=self.rope_cos.device)
    self.rope_cos, self.rope_sin = cos, sin

  @property
  def local_layer_ids(self) -> List[int]:
    return list(range(self.shard.start_layer, self.shard.end_layer + 1))

  def pack_decode_weights(self, reserve_bytes: int = 8 << 30) -> int:
    """Same greedy prepack policy as ShardedModel (descending measured-win
    order, free-memory guarded)."""
    from xotorch_amd.models.llama import ShardedModel
    return ShardedModel.pack_decode_weights(self, reserve_bytes)

  def head_weight(self):
    if hasattr(self, "embed_tokens"):
      return self.embed_tokens.weight
    return self.lm_head.weight

  def forward(self, x, caches, positions, start_pos: int, is_decode: bool = False,
              seq_lens=None, last_only: bool = True):
    cfg = self.cfg
    if x.dtype in (torch.int32, torch.int64):
      h = self.embed_tokens(x) * torch.tensor(cfg.dim ** 0.5, dtype=self.embed_tokens.weight.dtype)
    else:
      h = x
    if positions.dim() == 0:
      positions = positions.reshape(1)
    use_ckpt = (caches is None and self.training and torch.is_grad_enabled()
                and os.getenv("XOT_ACT_CKPT", "0") == "1")
    for idx, lid in enumerate(self.local_layer_ids):
      kv = caches[idx] if caches is not None else None
      if use_ckpt:
        h = torch.utils.checkpoint.checkpoint(
          self.layers[str(lid)], h, self.rope_cos, self.rope_sin, positions, kv,
          start_pos, is_decode, seq_lens, use_reentrant=False)
      else:
        h = self.layers[str(lid)](h, self.rope_cos, self.rope_sin, positions, kv,
                                  start_pos, is_decode, seq_lens)
    if not self.shard.is_last_layer:
      return h
    if last_only and h.shape[1] > 1:
      h = h[:, -1:, :]
    h = _rms(h, self.norm, cfg.norm_eps)
    logits = torch.nn.functional.linear(h, self.head_weight().to(h.dtype))
    logits = _softcap(logits.float(), cfg.final_logit_softcapping).to(logits.dtype)
    if is_decode or last_only:
      return logits[:, -1, :]
    return logits


def hf_key_map_gemma2(shard: Shard, cfg: ModelConfig):
  """HF Gemma2ForCausalLM checkpoint keys → Gemma2Model state-dict keys.
  q/k/v concatenate into the fused qkv_proj, gate/up into gate_up_proj
  (the loader concatenates `key#N` parts along dim 0 in N order)."""
  mapping = {}
  if shard.is_first_layer:
    mapping["model.embed_tokens.weight"] = "embed_tokens.weight"
  if shard.is_last_layer:
    mapping["model.norm.weight"] = "norm"
    if not shard.is_first_layer:
      mapping["model.embed_tokens.weight"] = "lm_head.weight"
  for lid in range(shard.start_layer, shard.end_layer + 1):
    hf = f"model.layers.{lid}."
    ours = f"layers.{lid}."
    for i, p in enumerate(("q_proj", "k_proj", "v_proj")):
      mapping[hf + f"self_attn.{p}.weight"] = ours + f"self_attn.qkv_proj.weight#{i}"
    mapping[hf + "self_attn.o_proj.weight"] = ours + "self_attn.o_proj.weight"
    for i, p in enumerate(("gate_proj", "up_proj")):
      mapping[hf + f"mlp.{p}.weight"] = ours + f"mlp.gate_up_proj.weight#{i}"
    mapping[hf + "mlp.down_proj.weight"] = ours + "mlp.down_proj.weight"
    for p in ("input_layernorm", "post_attention_layernorm",
              "pre_feedforward_layernorm", "post_feedforward_layernorm"):
      mapping[hf + f"{p}.weight"] = ours + p
  return mapping
