"""Model configuration: HF config.json → internal ModelConfig.

Parity with the reference's config loader
(/root/reference/xotorch/inference/torch/llm_utils.py:79-126), including the
XOT_MAX_SEQ_LEN clamp, rewritten for this framework's own decoder: one
generic GQA + SwiGLU + RMSNorm architecture covering the llama / qwen2 /
mistral families with per-family RoPE variants.
"""
from __future__ import annotations

import json
import math
import os
from dataclasses import dataclass, field
from pathlib import Path
from typing import Optional

import torch

_DTYPE_MAP = {
  "float16": torch.float16,
  "bfloat16": torch.bfloat16,
  "float32": torch.float32,
}


@dataclass
class RopeScaling:
  # llama3- or yarn-style frequency scaling; None fields → plain RoPE
  factor: float = 1.0
  low_freq_factor: float = 1.0
  high_freq_factor: float = 4.0
  original_max_position_embeddings: int = 8192
  rope_type: str = "default"
  # yarn (deepseek long-context) fields — HF _compute_yarn_parameters semantics
  beta_fast: float = 32.0
  beta_slow: float = 1.0
  mscale: float = 1.0
  mscale_all_dim: float = 0.0
  truncate: bool = True


@dataclass
class ModelConfig:
  model_id: str = "unknown"
  vocab_size: int = 32000
  dim: int = 4096
  n_layers: int = 32
  n_heads: int = 32
  n_kv_heads: int = 8
  head_dim: int = 128
  intermediate_dim: int = 14336
  norm_eps: float = 1e-5
  rope_theta: float = 500000.0
  rope_scaling: Optional[RopeScaling] = None
  max_seq_len: int = 8192
  tie_word_embeddings: bool = False
  attn_bias: bool = False  # qwen2 q/k/v bias
  qk_norm: bool = False    # qwen3 per-head RMSNorm on q/k before RoPE
  model_type: str = "llama"
  # gemma2 (separate decoder, models/gemma2.py):
  query_pre_attn_scalar: float = 0.0
  attn_logit_softcapping: float = 0.0
  final_logit_softcapping: float = 0.0
  sliding_window: int = 0
  # deepseek v3/r1 MLA + MoE (separate decoder, models/deepseek_v3.py):
  q_lora_rank: int = 0
  kv_lora_rank: int = 0
  qk_rope_head_dim: int = 0
  qk_nope_head_dim: int = 0
  v_head_dim: int = 0
  n_shared_experts: int = 0
  n_group: int = 0
  topk_group: int = 0
  routed_scaling_factor: float = 1.0
  norm_topk_prob: bool = True
  first_k_dense_replace: int = 0
  rope_interleave: bool = True

  def kv_cache_dims(self):
    """(heads, k_dim, v_dim) for per-layer cache allocation. MLA caches the
    LATENT (kv_lora_rank) + the shared roped key (qk_rope_head_dim) instead
    of expanded per-head K/V."""
    if self.model_type in ("deepseek_v3", "deepseek_v2"):
      return 1, self.kv_lora_rank, self.qk_rope_head_dim
    return self.n_kv_heads, self.head_dim, self.head_dim
  torch_dtype: torch.dtype = torch.bfloat16
  bos_token_id: Optional[int] = None
  eos_token_id: Optional[int] = None
  # MoE (Mixtral / qwen3-moe); n_experts == 0 → dense MLP
  n_experts: int = 0
  n_experts_per_tok: int = 2
  moe_intermediate_dim: int = 0   # per-expert I (qwen3moe); 0 → intermediate_dim
  moe_style: str = "mixtral"      # HF checkpoint key layout: mixtral | qwen3

  @property
  def kv_mult(self) -> int:
    return self.n_heads // self.n_kv_heads

  def estimate_param_bytes(self) -> int:
    """Rough bf16 parameter footprint — used by the partitioner for sizing."""
    per_layer = (
      self.dim * (self.n_heads + 2 * self.n_kv_heads) * self.head_dim  # qkv
      + self.n_heads * self.head_dim * self.dim  # o
      + 3 * self.dim * self.intermediate_dim * max(1, self.n_experts)  # mlp
      + 2 * self.dim  # norms
    )
    embed = self.vocab_size * self.dim * (1 if self.tie_word_embeddings else 2)
    return 2 * (self.n_layers * per_layer + embed + self.dim)


def config_from_hf(config_path: Path | str | dict, model_id: str = "unknown") -> ModelConfig:
  """Parse a HF-style config.json (path or dict) into a ModelConfig."""
  if isinstance(config_path, (str, Path)):
    with open(config_path) as f:
      raw = json.load(f)
  else:
    raw = dict(config_path)

  n_heads = raw.get("num_attention_heads", 32)
  dim = raw.get("hidden_size", 4096)
  head_dim = raw.get("head_dim") or dim // n_heads
  rope_scaling = None
  rs = raw.get("rope_scaling") or raw.get("rope_parameters")
  rs_type = rs.get("rope_type", rs.get("type", "default")) if rs else "default"
  if rs and rs_type == "llama3":
    rope_scaling = RopeScaling(
      factor=rs.get("factor", 8.0),
      low_freq_factor=rs.get("low_freq_factor", 1.0),
      high_freq_factor=rs.get("high_freq_factor", 4.0),
      original_max_position_embeddings=rs.get("original_max_position_embeddings", 8192),
      rope_type="llama3",
    )
  elif rs and rs_type == "yarn":
    orig = rs.get("original_max_position_embeddings", 4096)
    rope_scaling = RopeScaling(
      factor=rs.get("factor") or raw.get("max_position_embeddings", orig) / orig,
      original_max_position_embeddings=orig,
      rope_type="yarn",
      beta_fast=rs.get("beta_fast") or 32.0,
      beta_slow=rs.get("beta_slow") or 1.0,
      mscale=rs.get("mscale") or 1.0,
      mscale_all_dim=rs.get("mscale_all_dim") or 0.0,
      truncate=rs.get("truncate", True),
    )

  max_seq_len = raw.get("max_position_embeddings", 8192)
  env_max = os.getenv("XOT_MAX_SEQ_LEN")
  if env_max:
    max_seq_len = min(max_seq_len, int(env_max))

  eos = raw.get("eos_token_id")
  if isinstance(eos, list):
    eos = eos[0] if eos else None

  mtype = (raw.get("model_type") or "llama").lower()
  return ModelConfig(
    model_id=model_id,
    vocab_size=raw.get("vocab_size", 32000),
    dim=dim,
    n_layers=raw.get("num_hidden_layers", 32),
    n_heads=n_heads,
    n_kv_heads=raw.get("num_key_value_heads", n_heads),
    head_dim=head_dim,
    intermediate_dim=raw.get("intermediate_size", 4 * dim),
    norm_eps=raw.get("rms_norm_eps", 1e-5),
    rope_theta=raw.get("rope_theta", 10000.0),
    rope_scaling=rope_scaling,
    max_seq_len=max_seq_len,
    tie_word_embeddings=raw.get("tie_word_embeddings", False),
    attn_bias=mtype in ("qwen2",) or raw.get("attention_bias", False),
    qk_norm=mtype in ("qwen3", "qwen3_moe"),
    model_type=mtype,
    query_pre_attn_scalar=float(raw.get("query_pre_attn_scalar", 0) or 0),
    attn_logit_softcapping=float(raw.get("attn_logit_softcapping", 0) or 0),
    final_logit_softcapping=float(raw.get("final_logit_softcapping", 0) or 0),
    sliding_window=int(raw.get("sliding_window", 0) or 0),
    torch_dtype=_DTYPE_MAP.get(raw.get("torch_dtype", "bfloat16"), torch.bfloat16),
    bos_token_id=raw.get("bos_token_id"),
    eos_token_id=eos,
    n_experts=raw.get("num_local_experts", 0) or raw.get("num_experts", 0)
    or raw.get("n_routed_experts", 0) or 0,
    n_experts_per_tok=raw.get("num_experts_per_tok", 2),
    moe_intermediate_dim=raw.get("moe_intermediate_size", 0) or 0,
    moe_style="qwen3" if mtype == "qwen3_moe" else "mixtral",
    q_lora_rank=raw.get("q_lora_rank") or 0,
    kv_lora_rank=raw.get("kv_lora_rank") or 0,
    qk_rope_head_dim=raw.get("qk_rope_head_dim") or 0,
    qk_nope_head_dim=raw.get("qk_nope_head_dim") or 0,
    v_head_dim=raw.get("v_head_dim") or 0,
    n_shared_experts=raw.get("n_shared_experts") or 0,
    n_group=raw.get("n_group") or 0,
    topk_group=raw.get("topk_group") or 0,
    routed_scaling_factor=float(raw.get("routed_scaling_factor", 1.0) or 1.0),
    norm_topk_prob=bool(raw.get("norm_topk_prob", True)),
    first_k_dense_replace=raw.get("first_k_dense_replace") or 0,
    rope_interleave=bool(raw.get("rope_interleave", True)),
  )
