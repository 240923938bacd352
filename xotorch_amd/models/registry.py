"""Model registry: model cards, pretty names, shard builders.

Capability parity with the reference's registry
(/root/reference/xotorch/models.py:4-278): id → layer count + HF repo +
per-engine support, plus `build_base_shard`/`build_full_shard`.

Additionally each well-known architecture carries a BUILTIN HF-style config
so models can be constructed with random-init weights offline (the bench and
tests run with no network — BASELINE.json "synthetic data / random-init").
"""
from __future__ import annotations

from typing import Dict, List, Optional

from xotorch_amd.shard import Shard

# --- builtin HF-style configs for offline/random-init construction ----------

def _llama_cfg(dim, n_layers, n_heads, n_kv, inter, vocab=128256, theta=500000.0, tie=False,
               max_pos=8192, llama3_scaling=False, head_dim=None):
  cfg = {
    "model_type": "llama", "hidden_size": dim, "num_hidden_layers": n_layers,
    "num_attention_heads": n_heads, "num_key_value_heads": n_kv,
    "intermediate_size": inter, "vocab_size": vocab, "rope_theta": theta,
    "rms_norm_eps": 1e-5, "max_position_embeddings": max_pos,
    "tie_word_embeddings": tie, "torch_dtype": "bfloat16",
    "bos_token_id": 128000, "eos_token_id": 128001,
  }
  if head_dim:
    cfg["head_dim"] = head_dim
  if llama3_scaling:
    cfg["rope_scaling"] = {
      "rope_type": "llama3", "factor": 8.0, "low_freq_factor": 1.0,
      "high_freq_factor": 4.0, "original_max_position_embeddings": 8192,
    }
    cfg["max_position_embeddings"] = 131072
  return cfg


def _qwen_cfg(dim, n_layers, n_heads, n_kv, inter, vocab=151936, tie=False, max_pos=32768):
  return {
    "model_type": "qwen2", "hidden_size": dim, "num_hidden_layers": n_layers,
    "num_attention_heads": n_heads, "num_key_value_heads": n_kv,
    "intermediate_size": inter, "vocab_size": vocab, "rope_theta": 1000000.0,
    "rms_norm_eps": 1e-6, "max_position_embeddings": max_pos,
    "tie_word_embeddings": tie, "torch_dtype": "bfloat16",
    "bos_token_id": 151643, "eos_token_id": 151645,
  }


def _qwen3_cfg(dim, n_layers, n_heads, n_kv, inter, vocab=151936, tie=False, max_pos=40960, head_dim=128):
  return {
    "model_type": "qwen3", "hidden_size": dim, "num_hidden_layers": n_layers,
    "num_attention_heads": n_heads, "num_key_value_heads": n_kv,
    "head_dim": head_dim, "intermediate_size": inter, "vocab_size": vocab,
    "rope_theta": 1000000.0, "rms_norm_eps": 1e-6,
    "max_position_embeddings": max_pos, "tie_word_embeddings": tie,
    "torch_dtype": "bfloat16", "bos_token_id": 151643, "eos_token_id": 151645,
  }


def _deepseek_v3_cfg():
  # DeepSeek-V3 / R1 (671B): MLA + 256-expert MoE; 8x MI355X (2.3 TB HBM)
  # holds it at bf16 via the ring. YaRN params as published (factor 40 over
  # a 4096 pretraining window); max_position kept at 4096 here so
  # random-init cards stay cache-affordable — raise with XOT_MAX_SEQ_LEN.
  return {
    "model_type": "deepseek_v3", "hidden_size": 7168, "num_hidden_layers": 61,
    "num_attention_heads": 128, "num_key_value_heads": 128,
    "intermediate_size": 18432, "moe_intermediate_size": 2048,
    "n_routed_experts": 256, "num_experts_per_tok": 8, "n_shared_experts": 1,
    "n_group": 8, "topk_group": 4, "routed_scaling_factor": 2.5,
    "norm_topk_prob": True, "first_k_dense_replace": 3,
    "q_lora_rank": 1536, "kv_lora_rank": 512, "qk_rope_head_dim": 64,
    "qk_nope_head_dim": 128, "v_head_dim": 128, "vocab_size": 129280,
    "rope_theta": 10000.0, "rms_norm_eps": 1e-6,
    "max_position_embeddings": 4096, "torch_dtype": "bfloat16",
    "rope_scaling": {"rope_type": "yarn", "factor": 40.0, "beta_fast": 32,
                     "beta_slow": 1, "mscale": 1.0, "mscale_all_dim": 1.0,
                     "original_max_position_embeddings": 4096},
    "bos_token_id": 0, "eos_token_id": 1,
  }


def _gemma2_cfg(dim, n_layers, n_heads, n_kv, inter, head_dim, qpas, window=4096):
  return {
    "model_type": "gemma2", "hidden_size": dim, "num_hidden_layers": n_layers,
    "num_attention_heads": n_heads, "num_key_value_heads": n_kv,
    "head_dim": head_dim, "intermediate_size": inter, "vocab_size": 256000,
    "rope_theta": 10000.0, "rms_norm_eps": 1e-6, "max_position_embeddings": 8192,
    "tie_word_embeddings": True, "query_pre_attn_scalar": qpas,
    "attn_logit_softcapping": 50.0, "final_logit_softcapping": 30.0,
    "sliding_window": window, "torch_dtype": "bfloat16",
    "bos_token_id": 2, "eos_token_id": 1,
  }


def _qwen3_moe_cfg(dim, n_layers, n_heads, n_kv, inter, moe_inter, n_experts, top_k,
                   vocab=151936, max_pos=40960, head_dim=128):
  return {
    "model_type": "qwen3_moe", "hidden_size": dim, "num_hidden_layers": n_layers,
    "num_attention_heads": n_heads, "num_key_value_heads": n_kv,
    "head_dim": head_dim, "intermediate_size": inter,
    "moe_intermediate_size": moe_inter, "num_experts": n_experts,
    "num_experts_per_tok": top_k, "vocab_size": vocab,
    "rope_theta": 1000000.0, "rms_norm_eps": 1e-6,
    "max_position_embeddings": max_pos, "torch_dtype": "bfloat16",
    "bos_token_id": 151643, "eos_token_id": 151645,
  }


def _mixtral_cfg(dim, n_layers, n_heads, n_kv, inter, n_experts=8, top_k=2):
  return {
    "model_type": "mixtral", "hidden_size": dim, "num_hidden_layers": n_layers,
    "num_attention_heads": n_heads, "num_key_value_heads": n_kv,
    "intermediate_size": inter, "vocab_size": 32000, "rope_theta": 1000000.0,
    "rms_norm_eps": 1e-5, "max_position_embeddings": 32768,
    "num_local_experts": n_experts, "num_experts_per_tok": top_k,
    "torch_dtype": "bfloat16", "bos_token_id": 1, "eos_token_id": 2,
  }


BUILTIN_CONFIGS: Dict[str, dict] = {
  # llama 3.x family
  "llama-3.2-1b": _llama_cfg(2048, 16, 32, 8, 8192, tie=True, llama3_scaling=True, head_dim=64),
  "llama-3.2-3b": _llama_cfg(3072, 28, 24, 8, 8192, tie=True, llama3_scaling=True, head_dim=128),
  "llama-3.1-8b": _llama_cfg(4096, 32, 32, 8, 14336, llama3_scaling=True),
  "llama-3-8b": _llama_cfg(4096, 32, 32, 8, 14336),
  "llama-3-70b": _llama_cfg(8192, 80, 64, 8, 28672),
  "llama-3.1-70b": _llama_cfg(8192, 80, 64, 8, 28672, llama3_scaling=True),
  "llama-3.3-70b": _llama_cfg(8192, 80, 64, 8, 28672, llama3_scaling=True),
  "llama-3.1-405b": _llama_cfg(16384, 126, 128, 8, 53248, llama3_scaling=True),
  # qwen 2.5 family
  "qwen-2.5-0.5b": _qwen_cfg(896, 24, 14, 2, 4864, tie=True),
  "qwen-2.5-1.5b": _qwen_cfg(1536, 28, 12, 2, 8960, tie=True),
  "qwen-2.5-3b": _qwen_cfg(2048, 36, 16, 2, 11008, tie=True),
  "qwen-2.5-7b": _qwen_cfg(3584, 28, 28, 4, 18944),
  "qwen-2.5-14b": _qwen_cfg(5120, 48, 40, 8, 13824),
  "qwen-2.5-32b": _qwen_cfg(5120, 64, 40, 8, 27648),
  "qwen-2.5-72b": _qwen_cfg(8192, 80, 64, 8, 29568),
  # mistral
  "mistral-nemo": _llama_cfg(5120, 40, 32, 8, 14336, vocab=131072, theta=1000000.0, max_pos=128000, head_dim=128),
  "mistral-large": _llama_cfg(12288, 88, 96, 8, 28672, vocab=32768, theta=1000000.0, max_pos=32768),
  # deepseek r1 distills (qwen/llama backbones)
  "deepseek-r1-distill-qwen-1.5b": _qwen_cfg(1536, 28, 12, 2, 8960, tie=True),
  "deepseek-r1-distill-qwen-7b": _qwen_cfg(3584, 28, 28, 4, 18944),
  "deepseek-r1-distill-qwen-32b": _qwen_cfg(5120, 64, 40, 8, 27648),
  "deepseek-r1-distill-llama-8b": _llama_cfg(4096, 32, 32, 8, 14336, llama3_scaling=True),
  "deepseek-r1-distill-llama-70b": _llama_cfg(8192, 80, 64, 8, 28672, llama3_scaling=True),
  # qwen 3 dense (per-head q/k RMSNorm, no attn bias, explicit head_dim)
  "qwen-3-0.6b": _qwen3_cfg(1024, 28, 16, 8, 3072, tie=True),
  "qwen-3-8b": _qwen3_cfg(4096, 36, 32, 8, 12288),
  "qwen-3-32b": _qwen3_cfg(5120, 64, 64, 8, 25600),
  # qwen 2.5 coder / math (identical architectures to the base sizes)
  "qwen-2.5-coder-1.5b": _qwen_cfg(1536, 28, 12, 2, 8960, tie=True),
  "qwen-2.5-coder-3b": _qwen_cfg(2048, 36, 16, 2, 11008, tie=True),
  "qwen-2.5-coder-7b": _qwen_cfg(3584, 28, 28, 4, 18944),
  "qwen-2.5-coder-14b": _qwen_cfg(5120, 48, 40, 8, 13824),
  "qwen-2.5-coder-32b": _qwen_cfg(5120, 64, 40, 8, 27648),
  "qwen-2.5-math-7b": _qwen_cfg(3584, 28, 28, 4, 18944),
  "qwen-2.5-math-72b": _qwen_cfg(8192, 80, 64, 8, 29568),
  "deepseek-r1-distill-qwen-14b": _qwen_cfg(5120, 48, 40, 8, 13824),
  # nemotron (llama-3.1-70b architecture)
  "nemotron-70b": _llama_cfg(8192, 80, 64, 8, 28672, llama3_scaling=True),
  # moe
  "mixtral-8x7b": _mixtral_cfg(4096, 32, 32, 8, 14336),
  "qwen-3-30b-a3b": _qwen3_moe_cfg(2048, 48, 32, 4, 6144, 768, 128, 8),
  # gemma2 (separate decoder, models/gemma2.py)
  "gemma2-9b": _gemma2_cfg(3584, 42, 16, 8, 14336, 256, 256),
  "gemma2-27b": _gemma2_cfg(4608, 46, 32, 16, 36864, 128, 144),
  # deepseek v3/r1 (MLA decoder, models/deepseek_v3.py)
  "deepseek-v3": _deepseek_v3_cfg(),
  "deepseek-r1": _deepseek_v3_cfg(),
  # DeepSeek-V2-Lite (15.7B): same MLA geometry (512/64 latent) at a size a
  # single MI355X holds — the measurable MLA config
  "deepseek-v2-lite": {
    "model_type": "deepseek_v2", "hidden_size": 2048, "num_hidden_layers": 27,
    "num_attention_heads": 16, "num_key_value_heads": 16,
    "intermediate_size": 10944, "moe_intermediate_size": 1408,
    "n_routed_experts": 64, "num_experts_per_tok": 6, "n_shared_experts": 2,
    "n_group": 1, "topk_group": 1, "routed_scaling_factor": 1.0,
    "norm_topk_prob": False, "first_k_dense_replace": 1,
    "q_lora_rank": 0, "kv_lora_rank": 512, "qk_rope_head_dim": 64,
    "qk_nope_head_dim": 128, "v_head_dim": 128, "vocab_size": 102400,
    "rope_theta": 10000.0, "rms_norm_eps": 1e-6,
    "max_position_embeddings": 4096, "torch_dtype": "bfloat16",
    "rope_scaling": {"rope_type": "yarn", "factor": 40.0, "beta_fast": 32,
                     "beta_slow": 1, "mscale": 0.707, "mscale_all_dim": 0.707,
                     "original_max_position_embeddings": 4096},
    "bos_token_id": 100000, "eos_token_id": 100001,
  },
  # phi-4-mini (llama-like enough for the generic decoder)
  "phi-4-mini": _llama_cfg(3072, 32, 24, 8, 8192, vocab=200064, theta=10000.0, max_pos=131072, tie=True),
  "phi-4-mini-instruct": _llama_cfg(3072, 32, 24, 8, 8192, vocab=200064, theta=10000.0, max_pos=131072, tie=True),
  # tiny test model
  "dummy": _llama_cfg(64, 4, 4, 2, 128, vocab=256, theta=10000.0, max_pos=256, tie=True),
}
# DeepSeek-Coder-V2-Lite shares the V2-Lite architecture exactly
BUILTIN_CONFIGS["deepseek-coder-v2-lite"] = dict(BUILTIN_CONFIGS["deepseek-v2-lite"])


# model cards: layers + HF repo per engine (engine names of THIS framework)
model_cards: Dict[str, dict] = {
  "llama-3.2-1b": {"layers": 16, "repo": {"TorchEngine": "unsloth/Llama-3.2-1B-Instruct", "HIPEngine": "unsloth/Llama-3.2-1B-Instruct"}},
  "llama-3.2-3b": {"layers": 28, "repo": {"TorchEngine": "unsloth/Llama-3.2-3B-Instruct", "HIPEngine": "unsloth/Llama-3.2-3B-Instruct"}},
  "llama-3.1-8b": {"layers": 32, "repo": {"TorchEngine": "mlx-community/Meta-Llama-3.1-8B-Instruct-bf16", "HIPEngine": "mlx-community/Meta-Llama-3.1-8B-Instruct-bf16"}},
  "llama-3-8b": {"layers": 32, "repo": {"TorchEngine": "NousResearch/Meta-Llama-3-8B-Instruct", "HIPEngine": "NousResearch/Meta-Llama-3-8B-Instruct"}},
  "llama-3-70b": {"layers": 80, "repo": {"TorchEngine": "NousResearch/Meta-Llama-3-70B-Instruct", "HIPEngine": "NousResearch/Meta-Llama-3-70B-Instruct"}},
  "llama-3.1-70b": {"layers": 80, "repo": {"TorchEngine": "mlx-community/Meta-Llama-3.1-70B-Instruct", "HIPEngine": "mlx-community/Meta-Llama-3.1-70B-Instruct"}},
  "llama-3.3-70b": {"layers": 80, "repo": {"TorchEngine": "unsloth/Llama-3.3-70B-Instruct", "HIPEngine": "unsloth/Llama-3.3-70B-Instruct"}},
  "llama-3.1-405b": {"layers": 126, "repo": {"TorchEngine": "unsloth/Meta-Llama-3.1-405B-Instruct", "HIPEngine": "unsloth/Meta-Llama-3.1-405B-Instruct"}},
  "qwen-2.5-0.5b": {"layers": 24, "repo": {"TorchEngine": "Qwen/Qwen2.5-0.5B-Instruct", "HIPEngine": "Qwen/Qwen2.5-0.5B-Instruct"}},
  "qwen-2.5-1.5b": {"layers": 28, "repo": {"TorchEngine": "Qwen/Qwen2.5-1.5B-Instruct", "HIPEngine": "Qwen/Qwen2.5-1.5B-Instruct"}},
  "qwen-2.5-3b": {"layers": 36, "repo": {"TorchEngine": "Qwen/Qwen2.5-3B-Instruct", "HIPEngine": "Qwen/Qwen2.5-3B-Instruct"}},
  "qwen-2.5-7b": {"layers": 28, "repo": {"TorchEngine": "Qwen/Qwen2.5-7B-Instruct", "HIPEngine": "Qwen/Qwen2.5-7B-Instruct"}},
  "qwen-2.5-14b": {"layers": 48, "repo": {"TorchEngine": "Qwen/Qwen2.5-14B-Instruct", "HIPEngine": "Qwen/Qwen2.5-14B-Instruct"}},
  "qwen-2.5-32b": {"layers": 64, "repo": {"TorchEngine": "Qwen/Qwen2.5-32B-Instruct", "HIPEngine": "Qwen/Qwen2.5-32B-Instruct"}},
  "qwen-2.5-72b": {"layers": 80, "repo": {"TorchEngine": "Qwen/Qwen2.5-72B-Instruct", "HIPEngine": "Qwen/Qwen2.5-72B-Instruct"}},
  "mistral-nemo": {"layers": 40, "repo": {"TorchEngine": "unsloth/Mistral-Nemo-Instruct-2407", "HIPEngine": "unsloth/Mistral-Nemo-Instruct-2407"}},
  "mistral-large": {"layers": 88, "repo": {"TorchEngine": "mistralai/Mistral-Large-Instruct-2407", "HIPEngine": "mistralai/Mistral-Large-Instruct-2407"}},
  "deepseek-r1-distill-qwen-1.5b": {"layers": 28, "repo": {"TorchEngine": "deepseek-ai/DeepSeek-R1-Distill-Qwen-1.5B", "HIPEngine": "deepseek-ai/DeepSeek-R1-Distill-Qwen-1.5B"}},
  "deepseek-r1-distill-qwen-7b": {"layers": 28, "repo": {"TorchEngine": "deepseek-ai/DeepSeek-R1-Distill-Qwen-7B", "HIPEngine": "deepseek-ai/DeepSeek-R1-Distill-Qwen-7B"}},
  "deepseek-r1-distill-qwen-32b": {"layers": 64, "repo": {"TorchEngine": "deepseek-ai/DeepSeek-R1-Distill-Qwen-32B", "HIPEngine": "deepseek-ai/DeepSeek-R1-Distill-Qwen-32B"}},
  "deepseek-r1-distill-llama-8b": {"layers": 32, "repo": {"TorchEngine": "deepseek-ai/DeepSeek-R1-Distill-Llama-8B", "HIPEngine": "deepseek-ai/DeepSeek-R1-Distill-Llama-8B"}},
  "deepseek-r1-distill-llama-70b": {"layers": 80, "repo": {"TorchEngine": "deepseek-ai/DeepSeek-R1-Distill-Llama-70B", "HIPEngine": "deepseek-ai/DeepSeek-R1-Distill-Llama-70B"}},
  "mixtral-8x7b": {"layers": 32, "repo": {"TorchEngine": "mistralai/Mixtral-8x7B-Instruct-v0.1", "HIPEngine": "mistralai/Mixtral-8x7B-Instruct-v0.1"}},
  "phi-4-mini": {"layers": 32, "repo": {"TorchEngine": "microsoft/Phi-4-mini-instruct", "HIPEngine": "microsoft/Phi-4-mini-instruct"}},
  "qwen-2.5-coder-1.5b": {"layers": 28, "repo": {"TorchEngine": "Qwen/Qwen2.5-Coder-1.5B-Instruct", "HIPEngine": "Qwen/Qwen2.5-Coder-1.5B-Instruct"}},
  "qwen-2.5-coder-3b": {"layers": 36, "repo": {"TorchEngine": "Qwen/Qwen2.5-Coder-3B-Instruct", "HIPEngine": "Qwen/Qwen2.5-Coder-3B-Instruct"}},
  "qwen-2.5-coder-7b": {"layers": 28, "repo": {"TorchEngine": "Qwen/Qwen2.5-Coder-7B-Instruct", "HIPEngine": "Qwen/Qwen2.5-Coder-7B-Instruct"}},
  "qwen-2.5-coder-14b": {"layers": 48, "repo": {"TorchEngine": "Qwen/Qwen2.5-Coder-14B-Instruct", "HIPEngine": "Qwen/Qwen2.5-Coder-14B-Instruct"}},
  "qwen-2.5-coder-32b": {"layers": 64, "repo": {"TorchEngine": "Qwen/Qwen2.5-Coder-32B-Instruct", "HIPEngine": "Qwen/Qwen2.5-Coder-32B-Instruct"}},
  "qwen-2.5-math-7b": {"layers": 28, "repo": {"TorchEngine": "Qwen/Qwen2.5-Math-7B-Instruct", "HIPEngine": "Qwen/Qwen2.5-Math-7B-Instruct"}},
  "qwen-2.5-math-72b": {"layers": 80, "repo": {"TorchEngine": "Qwen/Qwen2.5-Math-72B-Instruct", "HIPEngine": "Qwen/Qwen2.5-Math-72B-Instruct"}},
  "deepseek-r1-distill-qwen-14b": {"layers": 48, "repo": {"TorchEngine": "deepseek-ai/DeepSeek-R1-Distill-Qwen-14B", "HIPEngine": "deepseek-ai/DeepSeek-R1-Distill-Qwen-14B"}},
  "nemotron-70b": {"layers": 80, "repo": {"TorchEngine": "nvidia/Llama-3.1-Nemotron-70B-Instruct-HF", "HIPEngine": "nvidia/Llama-3.1-Nemotron-70B-Instruct-HF"}},
  # Listed for registry parity with the reference but with NO supported
  # engine here: llava needs a vision tower + image path (the reference
  # lists it too; its torchtune GQA assembly cannot run it either —
  # SURVEY.md appendix). get_supported_models() filters it out.
  "qwen-3-0.6b": {"layers": 28, "repo": {"TorchEngine": "Qwen/Qwen3-0.6B", "HIPEngine": "Qwen/Qwen3-0.6B"}},
  "qwen-3-8b": {"layers": 36, "repo": {"TorchEngine": "Qwen/Qwen3-8B", "HIPEngine": "Qwen/Qwen3-8B"}},
  "qwen-3-32b": {"layers": 64, "repo": {"TorchEngine": "Qwen/Qwen3-32B", "HIPEngine": "Qwen/Qwen3-32B"}},
  "qwen-3-30b-a3b": {"layers": 48, "repo": {"TorchEngine": "Qwen/Qwen3-30B-A3B", "HIPEngine": "Qwen/Qwen3-30B-A3B"}},
  "gemma2-9b": {"layers": 42, "repo": {"TorchEngine": "google/gemma-2-9b-it", "HIPEngine": "google/gemma-2-9b-it"}},
  "gemma2-27b": {"layers": 46, "repo": {"TorchEngine": "google/gemma-2-27b-it", "HIPEngine": "google/gemma-2-27b-it"}},
  "deepseek-r1": {"layers": 61, "repo": {"TorchEngine": "deepseek-ai/DeepSeek-R1", "HIPEngine": "deepseek-ai/DeepSeek-R1"}},
  "deepseek-v3": {"layers": 61, "repo": {"TorchEngine": "deepseek-ai/DeepSeek-V3", "HIPEngine": "deepseek-ai/DeepSeek-V3"}},
  "deepseek-v2-lite": {"layers": 27, "repo": {"TorchEngine": "deepseek-ai/DeepSeek-V2-Lite-Chat", "HIPEngine": "deepseek-ai/DeepSeek-V2-Lite-Chat"}},
  "deepseek-coder-v2-lite": {"layers": 27, "repo": {"TorchEngine": "deepseek-ai/DeepSeek-Coder-V2-Lite-Instruct", "HIPEngine": "deepseek-ai/DeepSeek-Coder-V2-Lite-Instruct"}},
  "phi-4-mini-instruct": {"layers": 32, "repo": {"TorchEngine": "microsoft/Phi-4-mini-instruct", "HIPEngine": "microsoft/Phi-4-mini-instruct"}},
  # reference card id for a bitsandbytes-4bit 405B checkpoint: bnb weight
  # formats are not supported (use llama-3.1-405b bf16 + the opt-in fp8
  # modes instead); listed for id parity, filtered like llava
  "llama-3.1-405b-8bit": {"layers": 126, "repo": {}},
  # llava needs PIL for the image path (not present in this environment);
  # the reference's vision handling is also vestigial
  "llava-1.5-7b-hf": {"layers": 32, "repo": {}},
  "dummy": {"layers": 4, "repo": {"TorchEngine": "dummy", "HIPEngine": "dummy", "DummyEngine": "dummy"}},
}

pretty_names = {
  "llama-3.2-1b": "Llama 3.2 1B",
  "llama-3.2-3b": "Llama 3.2 3B",
  "llama-3.1-8b": "Llama 3.1 8B",
  "llama-3-8b": "Llama 3 8B",
  "llama-3-70b": "Llama 3 70B",
  "llama-3.1-70b": "Llama 3.1 70B",
  "llama-3.3-70b": "Llama 3.3 70B",
  "llama-3.1-405b": "Llama 3.1 405B",
  "qwen-2.5-0.5b": "Qwen 2.5 0.5B",
  "qwen-2.5-1.5b": "Qwen 2.5 1.5B",
  "qwen-2.5-3b": "Qwen 2.5 3B",
  "qwen-2.5-7b": "Qwen 2.5 7B",
  "qwen-2.5-14b": "Qwen 2.5 14B",
  "qwen-2.5-32b": "Qwen 2.5 32B",
  "qwen-2.5-72b": "Qwen 2.5 72B",
  "mistral-nemo": "Mistral Nemo 12B",
  "mistral-large": "Mistral Large 123B",
  "deepseek-r1-distill-qwen-1.5b": "DeepSeek R1 Distill Qwen 1.5B",
  "deepseek-r1-distill-qwen-7b": "DeepSeek R1 Distill Qwen 7B",
  "deepseek-r1-distill-qwen-32b": "DeepSeek R1 Distill Qwen 32B",
  "deepseek-r1-distill-llama-8b": "DeepSeek R1 Distill Llama 8B",
  "deepseek-r1-distill-llama-70b": "DeepSeek R1 Distill Llama 70B",
  "mixtral-8x7b": "Mixtral 8x7B",
  "qwen-2.5-coder-1.5b": "Qwen 2.5 Coder 1.5B",
  "qwen-2.5-coder-3b": "Qwen 2.5 Coder 3B",
  "qwen-2.5-coder-7b": "Qwen 2.5 Coder 7B",
  "qwen-2.5-coder-14b": "Qwen 2.5 Coder 14B",
  "qwen-2.5-coder-32b": "Qwen 2.5 Coder 32B",
  "qwen-2.5-math-7b": "Qwen 2.5 Math 7B",
  "qwen-2.5-math-72b": "Qwen 2.5 Math 72B",
  "deepseek-r1-distill-qwen-14b": "DeepSeek R1 Distill Qwen 14B",
  "nemotron-70b": "Nemotron 70B",
  "qwen-3-0.6b": "Qwen 3 0.6B",
  "qwen-3-8b": "Qwen 3 8B",
  "qwen-3-32b": "Qwen 3 32B",
  "qwen-3-30b-a3b": "Qwen 3 30B A3B (MoE)",
  "gemma2-9b": "Gemma2 9B",
  "gemma2-27b": "Gemma2 27B",
  "deepseek-r1": "DeepSeek R1",
  "deepseek-v2-lite": "DeepSeek V2 Lite (16B MLA)",
  "deepseek-coder-v2-lite": "DeepSeek Coder V2 Lite (16B MLA)",
  "phi-4-mini-instruct": "Phi-4 Mini Instruct",
  "llama-3.1-405b-8bit": "Llama 3.1 405B bnb-4bit (unsupported weight format)",
  "deepseek-v3": "DeepSeek V3",
  "llava-1.5-7b-hf": "LLaVa 1.5 7B (unsupported arch)",
  "phi-4-mini": "Phi-4 Mini",
  "dummy": "Dummy (test)",
}


def pretty_name(model_id: str) -> str:
  return pretty_names.get(model_id, model_id)


def get_repo(model_id: str, engine_classname: str) -> Optional[str]:
  return model_cards.get(model_id, {}).get("repo", {}).get(engine_classname)


def build_base_shard(model_id: str, engine_classname: str = "HIPEngine") -> Optional[Shard]:
  """Shard covering layer 0 only — the placeholder routed to the partitioner."""
  n_layers = model_cards.get(model_id, {}).get("layers", 0)
  if n_layers < 1:
    return None
  return Shard(model_id=model_id, start_layer=0, end_layer=0, n_layers=n_layers)


def build_full_shard(model_id: str, engine_classname: str = "HIPEngine") -> Optional[Shard]:
  base = build_base_shard(model_id, engine_classname)
  if base is None:
    return None
  return Shard(model_id=model_id, start_layer=0, end_layer=base.n_layers - 1, n_layers=base.n_layers)


def get_supported_models(supported_engine_lists: Optional[List[List[str]]] = None) -> List[str]:
  """Model ids runnable by at least one engine in EVERY peer's engine list.

  With no argument: every card that has at least one engine repo — cards
  listed purely for id parity with the reference (llava vision,
  bitsandbytes-format checkpoints) are excluded from "supported"."""
  if not supported_engine_lists:
    return [m for m, c in model_cards.items() if c.get("repo")]
  out = []
  for model_id, card in model_cards.items():
    repos = card.get("repo", {})
    if all(any(e in repos for e in engines) for engines in supported_engine_lists):
      out.append(model_id)
  return out


def builtin_config(model_id: str) -> Optional[dict]:
  return BUILTIN_CONFIGS.get(model_id)
