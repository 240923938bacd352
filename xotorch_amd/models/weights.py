"""Weight loading: shard-aware HF safetensors → ShardedModel, plus random init.

Parity with the reference's loader (/root/reference/xotorch/inference/torch/llm_utils.py:230-333)
minus the q/k permute (our RoPE uses the HF rotate-half convention, so HF
tensors load verbatim). Only files containing this shard's layers are read —
the same weight_map-driven selection the downloader uses for its
allow-patterns (/root/reference/xotorch/download/new_shard_download.py:181-194).
"""
from __future__ import annotations

import json
from pathlib import Path
from typing import Dict, Optional

import torch

from xotorch_amd.models.config import ModelConfig
from xotorch_amd.models.llama import ShardedModel, hf_key_map as _hf_key_map_llama


def hf_key_map(shard, cfg):
  """Checkpoint key map, routed by architecture."""
  mtype = getattr(cfg, "model_type", "llama")
  if mtype == "gemma2":
    from xotorch_amd.models.gemma2 import hf_key_map_gemma2
    return hf_key_map_gemma2(shard, cfg)
  if mtype in ("deepseek_v3", "deepseek_v2"):
    from xotorch_amd.models.deepseek_v3 import hf_key_map_deepseek
    return hf_key_map_deepseek(shard, cfg)
  return _hf_key_map_llama(shard, cfg)
from xotorch_amd.shard import Shard


def shard_file_set(model_dir: Path, shard: Shard, cfg: ModelConfig) -> Optional[Dict[str, list]]:
  """Map safetensors filename → [hf keys this shard needs from it]."""
  index_path = model_dir / "model.safetensors.index.json"
  mapping = hf_key_map(shard, cfg)
  if not index_path.exists():
    single = model_dir / "model.safetensors"
    if single.exists():
      return {single.name: list(mapping.keys())}
    return None
  with open(index_path) as f:
    weight_map: Dict[str, str] = json.load(f)["weight_map"]
  files: Dict[str, list] = {}
  for hf_key in mapping:
    fn = weight_map.get(hf_key)
    if fn is not None:
      files.setdefault(fn, []).append(hf_key)
  return files


def load_shard_weights(model: ShardedModel, model_dir: Path, device: str = "cpu") -> int:
  """Load this shard's tensors from HF safetensors files. Returns #tensors."""
  from safetensors import safe_open

  model_dir = Path(model_dir)
  cfg, shard = model.cfg, model.shard
  mapping = hf_key_map(shard, cfg)
  files = shard_file_set(model_dir, shard, cfg)
  if files is None:
    raise FileNotFoundError(f"no safetensors found under {model_dir}")
  state: Dict[str, torch.Tensor] = {}
  packed: Dict[str, Dict[int, torch.Tensor]] = {}
  for fn, keys in files.items():
    with safe_open(str(model_dir / fn), framework="pt", device=device) as f:
      names = set(f.keys())
      for hf_key in keys:
        if hf_key not in names:
          continue
        our = mapping[hf_key]
        if "#" in our:
          base, part = our.split("#")
          packed.setdefault(base, {})[int(part)] = f.get_tensor(hf_key)
        else:
          state[our] = f.get_tensor(hf_key)
  # fused qkv_proj / gate_up_proj: concatenate the HF fragments along dim 0
  for base, parts in packed.items():
    state[base] = torch.cat([parts[i] for i in sorted(parts)], dim=0)
  missing, unexpected = model.load_state_dict(state, strict=False)
  # tied-embedding models ship no lm_head tensor; rope buffers are computed
  real_missing = [m for m in missing if not m.startswith("rope_")]
  if cfg.tie_word_embeddings:
    real_missing = [m for m in real_missing if m != "lm_head.weight"]
  if real_missing:
    raise RuntimeError(f"missing weights for shard {shard}: {real_missing[:8]}")
  if cfg.tie_word_embeddings and hasattr(model, "lm_head") and not hasattr(model, "embed_tokens"):
    # last shard of a tied model without the embedding: need the embed tensor
    if "lm_head.weight" not in state:
      raise RuntimeError("tied-embedding model: last shard requires model.embed_tokens.weight")
  return len(state)


def remap_hf_state(sd: Dict[str, torch.Tensor], mapping: Dict[str, str]) -> Dict[str, torch.Tensor]:
  """Apply an hf_key_map to an in-memory HF state dict: rename keys and
  concatenate `key#N` fused parts along dim 0 in N order (same semantics as
  load_shard_weights' safetensors path; used by tests and converters)."""
  state: Dict[str, torch.Tensor] = {}
  packed: Dict[str, Dict[int, torch.Tensor]] = {}
  for hf_key, our in mapping.items():
    if hf_key not in sd:
      continue
    if "#" in our:
      base, part = our.split("#")
      packed.setdefault(base, {})[int(part)] = sd[hf_key]
    else:
      state[our] = sd[hf_key]
  for base, parts in packed.items():
    state[base] = torch.cat([parts[i] for i in sorted(parts)], dim=0)
  return state


@torch.no_grad()
def random_init(model: ShardedModel, seed: int = 1234, std: float = 0.02) -> None:
  """Deterministic random init (synthetic-weights benches and tests).

  Per-parameter generator seeding keyed by the parameter NAME so a split
  model and a full model get identical layer weights (the split-vs-full
  logits-equality oracle depends on this).
  """
  import zlib
  for name, p in model.named_parameters():
    # a tied-embedding last shard without the embedding holds a COPY of the
    # embedding matrix as its head: seed it as the embedding so split == full
    seed_name = name
    if name == "lm_head.weight" and model.cfg.tie_word_embeddings:
      seed_name = "embed_tokens.weight"
    g = torch.Generator(device="cpu")
    # stable across processes (python's str hash is salted per process)
    g.manual_seed((seed + zlib.crc32(seed_name.encode())) % (2**63))
    cpu_t = torch.empty(p.shape, dtype=torch.float32)
    if name.endswith("layernorm.weight") or name.endswith("norm.weight"):
      cpu_t.fill_(1.0)
    elif name.endswith(".bias"):
      cpu_t.zero_()
    else:
      cpu_t.normal_(0.0, std, generator=g)
    p.copy_(cpu_t.to(device=p.device, dtype=p.dtype))


@torch.no_grad()
def fast_random_init_gpu(model: ShardedModel, seed: int = 1234, std: float = 0.02) -> None:
  """GPU-side random init for huge models (70B+): not split-equal across
  shard boundaries, but orders of magnitude faster — weight VALUES don't
  affect decode throughput, only shapes do."""
  g = torch.Generator(device="cuda")
  g.manual_seed(seed)
  for name, p in model.named_parameters():
    if name.endswith("norm.weight"):
      p.fill_(1.0)
    elif name.endswith(".bias"):
      p.zero_()
    else:
      tmp = torch.empty(p.shape, dtype=torch.float32, device=p.device)
      tmp.normal_(0.0, std, generator=g)
      p.copy_(tmp.to(p.dtype))
      del tmp
