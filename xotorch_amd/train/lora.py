"""LoRA adapters for the sharded decoder.

The reference's train CLI is LoRA-flavored (sample data under train/data/lora,
SURVEY.md §3.4) but ships no engine implementation. Here: low-rank A/B
adapters wrap the projection Linears of a ShardedModel; base weights freeze;
merge/unmerge supported for inference; adapter state dicts save/load
separately (small checkpoints).
"""
from __future__ import annotations

import math
from typing import Iterable, List, Optional

import torch
import torch.nn as nn


class LoRALinear(nn.Module):
  def __init__(self, base: nn.Linear, rank: int = 8, alpha: float = 16.0, dropout: float = 0.0):
    super().__init__()
    self.base = base
    self.rank = rank
    self.scaling = alpha / rank
    # adapters live on the base weight's device, in fp32 (cast per-forward):
    # fp32 master adapters keep LoRA updates stable under bf16 bases
    dev = base.weight.device
    self.lora_a = nn.Parameter(torch.zeros(rank, base.in_features, device=dev, dtype=torch.float32))
    self.lora_b = nn.Parameter(torch.zeros(base.out_features, rank, device=dev, dtype=torch.float32))
    nn.init.kaiming_uniform_(self.lora_a, a=math.sqrt(5))
    self.dropout = nn.Dropout(dropout) if dropout > 0 else nn.Identity()
    self.merged = False
    base.weight.requires_grad_(False)
    if base.bias is not None:
      base.bias.requires_grad_(False)

  def forward(self, x):
    out = self.base(x)
    if not self.merged:
      lora = self.dropout(x) @ self.lora_a.to(x.dtype).t() @ self.lora_b.to(x.dtype).t()
      out = out + lora * self.scaling
    return out

  @torch.no_grad()
  def merge(self):
    if not self.merged:
      delta = (self.lora_b @ self.lora_a) * self.scaling
      self.base.weight.add_(delta.to(self.base.weight.dtype))
      self.merged = True

  @torch.no_grad()
  def unmerge(self):
    if self.merged:
      delta = (self.lora_b @ self.lora_a) * self.scaling
      self.base.weight.sub_(delta.to(self.base.weight.dtype))
      self.merged = False


DEFAULT_TARGETS = ("qkv_proj", "o_proj", "gate_up_proj", "down_proj")


def apply_lora(model: nn.Module, rank: int = 8, alpha: float = 16.0,
               targets: Iterable[str] = DEFAULT_TARGETS, dropout: float = 0.0) -> List[str]:
  """Wrap matching Linears with LoRALinear; freeze everything else.

  Returns the list of wrapped module paths.
  """
  targets = tuple(targets)
  for p in model.parameters():
    p.requires_grad_(False)
  wrapped = []
  for name, module in list(model.named_modules()):
    for child_name, child in list(module.named_children()):
      if isinstance(child, nn.Linear) and child_name in targets:
        setattr(module, child_name, LoRALinear(child, rank=rank, alpha=alpha, dropout=dropout))
        wrapped.append(f"{name}.{child_name}" if name else child_name)
  return wrapped


def lora_parameters(model: nn.Module):
  return [p for n, p in model.named_parameters() if "lora_" in n]


def lora_state_dict(model: nn.Module) -> dict:
  return {n: p.detach().cpu() for n, p in model.named_parameters() if "lora_" in n}


def load_lora_state_dict(model: nn.Module, sd: dict) -> int:
  own = dict(model.named_parameters())
  n = 0
  with torch.no_grad():
    for name, t in sd.items():
      if name in own:
        own[name].copy_(t.to(own[name].dtype))
        n += 1
  return n
