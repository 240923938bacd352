"""Per-request inference state.

The reference serializes tokens + the FULL causal mask + input_pos as JSON
lists on every pipeline hop (/root/reference/xotorch/inference/torch/llm_utils.py:593-631
— O(T·max_seq) per token, SURVEY.md §5 "known absurdity"). Here each stage
keeps positions/caches device-resident and only {curr_pos, seq_len} metadata
travels on the control plane.
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import List, Optional


@dataclass
class ShardInferenceState:
  curr_pos: int = 0          # absolute position of the NEXT cache slot
  total_len: int = 0         # prompt + max generation budget (cache capacity)
  tokens: List[int] = field(default_factory=list)  # decoded token ids (last shard appends)
  batch: int = 1

  def to_dict(self) -> dict:
    return {"curr_pos": self.curr_pos, "total_len": self.total_len, "batch": self.batch}

  @classmethod
  def from_dict(cls, d: Optional[dict]) -> "ShardInferenceState":
    if not d:
      return cls()
    return cls(curr_pos=int(d.get("curr_pos", 0)), total_len=int(d.get("total_len", 0)), batch=int(d.get("batch", 1)))
