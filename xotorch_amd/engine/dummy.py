"""Deterministic fake engine for plumbing tests (no model, no GPU).

Parity with /root/reference/xotorch/inference/dummy_inference_engine.py:7-37:
tokens pass through shards unchanged (+1 on hidden hops), fixed tokenizer,
eos=69 — lets the whole Node/ring/API stack run with --inference-engine dummy.
"""
from __future__ import annotations

from typing import Optional, Tuple

import numpy as np

from xotorch_amd.engine.interface import InferenceEngine
from xotorch_amd.engine.tokenizers import DummyTokenizer
from xotorch_amd.shard import Shard


class DummyEngine(InferenceEngine):
  def __init__(self):
    self.tokenizer = DummyTokenizer()
    self.shard: Optional[Shard] = None

  async def ensure_shard(self, shard: Shard):
    self.shard = shard

  async def encode(self, shard: Shard, prompt: str) -> np.ndarray:
    await self.ensure_shard(shard)
    return np.asarray(self.tokenizer.encode(prompt), dtype=np.int64)

  async def decode(self, shard: Shard, tokens) -> str:
    await self.ensure_shard(shard)
    return self.tokenizer.decode(list(np.asarray(tokens).reshape(-1)))

  async def sample(self, x: np.ndarray, temp: float = 0.0, top_k: int = 0, top_p: float = 0.0) -> np.ndarray:
    return np.asarray(x).reshape(np.asarray(x).shape[0], -1)[:, -1].astype(np.int64) % 256

  async def infer_tensor(
    self, request_id: str, shard: Shard, input_data: np.ndarray, inference_state: Optional[dict] = None
  ) -> Tuple[np.ndarray, Optional[dict]]:
    await self.ensure_shard(shard)
    x = np.asarray(input_data)
    if x.ndim == 3:
      out = x + 1.0  # hidden-state hop: +1 per shard (deterministic, checkable)
    else:
      out = x.astype(np.float32)
    state = dict(inference_state or {})
    state["curr_pos"] = int(state.get("curr_pos", 0)) + x.shape[1]
    return out, state
