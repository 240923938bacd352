"""InferenceEngine contract + factory.

Parity with the reference's engine ABC
(/root/reference/xotorch/inference/inference_engine.py:11-74) with the
capability gap closed: `train`/`evaluate` are part of the contract and are
actually implemented by the engines here (the reference declares the calls at
node.py:317-333 but ships no implementation — SURVEY.md §3.4).
"""
from __future__ import annotations

from abc import ABC, abstractmethod
from typing import Any, Dict, Optional, Tuple

import numpy as np

from xotorch_amd.shard import Shard


class InferenceEngine(ABC):
  @abstractmethod
  async def encode(self, shard: Shard, prompt: str) -> np.ndarray:
    ...

  @abstractmethod
  async def decode(self, shard: Shard, tokens: np.ndarray) -> str:
    ...

  @abstractmethod
  async def sample(self, x: np.ndarray, temp: float = 0.0, top_k: int = 0, top_p: float = 0.0) -> np.ndarray:
    ...

  @abstractmethod
  async def infer_tensor(
    self, request_id: str, shard: Shard, input_data: np.ndarray, inference_state: Optional[dict] = None
  ) -> Tuple[np.ndarray, Optional[dict]]:
    ...

  async def infer_prompt(
    self, request_id: str, shard: Shard, prompt: str, inference_state: Optional[dict] = None
  ) -> Tuple[np.ndarray, Optional[dict]]:
    tokens = await self.encode(shard, prompt)
    x = tokens.reshape(1, -1)
    return await self.infer_tensor(request_id, shard, x, inference_state)

  async def load_checkpoint(self, shard: Shard, path: str):
    pass

  async def save_checkpoint(self, shard: Shard, path: str):
    pass

  async def train(
    self, request_id: str, shard: Shard, inputs, targets, lengths, loss: str = "ce", opt=None
  ):
    """Forward+backward on this shard; returns (loss_or_step, grad_wrt_input)."""
    raise NotImplementedError(f"{type(self).__name__} does not implement train")

  async def evaluate(self, request_id: str, shard: Shard, inputs, targets, lengths, loss: str = "ce"):
    raise NotImplementedError(f"{type(self).__name__} does not implement evaluate")

  async def ensure_shard(self, shard: Shard):
    pass

  async def clear_session(self, request_id: str):
    pass


def inference_engine_classes() -> Dict[str, str]:
  return {
    "torch": "TorchEngine",
    "hip": "HIPEngine",
    "dummy": "DummyEngine",
  }


def get_inference_engine(engine_name: str, shard_downloader=None) -> InferenceEngine:
  """Lazy factory mirroring the reference's (inference_engine.py:61-74)."""
  if engine_name in ("torch", "hip"):
    from xotorch_amd.engine.torch_engine import TorchEngine
    # "hip" selects the same engine with the HIP op path required (GPU);
    # "torch" allows eager fallback (CPU dev boxes).
    return TorchEngine(shard_downloader=shard_downloader, require_hip=(engine_name == "hip"))
  if engine_name == "dummy":
    from xotorch_amd.engine.dummy import DummyEngine
    return DummyEngine()
  raise ValueError(f"unknown inference engine: {engine_name}")
