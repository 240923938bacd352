"""Shard — a contiguous transformer-layer range of one model on one worker.

Capability parity with the reference's frozen dataclass
(/root/reference/xotorch/inference/shard.py:5-40).
"""
from __future__ import annotations

from dataclasses import dataclass, field


@dataclass(frozen=True)
class Shard:
  model_id: str
  start_layer: int
  end_layer: int  # inclusive
  n_layers: int

  def __post_init__(self):
    if self.n_layers <= 0:
      raise ValueError(f"n_layers must be positive, got {self.n_layers}")
    if not (0 <= self.start_layer <= self.end_layer < self.n_layers):
      raise ValueError(
        f"invalid shard range [{self.start_layer}, {self.end_layer}] for n_layers={self.n_layers}"
      )

  @property
  def is_first_layer(self) -> bool:
    return self.start_layer == 0

  @property
  def is_last_layer(self) -> bool:
    return self.end_layer == self.n_layers - 1

  def get_layer_count(self) -> int:
    return self.end_layer - self.start_layer + 1

  def overlaps(self, other: "Shard") -> bool:
    return (
      self.model_id == other.model_id
      and max(self.start_layer, other.start_layer) <= min(self.end_layer, other.end_layer)
    )

  def to_dict(self) -> dict:
    return {
      "model_id": self.model_id,
      "start_layer": self.start_layer,
      "end_layer": self.end_layer,
      "n_layers": self.n_layers,
    }

  @classmethod
  def from_dict(cls, d: dict) -> "Shard":
    return cls(
      model_id=d["model_id"],
      start_layer=int(d["start_layer"]),
      end_layer=int(d["end_layer"]),
      n_layers=int(d["n_layers"]),
    )
