"""Layer partitioning: topology → ordered float ranges → contiguous layer shards.

Behavioral parity with the reference's partitioning layer
(/root/reference/xotorch/topology/partitioning_strategy.py:11-42 and
ring_memory_weighted_partitioning_strategy.py:7-19), including the rounding
fix-ups its regression tests pin down: the last partition always covers
through layer n-1, empty integer ranges are skipped, and coverage gaps from
float rounding are repaired by extending the previous shard.

On an 8×MI355X node all GPUs expose equal HBM (288 GB), so memory-weighted
ranges degenerate to equal layer counts — but heterogeneous weights are kept
because the product surface (multi-node rings, mixed fleets) needs them.
"""
from __future__ import annotations

from abc import ABC, abstractmethod
from dataclasses import dataclass
from typing import List

from xotorch_amd.shard import Shard


@dataclass(frozen=True)
class Partition:
  node_id: str
  start: float  # inclusive, in [0, 1)
  end: float  # exclusive, in (0, 1]


class PartitioningStrategy(ABC):
  @abstractmethod
  def partition(self, topology) -> List[Partition]:
    ...


def map_partitions_to_shards(partitions: List[Partition], n_layers: int, model_id: str) -> List[Shard]:
  """Convert float ranges to contiguous layer shards covering [0, n_layers)."""
  shards: List[Shard] = []
  for i, p in enumerate(partitions):
    start_layer = int(p.start * n_layers)
    # Last partition always covers through the final layer regardless of
    # float rounding of its end point.
    end_layer = n_layers - 1 if i == len(partitions) - 1 else int(p.end * n_layers) - 1
    # Skip partitions that round to an empty integer range.
    if start_layer > end_layer:
      continue
    # Repair coverage gaps introduced by rounding: this shard must begin
    # exactly one past the previous shard's end.
    if shards and start_layer != shards[-1].end_layer + 1:
      start_layer = shards[-1].end_layer + 1
    if start_layer > end_layer:
      continue
    shards.append(Shard(model_id=model_id, start_layer=start_layer, end_layer=end_layer, n_layers=n_layers))

  # Full-coverage fixup: extend the final shard to n_layers-1 if rounding
  # left a tail uncovered.
  if shards and shards[-1].end_layer != n_layers - 1:
    last = shards[-1]
    shards[-1] = Shard(model_id=model_id, start_layer=last.start_layer, end_layer=n_layers - 1, n_layers=n_layers)
  return shards


class RingMemoryWeightedPartitioningStrategy(PartitioningStrategy):
  """Split [0,1) across ring members proportionally to each node's memory.

  Ordering is (memory desc, node_id asc) so the assignment is deterministic
  under gossip-ordering churn, matching the reference's strategy.
  """

  def partition(self, topology) -> List[Partition]:
    nodes = sorted(topology.all_nodes(), key=lambda x: (-x[1].memory, x[0]))
    total = sum(cap.memory for _, cap in nodes)
    partitions: List[Partition] = []
    start = 0.0
    for node_id, cap in nodes:
      end = round(start + (cap.memory / total), 5) if total > 0 else start
      partitions.append(Partition(node_id=node_id, start=start, end=end))
      start = end
    if partitions:
      # Guarantee exact coverage of [0,1] at the ring's end.
      last = partitions[-1]
      partitions[-1] = Partition(node_id=last.node_id, start=last.start, end=1.0)
    return partitions
