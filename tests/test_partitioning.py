"""Partition-math tests, including the rounding regressions the reference pins
(/root/reference/xotorch/topology/test_map_partitions.py)."""
import pytest

from xotorch_amd.parallel.partitioning import (
  Partition,
  RingMemoryWeightedPartitioningStrategy,
  map_partitions_to_shards,
)
from xotorch_amd.parallel.topology import DeviceCapabilities, DeviceFlops, Topology


def caps(mem):
  return DeviceCapabilities(model="m", chip="c", memory=mem, flops=DeviceFlops())


def test_map_partitions_exact_split():
  parts = [Partition("a", 0.0, 0.5), Partition("b", 0.5, 1.0)]
  shards = map_partitions_to_shards(parts, 32, "m")
  assert [(s.start_layer, s.end_layer) for s in shards] == [(0, 15), (16, 31)]
  assert all(s.n_layers == 32 for s in shards)


def test_map_partitions_uneven():
  parts = [Partition("a", 0.0, 0.42857), Partition("b", 0.42857, 0.71428), Partition("c", 0.71428, 1.0)]
  shards = map_partitions_to_shards(parts, 32, "m")
  # contiguous full coverage regardless of rounding
  assert shards[0].start_layer == 0
  assert shards[-1].end_layer == 31
  for prev, nxt in zip(shards, shards[1:]):
    assert nxt.start_layer == prev.end_layer + 1


@pytest.mark.parametrize("n_layers", [1, 2, 3, 5, 16, 32, 80, 126])
@pytest.mark.parametrize("weights", [[1], [1, 1], [3, 1], [1, 2, 3], [1, 1, 1, 1, 1, 1, 1, 1]])
def test_full_coverage_all_splits(n_layers, weights):
  total = sum(weights)
  parts = []
  start = 0.0
  for i, w in enumerate(weights):
    end = round(start + w / total, 5)
    parts.append(Partition(f"n{i}", start, end))
    start = end
  parts[-1] = Partition(parts[-1].node_id, parts[-1].start, 1.0)
  shards = map_partitions_to_shards(parts, n_layers, "m")
  assert shards[0].start_layer == 0
  assert shards[-1].end_layer == n_layers - 1
  covered = sum(s.get_layer_count() for s in shards)
  assert covered == n_layers
  for prev, nxt in zip(shards, shards[1:]):
    assert nxt.start_layer == prev.end_layer + 1


def test_ring_memory_weighted_ordering():
  t = Topology()
  t.update_node("small", caps(1000))
  t.update_node("big", caps(3000))
  strat = RingMemoryWeightedPartitioningStrategy()
  parts = strat.partition(t)
  assert parts[0].node_id == "big"
  assert abs((parts[0].end - parts[0].start) - 0.75) < 1e-4
  assert parts[-1].end == 1.0


def test_ring_memory_weighted_id_tiebreak():
  t = Topology()
  for nid in ["c", "a", "b"]:
    t.update_node(nid, caps(100))
  parts = RingMemoryWeightedPartitioningStrategy().partition(t)
  assert [p.node_id for p in parts] == ["a", "b", "c"]


def test_equal_memory_eight_gpus_equal_layers():
  t = Topology()
  for i in range(8):
    t.update_node(f"gpu{i}", caps(294912))  # 288 GB MI355X
  parts = RingMemoryWeightedPartitioningStrategy().partition(t)
  shards = map_partitions_to_shards(parts, 80, "llama-3-70b")
  assert [s.get_layer_count() for s in shards] == [10] * 8


def test_empty_partition_skipped():
  # middle partition too small to round to one layer of 4
  parts = [Partition("a", 0.0, 0.5), Partition("b", 0.5, 0.55), Partition("c", 0.55, 1.0)]
  shards = map_partitions_to_shards(parts, 4, "m")
  assert sum(s.get_layer_count() for s in shards) == 4
  assert shards[0].start_layer == 0 and shards[-1].end_layer == 3


def test_full_coverage_property_fuzz():
  """Property fuzz over arbitrary positive memory weights and layer counts:
  shards must always tile [0, n_layers) exactly (the reference's rounding
  regressions lived exactly here)."""
  from hypothesis import given, settings, strategies as st

  @settings(max_examples=300, deadline=None)
  @given(
    n_layers=st.integers(min_value=1, max_value=200),
    weights=st.lists(st.floats(min_value=0.01, max_value=1000.0,
                               allow_nan=False, allow_infinity=False),
                     min_size=1, max_size=12),
  )
  def check(n_layers, weights):
    total = sum(weights)
    parts, start = [], 0.0
    for i, w in enumerate(weights):
      end = round(start + w / total, 5)
      parts.append(Partition(f"n{i}", start, end))
      start = end
    parts[-1] = Partition(parts[-1].node_id, parts[-1].start, 1.0)
    shards = map_partitions_to_shards(parts, n_layers, "m")
    assert shards[0].start_layer == 0
    assert shards[-1].end_layer == n_layers - 1
    assert sum(s.get_layer_count() for s in shards) == n_layers
    for prev, nxt in zip(shards, shards[1:]):
      assert nxt.start_layer == prev.end_layer + 1

  check()


def test_equal_ring_shards_property_fuzz():
  """equal_ring_shards tiles [0, n_layers) contiguously with sizes differing
  by at most 1 (balanced pipeline stages) for every world size."""
  from hypothesis import given, settings, strategies as st
  from xotorch_amd.parallel.ring import equal_ring_shards

  @settings(max_examples=200, deadline=None)
  @given(n_layers=st.integers(min_value=1, max_value=200),
         world=st.integers(min_value=1, max_value=16))
  def check(n_layers, world):
    if world > n_layers:
      return  # not enough layers for every stage
    shards = equal_ring_shards("m", n_layers, world)
    assert len(shards) == world
    assert shards[0].start_layer == 0
    assert shards[-1].end_layer == n_layers - 1
    for a, b in zip(shards, shards[1:]):
      assert b.start_layer == a.end_layer + 1
    sizes = [s.get_layer_count() for s in shards]
    assert max(sizes) - min(sizes) <= 1

  check()
