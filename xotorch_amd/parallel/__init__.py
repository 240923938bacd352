from xotorch_amd.parallel.partitioning import (  # noqa: F401
  Partition,
  PartitioningStrategy,
  RingMemoryWeightedPartitioningStrategy,
  map_partitions_to_shards,
)
