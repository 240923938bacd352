import pytest

from xotorch_amd.shard import Shard


def test_shard_properties():
  s = Shard("m", 0, 3, 8)
  assert s.is_first_layer and not s.is_last_layer
  assert s.get_layer_count() == 4
  s2 = Shard("m", 4, 7, 8)
  assert s2.is_last_layer and not s2.is_first_layer
  assert not s.overlaps(s2)
  assert s.overlaps(Shard("m", 3, 5, 8))
  assert not s.overlaps(Shard("other", 0, 3, 8))


def test_shard_validation():
  with pytest.raises(ValueError):
    Shard("m", 3, 2, 8)
  with pytest.raises(ValueError):
    Shard("m", 0, 8, 8)
  with pytest.raises(ValueError):
    Shard("m", -1, 2, 8)


def test_shard_roundtrip():
  s = Shard("m", 2, 5, 10)
  assert Shard.from_dict(s.to_dict()) == s
