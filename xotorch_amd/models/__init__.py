from xotorch_amd.models.config import ModelConfig, config_from_hf  # noqa: F401
from xotorch_amd.models.registry import (  # noqa: F401
  model_cards,
  pretty_name,
  get_repo,
  build_base_shard,
  build_full_shard,
  get_supported_models,
)
