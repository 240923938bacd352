from xotorch_amd.models.config import ModelConfig, config_from_hf  # noqa: F401
from xotorch_amd.models.registry import (  # noqa: F401
  model_cards,
  pretty_name,
  get_repo,
  build_base_shard,
  build_full_shard,
  get_supported_models,
)


def model_class_for(cfg: ModelConfig):
  """Decoder class for a parsed config (ShardedModel / Gemma2 / DeepSeek-MLA)."""
  if cfg.model_type == "gemma2":
    from xotorch_amd.models.gemma2 import Gemma2Model
    return Gemma2Model
  if cfg.model_type in ("deepseek_v3", "deepseek_v2"):
    from xotorch_amd.models.deepseek_v3 import DeepseekV3Model
    return DeepseekV3Model
  from xotorch_amd.models.llama import ShardedModel
  return ShardedModel
