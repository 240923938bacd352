from xotorch_amd.engine.interface import InferenceEngine, get_inference_engine  # noqa: F401
from xotorch_amd.engine.state import ShardInferenceState  # noqa: F401
