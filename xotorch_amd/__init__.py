"""xotorch_amd — MI355X-native peer-partitioned LLM inference & training runtime.

A from-scratch framework with the capabilities of shamantechnology/xotorch
(see SURVEY.md): ring memory-weighted layer partitioning, pipeline decode over
RCCL/xGMI, ChatGPT-compatible API, `xot` CLI, HF model registry + downloader,
and a LoRA train path — built MI355X-first: hand-written CDNA4 HIP kernels
(MFMA/LDS) for the per-token hot path, hipGraph-captured decode steps, and
torch.distributed (RCCL) collectives instead of gRPC serialization.
"""

__version__ = "0.1.0"

from xotorch_amd.shard import Shard  # noqa: F401
