"""Pin the N=8 ring memory budget by arithmetic, not at bench time.

The driver's first real 8-GPU run must not faceplant on HBM: this test
reproduces exactly what RingPipeline allocates per rank for the headline
config (llama-3-70b bf16, B=128 per micro-batch, M=world micro-batches)
and asserts it fits one MI355X's 288 GB with headroom, for every ring
size the driver measures (1/2/4/8) at a generous generation budget.
"""
import pytest

from xotorch_amd.models.config import config_from_hf
from xotorch_amd.models.registry import builtin_config
from xotorch_amd.parallel.ring import equal_ring_shards

HBM_BYTES = 288 * 1024**3
BF16 = 2


def ring_rank_bytes(model_id: str, world: int, mb_batch: int, prompt_len: int,
                    max_gen: int, packed_attn: bool = True,
                    include_packed: bool = True) -> int:
  """Worst-case resident bytes on the largest rank of an N-way ring."""
  cfg = config_from_hf(builtin_config(model_id), model_id)
  shards = equal_ring_shards(model_id, cfg.n_layers, world)
  M = world  # micro-batches in flight (RingPipeline default)
  T = prompt_len + max_gen
  D, H, KVH, hd, I, V = (cfg.dim, cfg.n_heads, cfg.n_kv_heads, cfg.head_dim,
                         cfg.intermediate_dim, cfg.vocab_size)
  worst = 0
  for r, sh in enumerate(shards):
    L = sh.get_layer_count()
    # --- weights (bf16) ---
    per_layer = (
      D * (H + 2 * KVH) * hd    # fused qkv
      + H * hd * D              # o_proj
      + D * 2 * I               # fused gate|up
      + I * D                   # down_proj
      + 2 * D                   # the two norms
    )
    params = L * per_layer
    if sh.is_first_layer:
      params += V * D           # embedding
    if sh.is_last_layer:
      params += V * D + D       # lm_head + final norm
    weights = params * BF16
    # --- packed decode-GEMM copies (pack_decode_weights default groups:
    # down_proj, lm_head, gate_up — a second bf16 copy of each) ---
    packed = L * (I * D + D * 2 * I) * BF16
    if sh.is_last_layer:
      packed += V * D * BF16
    # --- KV caches: M micro-batches x 2 (K,V) x L layers, [B, KVH, T, hd];
    # the MFMA fragment-packed copies double residency ---
    kv = M * 2 * L * mb_batch * KVH * T * hd * BF16
    if packed_attn and hd == 128:
      kv *= 2
    # --- static I/O + position buffers (small) + prefill activation chunk
    # (bounded by XOT_PREFILL_CHUNK=128: chunk x S x 2I bf16 gate_up out) ---
    bufs = M * mb_batch * (D * BF16 + 8 + 8)
    prefill_act = min(mb_batch, 128) * prompt_len * 2 * I * BF16 * 2  # gate_up + slack
    total = (weights + kv + bufs + prefill_act) + (packed if include_packed else 0)
    worst = max(worst, total)
  return worst


@pytest.mark.parametrize("world", [1, 2, 4, 8])
def test_70b_ring_mandatory_fits_hbm(world):
  """MANDATORY residency (weights + dual-layout KV + buffers — everything
  RingPipeline allocates unconditionally) must fit. The packed decode-GEMM
  copies are applied GREEDILY under a free-memory guard
  (llama.py pack_decode_weights: mem_get_info + reserve check), so they can
  never OOM — they are asserted separately for N>=2 where they all fit."""
  # generation budget well beyond any driver run (steps+warmup <= 252)
  need = ring_rank_bytes("llama-3-70b", world, mb_batch=128, prompt_len=512,
                         max_gen=256, include_packed=False)
  # 85% ceiling: allocator fragmentation + RCCL buffers + hipGraph pools
  assert need < HBM_BYTES * 0.85, (
    f"world={world}: worst rank needs {need/2**30:.1f} GiB "
    f"> {HBM_BYTES*0.85/2**30:.1f} GiB budget"
  )


@pytest.mark.parametrize("world", [2, 4, 8])
def test_70b_ring_packed_fits_hbm_multi(world):
  """At N>=2 the full fast path (packed GEMM copies included) fits, so the
  driver's scaling runs get the packed kernels everywhere."""
  need = ring_rank_bytes("llama-3-70b", world, mb_batch=128, prompt_len=512,
                         max_gen=256, include_packed=True)
  assert need < HBM_BYTES * 0.85, f"world={world}: {need/2**30:.1f} GiB"


def test_70b_n1_fits_without_packed_weights():
  """N=1 holds ALL 80 layers: the greedy pack policy must have room to skip
  packing (weights+KV alone must fit), and the test documents that the
  pack groups are optional there."""
  cfg = config_from_hf(builtin_config("llama-3-70b"), "llama-3-70b")
  D, H, KVH, hd, I, V = (cfg.dim, cfg.n_heads, cfg.n_kv_heads, cfg.head_dim,
                         cfg.intermediate_dim, cfg.vocab_size)
  L = cfg.n_layers
  weights = (L * (D*(H+2*KVH)*hd + H*hd*D + D*2*I + I*D + 2*D) + 2*V*D + D) * BF16
  T = 512 + 256
  kv = 2 * L * 128 * KVH * T * hd * BF16 * 2  # M=1, dual-layout (MFMA) copies
  assert weights + kv < HBM_BYTES * 0.85


def test_8b_two_ranks_one_gpu_fits():
  """The 1-GPU lease verification runs 2 ranks of llama-3-8b on ONE device
  (gloo group): both ranks' residency must fit a single 288 GB GPU."""
  total = sum(
    ring_rank_bytes("llama-3-8b", 2, mb_batch=32, prompt_len=512, max_gen=128)
    for _ in range(1)
  ) * 2  # both ranks share the device; worst-rank x2 is an upper bound
  assert total < HBM_BYTES * 0.7


def test_405b_ring_fits_hbm():
  """llama-3.1-405b across 8 ranks: the 2.3 TB ring holds it at bf16
  (B=64 per micro-batch; B=128 exceeds the comfortable ceiling — use
  --mb-batch 64)."""
  need = ring_rank_bytes("llama-3.1-405b", 8, mb_batch=64, prompt_len=512,
                         max_gen=256, include_packed=False)
  assert need < HBM_BYTES * 0.85, f"{need/2**30:.1f} GiB"


@pytest.mark.parametrize("model_id", ["qwen-2.5-72b", "mistral-large", "nemotron-70b"])
def test_big_dense_cards_fit_one_gpu(model_id):
  """Every large dense card a user might `xot run` on one MI355X fits 288 GB
  at bf16 with the bench batch (B=64, 512+64 tokens) WITHOUT packed weights
  (pack_decode_weights backs off under memory pressure by design)."""
  if builtin_config(model_id) is None:
    pytest.skip(f"no builtin config for {model_id}")
  need = ring_rank_bytes(model_id, 1, 64, 512, 64, include_packed=False)
  assert need < HBM_BYTES * 0.92, f"{model_id}: {need/2**30:.0f} GiB"
