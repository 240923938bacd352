"""GPU ring-pipeline integration: hipGraph-captured decode steps through the
full HIP kernel stack (packed GEMMs, MFMA attention on the packed KV cache,
fused RoPE/append, sampling) must match the eager (no-graph) run exactly.
This is what the driver's round-end smoke + bench exercise; keeping it in
`-m gpu` catches capture regressions early."""
import pytest
import torch

pytestmark = pytest.mark.gpu

# tiny hd=128 config so the MFMA attention + packed-cache paths engage
TINY128 = {
  "model_type": "llama", "hidden_size": 256, "num_hidden_layers": 3,
  "num_attention_heads": 2, "num_key_value_heads": 1, "intermediate_size": 512,
  "vocab_size": 512, "rope_theta": 10000.0, "rms_norm_eps": 1e-5,
  "max_position_embeddings": 512, "torch_dtype": "bfloat16",
}


def _run(use_graphs: bool, pack_mode: str):
  import os
  from xotorch_amd.parallel.ring import RingPipeline
  os.environ["XOT_PACK"] = pack_mode
  try:
    ring = RingPipeline(model_id="tiny128", rank=0, world=1, device="cuda",
                        dtype=torch.bfloat16, mb_batch=64, prompt_len=96, max_gen=12,
                        use_graphs=use_graphs, seed=77, cfg_override=TINY128)
    ring.capture_tokens = True
    ring.prefill()
    for _ in range(8):
      ring.decode_step()
    torch.cuda.synchronize()
    return torch.cat(ring.generated[0], dim=1).cpu()
  finally:
    os.environ.pop("XOT_PACK", None)


def test_graphed_decode_matches_eager():
  toks_eager = _run(use_graphs=False, pack_mode="none")
  toks_graph = _run(use_graphs=True, pack_mode="none")
  assert torch.equal(toks_eager, toks_graph)


def test_packed_gemms_match_unpacked():
  toks_unpacked = _run(use_graphs=True, pack_mode="none")
  toks_packed = _run(use_graphs=True, pack_mode="all")
  # packed MFMA GEMMs vs hipBLASLt: identical greedy tokens expected on a
  # 96-token prompt at these scales (both fp32-accumulate)
  agree = (toks_unpacked == toks_packed).float().mean().item()
  assert agree > 0.95, f"token agreement {agree}"


def test_mfma_attention_cache_consistency():
  """VALU path (XOT_MFMA_ATTN=0) vs MFMA path must agree."""
  import os
  os.environ["XOT_MFMA_ATTN"] = "0"
  try:
    toks_valu = _run(use_graphs=True, pack_mode="none")
  finally:
    os.environ["XOT_MFMA_ATTN"] = "1"
  toks_mfma = _run(use_graphs=True, pack_mode="none")
  agree = (toks_valu == toks_mfma).float().mean().item()
  assert agree > 0.95, f"token agreement {agree}"
