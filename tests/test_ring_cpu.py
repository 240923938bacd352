"""Ring-pipeline correctness over real torch.distributed transport (gloo, CPU):
2-stage ring decode must produce exactly the tokens of the single-process run
(the reference's split-vs-full oracle, run over the wire)."""
import json
import os
import sys
import tempfile
from pathlib import Path

import pytest
import torch
import torch.multiprocessing as mp

TINY = {
  "model_type": "llama", "hidden_size": 64, "num_hidden_layers": 4,
  "num_attention_heads": 4, "num_key_value_heads": 2, "intermediate_size": 128,
  "vocab_size": 211, "rope_theta": 10000.0, "rms_norm_eps": 1e-5,
  "max_position_embeddings": 128, "torch_dtype": "float32",
}
STEPS = 6


def run_ring(world: int, out_dir: str, port: int):
  """Run a ring decode; last stage dumps its generated tokens to a file."""
  from xotorch_amd.parallel.ring import RingPipeline
  rank = int(os.environ.get("RANK", "0"))
  ring = RingPipeline(
    model_id="tinytest", rank=rank, world=world, device="cpu", dtype=torch.float32,
    mb_batch=2, n_microbatches=2, prompt_len=12, max_gen=STEPS + 2,
    use_graphs=False, cfg_override=TINY,
  )
  ring.capture_tokens = True
  ring.prefill()
  for _ in range(STEPS):
    ring.decode_step()
  ring.finish()
  if ring.is_last:
    toks = [[t.tolist() for t in mbtoks] for mbtoks in ring.generated]
    Path(out_dir, f"tokens_w{world}.json").write_text(json.dumps(toks))


def _worker(rank, world, out_dir, port):
  os.environ["RANK"] = str(rank)
  os.environ["WORLD_SIZE"] = str(world)
  os.environ["MASTER_ADDR"] = "127.0.0.1"
  os.environ["MASTER_PORT"] = str(port)
  import torch.distributed as dist
  dist.init_process_group("gloo", rank=rank, world_size=world)
  try:
    run_ring(world, out_dir, port)
  finally:
    dist.destroy_process_group()


@pytest.mark.timeout(240)
def test_two_stage_ring_matches_single():
  from xotorch_amd.helpers import find_available_port
  with tempfile.TemporaryDirectory() as d:
    # single process reference
    os.environ.pop("RANK", None)
    os.environ.pop("WORLD_SIZE", None)
    run_ring(1, d, 0)
    # two-stage ring over gloo
    port = find_available_port("127.0.0.1")
    mp.spawn(_worker, args=(2, d, port), nprocs=2, join=True)
    ref = json.loads(Path(d, "tokens_w1.json").read_text())
    two = json.loads(Path(d, "tokens_w2.json").read_text())
    assert ref == two, "2-stage ring tokens diverged from single-process decode"
    # sanity: we actually generated STEPS+1 tokens per micro-batch
    assert len(ref[0]) == STEPS + 1


def test_chunked_prefill_matches_unchunked(tmp_path, monkeypatch):
  """Batch-chunked prefill (bounded activation peak) must produce the same
  tokens as single-shot prefill."""
  import torch
  from xotorch_amd.parallel.ring import RingPipeline

  def run(chunk):
    if chunk:
      monkeypatch.setenv("XOT_PREFILL_CHUNK", "4")
    else:
      monkeypatch.delenv("XOT_PREFILL_CHUNK", raising=False)
    ring = RingPipeline(model_id="dummy", rank=0, world=1, device="cpu",
                        dtype=torch.float32, mb_batch=12, prompt_len=48, max_gen=8,
                        use_graphs=False, seed=7)
    ring.capture_tokens = True
    ring.prefill()
    for _ in range(6):
      ring.decode_step()
    return torch.cat([t for t in ring.generated[0]], dim=1)

  toks_chunked = run(True)
  toks_full = run(False)
  assert torch.equal(toks_chunked, toks_full)


@pytest.mark.timeout(360)
def test_four_stage_ring_matches_single():
  """4-stage pipeline (the multi-GPU shape the driver runs at N=4/8): four
  micro-batches in flight, token wrap-around across 3 hops."""
  from xotorch_amd.helpers import find_available_port
  with tempfile.TemporaryDirectory() as d:
    os.environ.pop("RANK", None)
    os.environ.pop("WORLD_SIZE", None)
    run_ring(1, d, 0)
    port = find_available_port("127.0.0.1")
    mp.spawn(_worker, args=(4, d, port), nprocs=4, join=True)
    ref = json.loads(Path(d, "tokens_w1.json").read_text())
    four = json.loads(Path(d, "tokens_w4.json").read_text())
    # world=1 runs M=1 micro-batch; world=4 runs M=4 — compare the shared mb 0
    assert ref[0] == four[0], "4-stage ring tokens diverged from single-process decode"


@pytest.mark.timeout(240)
def test_seq_chunked_prefill_matches(monkeypatch):
  """Sequence-chunked multi-rank prefill produces identical tokens."""
  from xotorch_amd.helpers import find_available_port
  with tempfile.TemporaryDirectory() as d:
    os.environ.pop("RANK", None)
    os.environ.pop("WORLD_SIZE", None)
    run_ring(1, d, 0)
    os.environ["XOT_RING_PREFILL_SEQ_CHUNK"] = "5"  # 12-token prompt -> 3 chunks
    try:
      port = find_available_port("127.0.0.1")
      mp.spawn(_worker, args=(2, d, port), nprocs=2, join=True)
    finally:
      os.environ.pop("XOT_RING_PREFILL_SEQ_CHUNK", None)
    ref = json.loads(Path(d, "tokens_w1.json").read_text())
    two = json.loads(Path(d, "tokens_w2.json").read_text())
    assert ref == two, "seq-chunked prefill diverged"
