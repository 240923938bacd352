"""Ring serving worker test (world=1, CPU): one request through RingWorker."""
import torch

from xotorch_amd.serve_ring import RingWorker

TINY_ID = "dummy"  # builtin tiny llama config


def test_ring_worker_single():
  w = RingWorker(TINY_ID, rank=0, world=1, device="cpu", dtype=torch.float32)
  tokens = torch.randint(0, 200, (1, 7))
  got = []
  out = w.serve_request(tokens, max_new=5, temp=0.0, on_token=lambda t: got.append(t))
  assert len(out) >= 1 and got == out
  # greedy determinism: same prompt, fresh worker → same tokens
  w2 = RingWorker(TINY_ID, rank=0, world=1, device="cpu", dtype=torch.float32)
  out2 = w2.serve_request(tokens, max_new=5, temp=0.0)
  assert out == out2
