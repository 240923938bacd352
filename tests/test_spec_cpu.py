"""Speculative decoding invariants (CPU, tiny models).

The load-bearing property: greedy speculative output is EXACTLY the
target-only greedy output, for ANY draft — a good draft only changes how
many tokens each round accepts, never what is generated.
"""
import numpy as np
import torch

from xotorch_amd.engine.spec import SpeculativeDecoder
from xotorch_amd.models.config import config_from_hf
from xotorch_amd.models.llama import ShardedModel
from xotorch_amd.models.registry import builtin_config
from xotorch_amd.models.weights import random_init
from xotorch_amd.shard import Shard


def _build(seed):
  cfg = config_from_hf(builtin_config("dummy"), "dummy")
  shard = Shard("dummy", 0, cfg.n_layers - 1, cfg.n_layers)
  m = ShardedModel(cfg, shard).float()
  random_init(m, seed=seed)
  m.eval()
  return m, cfg


def _spec(target_seed, draft_seed, gamma=4):
  t, tc = _build(target_seed)
  d, dc = _build(draft_seed)
  return SpeculativeDecoder(t, d, tc, dc, device="cpu", dtype=torch.float32,
                            gamma=gamma, max_seq=256)


def test_identical_draft_accepts_everything():
  sd = _spec(7, 7)
  prompt = torch.randint(0, 200, (1, 12))
  toks, stats = sd.generate(prompt, max_new=24)
  assert stats.accept_rate == 1.0, stats
  ref = sd.generate_plain(prompt, max_new=24)
  assert toks == ref


def test_mismatched_draft_output_identical_to_target():
  for gamma in (1, 3, 5):
    sd = _spec(7, 99, gamma=gamma)
    prompt = torch.randint(0, 200, (1, 9))
    toks, stats = sd.generate(prompt, max_new=30)
    sd.reset()
    ref = sd.generate_plain(prompt, max_new=30)
    assert toks == ref, (gamma, toks, ref)
    assert stats.proposed > 0


def test_eos_stops_generation():
  sd = _spec(3, 3)
  prompt = torch.randint(0, 200, (1, 6))
  full = sd.generate_plain(prompt, max_new=40)
  if len(set(full)) < 2:
    return  # degenerate random model; nothing to cut on
  eos = full[len(full) // 2]
  sd.reset()
  toks, _ = sd.generate(prompt, max_new=40, eos_id=eos)
  assert toks == full[: full.index(eos) + 1]
