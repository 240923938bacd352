"""Speculative decoding: draft-model proposal + single-pass target verify.

Beyond the reference (its engine decodes strictly one token per full-model
pass): a small draft model proposes gamma tokens autoregressively, the
target model scores all of them in ONE chunked forward (prefill-style, so
the target's per-token cost amortizes towards its weight-stream floor), and
the longest agreeing prefix is accepted plus one corrected token.

Greedy (temperature 0) acceptance: the output is PROVABLY IDENTICAL to
running the target alone — the draft only changes latency, never content
(pinned by tests/test_spec_cpu.py with a deliberately mismatched draft).

KV bookkeeping uses position rewind: verify writes the target cache at
positions [P, P+gamma]; on acceptance of m tokens the cache is truthful
through P+m (rejected suffix positions are simply overwritten by the next
round before they are ever attended to, because attention reads [0, seq_len)
with seq_len = accepted length). The draft cache rewinds the same way.

Registry pairs (llama-3.2-1b draft for llama-3-8b/70b, qwen 0.5b for qwen
7b+) share tokenizers, which is the only compatibility requirement.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import List, Optional, Tuple

import torch

from xotorch_amd.engine.kvcache import ShardKVCache
from xotorch_amd.models import model_class_for
from xotorch_amd.models.config import ModelConfig, config_from_hf
from xotorch_amd.models.registry import builtin_config
from xotorch_amd.shard import Shard


@dataclass
class SpecStats:
  proposed: int = 0
  accepted: int = 0
  rounds: int = 0

  @property
  def accept_rate(self) -> float:
    return self.accepted / max(1, self.proposed)


class SpeculativeDecoder:
  """Greedy speculative decoding over two full (unsharded) models."""

  def __init__(self, target: torch.nn.Module, draft: torch.nn.Module,
               target_cfg: ModelConfig, draft_cfg: ModelConfig,
               device: str = "cuda", dtype: torch.dtype = torch.bfloat16,
               gamma: int = 4, max_seq: int = 2048):
    self.target, self.draft = target, draft
    self.tc, self.dc = target_cfg, draft_cfg
    self.device, self.dtype = device, dtype
    self.gamma = gamma
    self.max_seq = min(max_seq, target_cfg.max_seq_len, draft_cfg.max_seq_len)
    th, tk, tv = target_cfg.kv_cache_dims()
    dh, dk, dv = draft_cfg.kv_cache_dims()
    self.t_cache = ShardKVCache(target_cfg.n_layers, 1, th, self.max_seq, tk, dtype, device, v_dim=tv)
    self.d_cache = ShardKVCache(draft_cfg.n_layers, 1, dh, self.max_seq, dk, dtype, device, v_dim=dv)

  @classmethod
  def from_model_ids(cls, target_id: str, draft_id: str, device="cuda",
                     dtype=torch.bfloat16, gamma: int = 4, seed: int = 1234, **kw):
    from xotorch_amd.models.weights import fast_random_init_gpu, random_init
    models = []
    cfgs = []
    for mid in (target_id, draft_id):
      cfg = config_from_hf(builtin_config(mid), mid)
      shard = Shard(mid, 0, cfg.n_layers - 1, cfg.n_layers)
      cls_ = model_class_for(cfg)
      prev = torch.get_default_dtype()
      torch.set_default_dtype(dtype)
      try:
        with torch.device("meta"):
          m = cls_(cfg, shard)
      finally:
        torch.set_default_dtype(prev)
      m = m.to_empty(device=device).to(dtype)
      if device == "cuda" and cfg.dim >= 2048:
        fast_random_init_gpu(m, seed)
      else:
        random_init(m, seed)
      m.reset_rope()
      m.eval()
      if device == "cuda":
        m.pack_decode_weights(reserve_bytes=8 << 30)
      models.append(m)
      cfgs.append(cfg)
    return cls(models[0], models[1], cfgs[0], cfgs[1], device, dtype, gamma, **kw)

  def _forward(self, model, cache, tokens: torch.Tensor, pos0: int) -> torch.Tensor:
    """Run `tokens` [1, S] at positions pos0..pos0+S-1; returns logits
    [1, S, V] (all positions)."""
    S = tokens.shape[1]
    pos = torch.arange(pos0, pos0 + S, dtype=torch.int32, device=self.device)
    out = model(tokens, caches=cache.caches, positions=pos, start_pos=pos0,
                last_only=False)
    if out.dim() == 2:
      out = out.unsqueeze(1)
    return out

  @torch.inference_mode()
  def generate(self, prompt: torch.Tensor, max_new: int,
               eos_id: Optional[int] = None) -> Tuple[List[int], SpecStats]:
    """prompt: [1, S] int64. Greedy. Returns (tokens, stats)."""
    dev = self.device
    prompt = prompt.to(dev)
    S = prompt.shape[1]
    stats = SpecStats()
    # prefill both models; t_cur = target's first prediction
    t_logits = self._forward(self.target, self.t_cache, prompt, 0)
    self._forward(self.draft, self.d_cache, prompt, 0)
    t_cur = int(t_logits[0, -1].argmax())
    out: List[int] = [t_cur]
    P = S          # both caches truthful through position P-1; t_cur sits at P (unfed)
    d_next = S     # first cache row the DRAFT has not yet written
    while len(out) < max_new and not (eos_id is not None and t_cur == eos_id):
      g = min(self.gamma, self.max_seq - P - 2)
      if g <= 0:
        break
      # ---- draft proposes g tokens ----
      d_toks: List[int] = []
      cur = t_cur
      dp = P
      for i in range(g):
        if i == 0 and dp > d_next:
          # after a fully-accepted round the draft never saw its own last
          # proposal: feed the accepted-but-unfed positions in one chunk so
          # the draft cache has no zero-KV hole (hole -> proposals diverge
          # from the target forever and acceptance collapses)
          toks = [out[p - S] for p in range(d_next, dp)] + [cur]
          dl = self._forward(self.draft, self.d_cache,
                             torch.tensor([toks], dtype=torch.int64, device=dev), d_next)
        else:
          dl = self._forward(self.draft, self.d_cache,
                             torch.tensor([[cur]], dtype=torch.int64, device=dev), dp)
        d_next = dp + 1
        cur = int(dl[0, -1].argmax())
        d_toks.append(cur)
        dp += 1
      # ---- target verifies in one chunk ----
      chunk = torch.tensor([[t_cur] + d_toks], dtype=torch.int64, device=dev)
      tl = self._forward(self.target, self.t_cache, chunk, P)  # [1, g+1, V]
      greedy = tl[0].argmax(dim=-1).tolist()                   # predictions for P+1..P+g+1
      m = 0
      while m < g and greedy[m] == d_toks[m]:
        m += 1
      stats.proposed += g
      stats.accepted += m
      stats.rounds += 1
      emitted = d_toks[:m] + [greedy[m]]
      for t in emitted:
        out.append(t)
        if len(out) >= max_new or (eos_id is not None and t == eos_id):
          break
      t_cur = out[-1]
      P = P + m + 1
      # draft cache: positions P.. hold stale tokens; truthful through the
      # accepted prefix, so continuing from P with t_cur overwrites them
    return out[:max_new], stats

  @torch.inference_mode()
  def generate_plain(self, prompt: torch.Tensor, max_new: int,
                     eos_id: Optional[int] = None) -> List[int]:
    """Target-only greedy (the correctness oracle / baseline timing)."""
    dev = self.device
    self.t_cache.reset()
    prompt = prompt.to(dev)
    S = prompt.shape[1]
    logits = self._forward(self.target, self.t_cache, prompt, 0)
    cur = int(logits[0, -1].argmax())
    out = [cur]
    P = S
    while len(out) < max_new and not (eos_id is not None and cur == eos_id):
      if P + 1 >= self.max_seq:
        break
      tl = self._forward(self.target, self.t_cache,
                         torch.tensor([[cur]], dtype=torch.int64, device=dev), P)
      cur = int(tl[0, -1].argmax())
      out.append(cur)
      P += 1
    return out[:max_new]

  def reset(self):
    self.t_cache.reset()
    self.d_cache.reset()
