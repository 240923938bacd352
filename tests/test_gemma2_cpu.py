"""Gemma2 decoder vs the transformers Gemma2ForCausalLM oracle (random
weights, CPU fp32): prefill logits, cached decode step, and shard splitting.
The reference lists gemma2 cards but cannot run the architecture
(general_mha.py has no softcap / sliding window / sandwich norms)."""
import pytest
import torch

TINY = dict(vocab_size=101, hidden_size=64, intermediate_size=128, num_hidden_layers=4,
            num_attention_heads=4, num_key_value_heads=2, head_dim=16,
            sliding_window=8, attn_logit_softcapping=50.0, final_logit_softcapping=30.0,
            query_pre_attn_scalar=16, rms_norm_eps=1e-6, rope_theta=10000.0,
            max_position_embeddings=64, tie_word_embeddings=True)


def _build_pair(seed=0):
  from transformers import Gemma2Config, Gemma2ForCausalLM
  from xotorch_amd.models.config import config_from_hf
  from xotorch_amd.models.gemma2 import Gemma2Model, hf_key_map_gemma2
  from xotorch_amd.shard import Shard
  torch.manual_seed(seed)
  ref = Gemma2ForCausalLM(Gemma2Config(**TINY)).eval().float()
  raw = dict(TINY)
  raw["model_type"] = "gemma2"
  cfg = config_from_hf(raw, "gemma2-tiny")
  shard = Shard("gemma2-tiny", 0, cfg.n_layers - 1, cfg.n_layers)
  ours = Gemma2Model(cfg, shard).float()
  from xotorch_amd.models.weights import remap_hf_state
  sd = ref.state_dict()
  mapping = hf_key_map_gemma2(shard, cfg)
  ours.load_state_dict(remap_hf_state(sd, mapping), strict=False)
  ours.eval()
  return ref, ours, cfg, shard


def _caches(cfg, n_layers, B, T):
  from xotorch_amd.engine.kvcache import ShardKVCache
  return ShardKVCache(n_layers, B, cfg.n_kv_heads, T, cfg.head_dim, torch.float32, "cpu")


def test_gemma2_matches_transformers_prefill_and_decode():
  ref, ours, cfg, shard = _build_pair()
  B, S = 2, 12
  x = torch.randint(0, cfg.vocab_size, (B, S))
  cache = _caches(cfg, cfg.n_layers, B, S + 4)
  with torch.no_grad():
    lref = ref(x).logits
    lours = ours(x, caches=cache.caches, positions=torch.arange(S), start_pos=0,
                 last_only=False)
    assert torch.allclose(lours, lref, atol=2e-4, rtol=2e-4), (lours - lref).abs().max()
    # one cached decode step vs the oracle re-running the full sequence
    nxt = lref[:, -1].argmax(-1, keepdim=True)
    lref2 = ref(torch.cat([x, nxt], dim=1)).logits[:, -1]
    lours2 = ours(nxt, caches=cache.caches, positions=torch.tensor([S]), start_pos=S,
                  is_decode=True)
    assert torch.allclose(lours2, lref2, atol=2e-4, rtol=2e-4), (lours2 - lref2).abs().max()


def test_gemma2_split_equals_full():
  ref, ours, cfg, shard = _build_pair(seed=1)
  from xotorch_amd.models.gemma2 import Gemma2Model, hf_key_map_gemma2
  from xotorch_amd.shard import Shard
  B, S = 2, 9
  x = torch.randint(0, cfg.vocab_size, (B, S))
  cache = _caches(cfg, cfg.n_layers, B, S + 2)
  with torch.no_grad():
    full = ours(x, caches=cache.caches, positions=torch.arange(S), start_pos=0, last_only=False)
    s0, s1 = Shard("gemma2-tiny", 0, 1, 4), Shard("gemma2-tiny", 2, 3, 4)
    sd = ref.state_dict()
    m0 = Gemma2Model(cfg, s0).float()
    from xotorch_amd.models.weights import remap_hf_state
    m0.load_state_dict(remap_hf_state(sd, hf_key_map_gemma2(s0, cfg)), strict=False)
    m1 = Gemma2Model(cfg, s1).float()
    m1.load_state_dict(remap_hf_state(sd, hf_key_map_gemma2(s1, cfg)), strict=False)
    m0.eval(); m1.eval()
    c0, c1 = _caches(cfg, 2, B, S + 2), _caches(cfg, 2, B, S + 2)
    h = m0(x, caches=c0.caches, positions=torch.arange(S), start_pos=0)
    split = m1(h, caches=c1.caches, positions=torch.arange(S), start_pos=0, last_only=False)
    assert torch.allclose(full, split, atol=1e-5), (full - split).abs().max()


def test_gemma2_through_engine_and_node(monkeypatch):
  """Serving path: TorchEngine routes gemma2 to the Gemma2 decoder; a node
  decode loop completes."""
  import asyncio
  from xotorch_amd.models import registry

  tiny = dict(TINY)
  tiny["model_type"] = "gemma2"
  tiny["vocab_size"] = 256  # DummyTokenizer emits ids up to 200
  monkeypatch.setitem(registry.BUILTIN_CONFIGS, "gemma2-tiny", tiny)
  monkeypatch.setitem(registry.model_cards, "gemma2-tiny",
                      {"layers": 4, "repo": {"TorchEngine": "dummy"}})

  async def go():
    from xotorch_amd.engine.torch_engine import TorchEngine
    from xotorch_amd.models.registry import build_base_shard
    from xotorch_amd.orchestration.node import Node
    eng = TorchEngine(device="cpu", dtype=torch.float32)
    node = Node("g2-test", None, eng, None, max_generate_tokens=6)
    await node.start(0)
    try:
      shard = build_base_shard("gemma2-tiny", "TorchEngine")
      done = asyncio.Event()
      got = []
      def on_token(rid, toks, fin):
        got.extend(toks)
        if fin:
          done.set()
      node.on_token.register("g2").on_next(on_token)
      await node.process_prompt(shard, "hello gemma", "g2-req")
      await asyncio.wait_for(done.wait(), 90)
      from xotorch_amd.models.gemma2 import Gemma2Model
      assert isinstance(eng.model, Gemma2Model)
      assert 0 < len(got) <= 6
    finally:
      await node.stop()
  asyncio.run(go())


def test_gemma2_two_node_tcp_ring(monkeypatch, tmp_path):
  """Gemma2 over the ring: two nodes split the layers over the TCP wire."""
  import asyncio
  from xotorch_amd.models import registry
  from tests.test_node_cpu import _make_tcp_pair, run

  tiny = dict(TINY)
  tiny["model_type"] = "gemma2"
  tiny["vocab_size"] = 256
  monkeypatch.setitem(registry.BUILTIN_CONFIGS, "gemma2-tiny2", tiny)
  monkeypatch.setitem(registry.model_cards, "gemma2-tiny2",
                      {"layers": 4, "repo": {"TorchEngine": "dummy"}})

  async def go():
    import torch
    from xotorch_amd.engine.torch_engine import TorchEngine
    from xotorch_amd.models.registry import build_base_shard
    nodes = await _make_tcp_pair(tmp_path, lambda: TorchEngine(device="cpu", dtype=torch.float32))
    a, b = nodes
    shard = build_base_shard("gemma2-tiny2", "TorchEngine")
    done = asyncio.Event()
    got = []
    def on_token(rid, toks, fin):
      got[:] = toks
      if fin:
        done.set()
    a.on_token.register("g2-t").on_next(on_token)
    await a.process_prompt(shard, "softcapped ring", "g2-tcp")
    await asyncio.wait_for(done.wait(), 120)
    assert len(got) >= 1
    from xotorch_amd.models.gemma2 import Gemma2Model
    assert isinstance(a.inference_engine.model, Gemma2Model)
    assert isinstance(b.inference_engine.model, Gemma2Model)
    for n in nodes:
      await n.stop()
    return True
  assert run(go())


def test_gemma2_per_row_positions():
  """Ragged per-slot decode equals per-row sequential decode (eager path)."""
  import torch
  from xotorch_amd.engine.kvcache import ShardKVCache
  from xotorch_amd.models.config import config_from_hf
  from xotorch_amd.models.gemma2 import Gemma2Model
  from xotorch_amd.models.weights import random_init
  from xotorch_amd.shard import Shard
  raw = dict(TINY)
  raw["model_type"] = "gemma2"
  cfg = config_from_hf(raw, "gemma2-tiny")
  shard = Shard("gemma2-tiny", 0, cfg.n_layers - 1, cfg.n_layers)
  torch.manual_seed(2)
  m = Gemma2Model(cfg, shard).float()
  random_init(m, std=0.15)
  m.eval()
  B, lens = 3, [4, 8, 6]
  cache = ShardKVCache(cfg.n_layers, B, cfg.n_kv_heads, 32, cfg.head_dim, torch.float32, "cpu")
  toks = torch.randint(0, cfg.vocab_size, (B, 10))
  with torch.inference_mode():
    for b, L in enumerate(lens):
      sliced = [tuple(t[b:b + 1] if t is not None else None for t in layer) for layer in cache.caches]
      m(toks[b:b + 1, :L], caches=sliced, positions=torch.arange(L), start_pos=0)
    nxt = toks[:, :1]
    refs = []
    for b, L in enumerate(lens):
      cc = [tuple(t[b:b + 1].clone() if t is not None else None for t in layer) for layer in cache.caches]
      refs.append(m(nxt[b:b + 1], caches=cc, positions=torch.tensor([L], dtype=torch.int32),
                    start_pos=L, is_decode=True))
    out = m(nxt, caches=cache.caches, positions=torch.tensor(lens, dtype=torch.int32),
            start_pos=-1, is_decode=True,
            seq_lens=torch.tensor([l + 1 for l in lens], dtype=torch.int32))
    for b in range(B):
      assert torch.allclose(out[b:b + 1], refs[b], atol=1e-5), \
        (b, float((out[b:b + 1] - refs[b]).abs().max()))
