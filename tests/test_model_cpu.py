"""Model correctness on CPU: the split-vs-full logits-equality oracle.

This is the canonical sharding test from the reference
(/root/reference/xotorch/inference/test_inference_engine.py:12-47) made
hermetic: a tiny random-init llama-style config, fp32 CPU, asserting the
full-model logits equal the N-way-split logits bitwise for prefill and for
a cached decode step.
"""
import pytest
import torch

from xotorch_amd.engine.kvcache import ShardKVCache
from xotorch_amd.models.config import config_from_hf
from xotorch_amd.models.llama import ShardedModel
from xotorch_amd.models.registry import BUILTIN_CONFIGS
from xotorch_amd.models.weights import random_init
from xotorch_amd.shard import Shard

TINY = {
  "model_type": "llama", "hidden_size": 64, "num_hidden_layers": 6,
  "num_attention_heads": 4, "num_key_value_heads": 2, "intermediate_size": 128,
  "vocab_size": 199, "rope_theta": 10000.0, "rms_norm_eps": 1e-5,
  "max_position_embeddings": 64, "tie_word_embeddings": False, "torch_dtype": "float32",
}


def make_model(shard, cfg_dict=TINY):
  cfg = config_from_hf(cfg_dict, "tiny")
  cfg.torch_dtype = torch.float32
  m = ShardedModel(cfg, shard).float()
  random_init(m)
  m.eval()
  return m, cfg


def make_cache(m, cfg, B, T):
  return ShardKVCache(m.shard.get_layer_count(), B, cfg.n_kv_heads, T, cfg.head_dim,
                      dtype=torch.float32, device="cpu")


def run_pipeline(shards, tokens, n_decode=3):
  """Run prefill + n_decode greedy steps through a list of shard models."""
  B, S = tokens.shape
  T = S + n_decode + 1
  models = []
  caches = []
  for sh in shards:
    m, cfg = make_model(sh)
    models.append((m, cfg))
    caches.append(make_cache(m, cfg, B, T))
  all_logits = []
  x = tokens
  pos = 0
  cur = tokens
  with torch.inference_mode():
    for step in range(n_decode + 1):
      S_cur = cur.shape[1]
      positions = torch.arange(pos, pos + S_cur, dtype=torch.long)
      h = cur
      for (m, cfg), cache in zip(models, caches):
        h = m(h, caches=cache.caches, positions=positions, start_pos=pos,
              is_decode=(S_cur == 1 and pos > 0))
      logits = h  # last shard returns [B, V]
      if logits.dim() == 3:
        logits = logits[:, -1, :]
      all_logits.append(logits.clone())
      nxt = logits.argmax(dim=-1, keepdim=True)
      pos += S_cur
      cur = nxt
  return all_logits


@pytest.mark.parametrize("n_splits", [1, 2, 3])
def test_split_equals_full(n_splits):
  torch.manual_seed(0)
  tokens = torch.randint(0, 199, (2, 11))
  n_layers = TINY["num_hidden_layers"]
  full = run_pipeline([Shard("tiny", 0, n_layers - 1, n_layers)], tokens)
  if n_splits == 1:
    split_shards = [Shard("tiny", 0, n_layers - 1, n_layers)]
  elif n_splits == 2:
    split_shards = [Shard("tiny", 0, 2, n_layers), Shard("tiny", 3, 5, n_layers)]
  else:
    split_shards = [Shard("tiny", 0, 1, n_layers), Shard("tiny", 2, 3, n_layers), Shard("tiny", 4, 5, n_layers)]
  split = run_pipeline(split_shards, tokens)
  for lf, ls in zip(full, split):
    assert torch.equal(lf, ls), "split-vs-full logits diverged"


def test_decode_matches_prefill():
  """Decoding token-by-token must equal prefilling the whole sequence."""
  torch.manual_seed(1)
  n_layers = TINY["num_hidden_layers"]
  sh = Shard("tiny", 0, n_layers - 1, n_layers)
  m, cfg = make_model(sh)
  B, S = 1, 9
  tokens = torch.randint(0, 199, (B, S))
  with torch.inference_mode():
    cache1 = make_cache(m, cfg, B, S + 1)
    positions = torch.arange(0, S, dtype=torch.long)
    logits_full = m(tokens, caches=cache1.caches, positions=positions, start_pos=0, last_only=False)
    cache2 = make_cache(m, cfg, B, S + 1)
    logits_inc = []
    for i in range(S):
      li = m(tokens[:, i:i + 1], caches=cache2.caches,
             positions=torch.tensor([i]), start_pos=i, is_decode=(i > 0))
      logits_inc.append(li)
  inc = torch.stack(logits_inc, dim=1)
  assert torch.allclose(logits_full, inc, atol=1e-4, rtol=1e-4)


def test_tied_embeddings_head():
  cfg_dict = dict(TINY, tie_word_embeddings=True)
  n_layers = cfg_dict["num_hidden_layers"]
  m, cfg = make_model(Shard("tiny", 0, n_layers - 1, n_layers), cfg_dict)
  assert m.head_weight() is m.embed_tokens.weight


def test_builtin_configs_parse():
  from xotorch_amd.models.config import config_from_hf
  for mid, raw in BUILTIN_CONFIGS.items():
    cfg = config_from_hf(raw, mid)
    assert cfg.n_layers > 0 and cfg.dim > 0
    assert cfg.n_heads % cfg.n_kv_heads == 0


def test_qwen_bias_and_moe_construct():
  qcfg = config_from_hf(BUILTIN_CONFIGS["qwen-2.5-0.5b"], "qwen-2.5-0.5b")
  assert qcfg.attn_bias
  mcfg = config_from_hf(BUILTIN_CONFIGS["mixtral-8x7b"], "mixtral-8x7b")
  assert mcfg.n_experts == 8
  # tiny moe construct + forward
  tiny_moe = dict(TINY, num_local_experts=4, num_experts_per_tok=2)
  n_layers = TINY["num_hidden_layers"]
  m, cfg = make_model(Shard("tiny", 0, n_layers - 1, n_layers), tiny_moe)
  cache = make_cache(m, cfg, 1, 8)
  with torch.inference_mode():
    out = m(torch.randint(0, 199, (1, 5)), caches=cache.caches,
            positions=torch.arange(5), start_pos=0)
  assert out.shape == (1, 199)


def test_moe_routed_decode_matches_loop():
  """The static-capacity routed MoE path (hipGraph-capturable) must match the
  dynamic per-expert gather loop (reference semantics, llm_utils.py:502-590)."""
  import torch
  from xotorch_amd.models.config import config_from_hf
  from xotorch_amd.models.registry import builtin_config
  from xotorch_amd.models.llama import MoEMLP

  raw = dict(builtin_config("mixtral-8x7b"))
  raw.update(hidden_size=64, intermediate_size=128, num_hidden_layers=2,
             num_attention_heads=4, num_key_value_heads=2)
  cfg = config_from_hf(raw, "mixtral-tiny")
  torch.manual_seed(0)
  moe = MoEMLP(cfg).eval()
  x = torch.randn(5, 1, 64)
  with torch.no_grad():
    ref = moe(x)  # CPU -> dynamic loop
    routed = moe._forward_decode(x.view(-1, 64)).view(5, 1, 64).to(x.dtype)
  assert torch.allclose(ref, routed, atol=1e-4, rtol=1e-4), (ref - routed).abs().max()


def test_moe_capacity_is_lossless():
  """Every (token, expert) assignment must land in a slot (top-k experts are
  distinct per token, so per-expert load <= T <= capacity)."""
  import torch
  from xotorch_amd.models.config import config_from_hf
  from xotorch_amd.models.registry import builtin_config
  from xotorch_amd.models.llama import MoEMLP

  raw = dict(builtin_config("mixtral-8x7b"))
  raw.update(hidden_size=32, intermediate_size=64)
  cfg = config_from_hf(raw, "mixtral-tiny2")
  torch.manual_seed(1)
  moe = MoEMLP(cfg).eval()
  # adversarial: bias the router so one expert gets every token
  with torch.no_grad():
    moe.gate.weight.zero_()
    moe.gate.weight[3].fill_(5.0)
    x = torch.randn(64, 1, 32)
    ref = moe(x)
    routed = moe._forward_decode(x.view(-1, 32)).view(64, 1, 32).to(x.dtype)
  assert torch.allclose(ref, routed, atol=1e-4, rtol=1e-4)


def test_qwen3_qk_norm_model():
  """qwen3 family: per-head q/k RMSNorm before RoPE, no attn bias, explicit
  head_dim; full-vs-split logits equality must hold with qk-norm active."""
  import torch
  from xotorch_amd.engine.kvcache import ShardKVCache
  from xotorch_amd.models.config import config_from_hf
  from xotorch_amd.models.llama import ShardedModel, hf_key_map
  from xotorch_amd.models.weights import random_init
  from xotorch_amd.shard import Shard

  tiny = {
    "model_type": "qwen3", "hidden_size": 128, "num_hidden_layers": 4,
    "num_attention_heads": 4, "num_key_value_heads": 2, "head_dim": 32,
    "intermediate_size": 256, "vocab_size": 151, "rope_theta": 1000000.0,
    "rms_norm_eps": 1e-6, "max_position_embeddings": 64, "torch_dtype": "float32",
  }
  cfg = config_from_hf(tiny, "qwen3-tiny")
  assert cfg.qk_norm and not cfg.attn_bias and cfg.head_dim == 32
  full = Shard("qwen3-tiny", 0, 3, 4)
  m = ShardedModel(cfg, full).float()
  random_init(m)
  # perturb the norms so they actually do something
  with torch.no_grad():
    for lid in m.local_layer_ids:
      m.layers[str(lid)].self_attn.q_norm.mul_(1.5)
      m.layers[str(lid)].self_attn.k_norm.mul_(0.8)
  m.eval()
  mapping = hf_key_map(full, cfg)
  assert "model.layers.0.self_attn.q_norm.weight" in mapping
  B, S = 2, 10
  toks = torch.randint(0, 151, (B, S))
  cache_f = ShardKVCache(4, B, 2, S + 2, 32, torch.float32, "cpu")
  with torch.inference_mode():
    pos = torch.arange(S)
    lf = m(toks, caches=cache_f.caches, positions=pos, start_pos=0)
    assert torch.isfinite(lf).all()
    # split in two shards, run sequentially, compare logits
    s0, s1 = Shard("qwen3-tiny", 0, 1, 4), Shard("qwen3-tiny", 2, 3, 4)
    m0, m1 = ShardedModel(cfg, s0).float(), ShardedModel(cfg, s1).float()
    sd = m.state_dict()
    m0.load_state_dict({k: v for k, v in sd.items() if k in m0.state_dict()})
    m1.load_state_dict({k: v for k, v in sd.items() if k in m1.state_dict()})
    m0.eval(); m1.eval()
    c0 = ShardKVCache(2, B, 2, S + 2, 32, torch.float32, "cpu")
    c1 = ShardKVCache(2, B, 2, S + 2, 32, torch.float32, "cpu")
    h = m0(toks, caches=c0.caches, positions=pos, start_pos=0)
    ls = m1(h, caches=c1.caches, positions=pos, start_pos=0)
    assert torch.equal(lf, ls), (lf - ls).abs().max()


def test_qwen3_moe_tiny_forward():
  """qwen3-moe: qk-norm + per-expert moe_intermediate + qwen3 expert key map;
  routed decode path must match the dynamic loop."""
  import torch
  from xotorch_amd.models.config import config_from_hf
  from xotorch_amd.models.llama import MoEMLP, ShardedModel, hf_key_map
  from xotorch_amd.models.weights import random_init
  from xotorch_amd.engine.kvcache import ShardKVCache
  from xotorch_amd.shard import Shard

  tiny = {
    "model_type": "qwen3_moe", "hidden_size": 64, "num_hidden_layers": 2,
    "num_attention_heads": 4, "num_key_value_heads": 2, "head_dim": 16,
    "intermediate_size": 128, "moe_intermediate_size": 48, "num_experts": 8,
    "num_experts_per_tok": 3, "vocab_size": 131, "rope_theta": 1000000.0,
    "rms_norm_eps": 1e-6, "max_position_embeddings": 64, "torch_dtype": "float32",
  }
  cfg = config_from_hf(tiny, "qwen3moe-tiny")
  assert cfg.qk_norm and cfg.moe_style == "qwen3" and cfg.moe_intermediate_dim == 48
  shard = Shard("qwen3moe-tiny", 0, 1, 2)
  m = ShardedModel(cfg, shard).float()
  random_init(m)
  m.eval()
  # expert shapes use moe_intermediate
  assert m.layers["0"].mlp.experts[0].gate_up_proj.weight.shape == (96, 64)
  mapping = hf_key_map(shard, cfg)
  assert "model.layers.0.mlp.experts.0.gate_proj.weight" in mapping
  assert "model.layers.0.mlp.gate.weight" in mapping
  cache = ShardKVCache(2, 3, 2, 20, 16, torch.float32, "cpu")
  with torch.inference_mode():
    logits = m(torch.randint(0, 131, (3, 8)), caches=cache.caches,
               positions=torch.arange(8), start_pos=0)
    assert torch.isfinite(logits).all()
    # routed static-capacity path vs dynamic loop
    moe = m.layers["0"].mlp
    x = torch.randn(5, 1, 64)
    ref = moe(x)
    routed = moe._forward_decode(x.view(-1, 64)).view(5, 1, 64).to(x.dtype)
    assert torch.allclose(ref, routed, atol=1e-4, rtol=1e-4)


def test_registry_consistency():
  """Every model card: pretty name present; when a builtin config exists its
  layer count must match the card; supported models resolve base shards."""
  from xotorch_amd.models.registry import (
    BUILTIN_CONFIGS, build_base_shard, builtin_config, get_supported_models,
    model_cards, pretty_name)
  for mid, card in model_cards.items():
    assert pretty_name(mid), mid
    assert card.get("layers", 0) > 0, mid
    cfg = builtin_config(mid)
    if cfg is not None:
      assert cfg["num_hidden_layers"] == card["layers"], \
        f"{mid}: card layers {card['layers']} != builtin {cfg['num_hidden_layers']}"
  # every builtin config belongs to a card
  for mid in BUILTIN_CONFIGS:
    assert mid in model_cards or mid in ("dummy",), mid
  for mid in get_supported_models([["HIPEngine"]]):
    assert build_base_shard(mid, "HIPEngine") is not None, mid


def test_pack_decode_weight_is_a_permutation():
  """The MFMA prepack is a pure permutation of W: inverting the documented
  index map (docs/KERNELS.md) must recover W exactly, and every element must
  land where the kernel's A-fragment load expects it:
  packed[n32][kt][half][n][e] == W[n32*32+n][kt*16+half*8+e]."""
  import torch
  from xotorch_amd.ops import pack_decode_weight
  N, K = 96, 128
  w = torch.arange(N * K, dtype=torch.float32).reshape(N, K).to(torch.bfloat16)
  p = pack_decode_weight(w)
  assert p.shape == (N // 32, K // 16, 2, 32, 8)
  # invert
  back = p.permute(0, 3, 1, 2, 4).reshape(N, K)
  assert torch.equal(back, w)
  # spot-check the lane address math the kernel uses: element loaded by lane
  # l at (n32, kt) from flat offset (n32*(K/16)+kt)*512 + l*8 + j must be
  # W[n32*32 + (l&31)][kt*16 + (l>>5)*8 + j]
  flat = p.reshape(-1)
  for (n32, kt, l, j) in [(0, 0, 0, 0), (1, 3, 45, 7), (2, 7, 31, 4), (0, 1, 63, 0)]:
    got = flat[(n32 * (K // 16) + kt) * 512 + l * 8 + j]
    want = w[n32 * 32 + (l & 31)][kt * 16 + (l >> 5) * 8 + j]
    assert torch.equal(got, want), (n32, kt, l, j)


def test_qwen2_yarn_matches_transformers():
  """YaRN rope on the GQA family (qwen2 long-context configs): our generic
  yarn branch in rope_cos_sin vs transformers' yarn implementation, prefill
  past the pretraining window + one cached decode step."""
  from transformers import Qwen2Config, Qwen2ForCausalLM
  from xotorch_amd.models.weights import hf_key_map, remap_hf_state
  kw = dict(vocab_size=151, hidden_size=64, intermediate_size=128, num_hidden_layers=3,
            num_attention_heads=4, num_key_value_heads=2, max_position_embeddings=64,
            rope_theta=10000.0, rms_norm_eps=1e-6, tie_word_embeddings=False,
            rope_scaling={"rope_type": "yarn", "factor": 4.0, "beta_fast": 32,
                          "beta_slow": 1, "original_max_position_embeddings": 16})
  torch.manual_seed(11)
  ref = Qwen2ForCausalLM(Qwen2Config(**kw)).eval().float()
  cfg = config_from_hf({**kw, "model_type": "qwen2"}, "qwen-tiny")
  assert cfg.rope_scaling is not None and cfg.rope_scaling.rope_type == "yarn"
  cfg.torch_dtype = torch.float32
  shard = Shard("qwen-tiny", 0, cfg.n_layers - 1, cfg.n_layers)
  ours = ShardedModel(cfg, shard).float()
  sd = ref.state_dict()
  missing = ours.load_state_dict(remap_hf_state(sd, hf_key_map(shard, cfg)), strict=False).missing_keys
  assert not [m for m in missing if "rope" not in m], missing
  ours.eval()
  B, S = 2, 24  # past the 16-token pretraining window
  x = torch.randint(0, 151, (B, S), generator=torch.Generator().manual_seed(4))
  cache = ShardKVCache(cfg.n_layers, B, cfg.n_kv_heads, S + 4, cfg.head_dim,
                       dtype=torch.float32, device="cpu")
  with torch.no_grad():
    lref = ref(x).logits
    lours = ours(x, caches=cache.caches, positions=torch.arange(S), start_pos=0, last_only=False)
    torch.testing.assert_close(lours, lref, atol=3e-4, rtol=3e-4)
    nxt = lref[:, -1].argmax(-1, keepdim=True)
    lref2 = ref(torch.cat([x, nxt], 1)).logits[:, -1]
    lours2 = ours(nxt, caches=cache.caches, positions=torch.tensor([S]), start_pos=S, is_decode=True)
    torch.testing.assert_close(lours2, lref2, atol=3e-4, rtol=3e-4)


def test_config_parse_key_subsets_fuzz():
  """config_from_hf must parse any subset of a real config's keys without
  crashing (HF configs in the wild omit fields freely) and keep derived
  invariants (head_dim, kv heads <= heads fallback)."""
  from hypothesis import given, settings, strategies as st

  FULL = {
    "model_type": "llama", "hidden_size": 128, "num_hidden_layers": 4,
    "num_attention_heads": 8, "num_key_value_heads": 4, "intermediate_size": 256,
    "vocab_size": 500, "rope_theta": 10000.0, "rms_norm_eps": 1e-5,
    "max_position_embeddings": 512, "tie_word_embeddings": True,
    "head_dim": 16, "attention_bias": False, "torch_dtype": "bfloat16",
    "eos_token_id": [7, 8], "bos_token_id": 1,
    "rope_scaling": {"rope_type": "llama3", "factor": 8.0,
                     "low_freq_factor": 1.0, "high_freq_factor": 4.0,
                     "original_max_position_embeddings": 256},
  }
  keys = sorted(FULL)

  @settings(max_examples=120, deadline=None)
  @given(mask=st.lists(st.booleans(), min_size=len(keys), max_size=len(keys)))
  def check(mask):
    raw = {k: FULL[k] for k, keep in zip(keys, mask) if keep}
    cfg = config_from_hf(raw, "fuzz")
    assert cfg.n_heads >= 1 and cfg.n_kv_heads >= 1
    assert cfg.n_kv_heads <= cfg.n_heads
    assert cfg.head_dim * 1 > 0
    assert cfg.max_seq_len >= 1
    if "eos_token_id" in raw:
      assert cfg.eos_token_id == 7  # list collapses to first

  check()


def test_kv_cache_dims_all_builtin_configs():
  """Every builtin config yields a constructible KV cache whose per-token
  byte cost is positive and (for MLA) much smaller than GQA equivalents."""
  from xotorch_amd.engine.kvcache import ShardKVCache
  for mid, raw in BUILTIN_CONFIGS.items():
    cfg = config_from_hf(raw, mid)
    heads, kd, vd = cfg.kv_cache_dims()
    assert heads >= 1 and kd >= 1
    c = ShardKVCache(1, 1, heads, 32, kd, torch.float32, "cpu", v_dim=vd)
    lk = c.caches[0]
    assert lk.k.shape[2] >= 32 - 31  # T dim exists (may be padded)
    if cfg.model_type in ("deepseek_v3", "deepseek_v2"):
      gqa_bytes = cfg.n_kv_heads * cfg.head_dim * 2 * 2
      mla_bytes = heads * (kd + (vd or kd)) * 2
      assert mla_bytes < gqa_bytes / 2, mid  # the latent cache is the point
