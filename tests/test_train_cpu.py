"""Training tests: LoRA adapters, DP bucketed all-reduce (gloo, 2 procs),
dataset loading."""
import json
import os
from pathlib import Path

import numpy as np
import pytest
import torch
import torch.multiprocessing as mp

TINY = {
  "model_type": "llama", "hidden_size": 64, "num_hidden_layers": 2,
  "num_attention_heads": 4, "num_key_value_heads": 2, "intermediate_size": 128,
  "vocab_size": 128, "rope_theta": 10000.0, "rms_norm_eps": 1e-5,
  "max_position_embeddings": 64, "torch_dtype": "float32",
}


def tiny_model():
  from xotorch_amd.models.config import config_from_hf
  from xotorch_amd.models.llama import ShardedModel
  from xotorch_amd.models.weights import random_init
  from xotorch_amd.shard import Shard
  cfg = config_from_hf(TINY, "tiny")
  m = ShardedModel(cfg, Shard("tiny", 0, 1, 2)).float()
  random_init(m)
  return m, cfg


def fwd_fn(model):
  def fn(tokens):
    from xotorch_amd.ops import torch_ref as tr
    h = model.embed_tokens(tokens)
    S = h.shape[1]
    positions = torch.arange(S)
    for lid in model.local_layer_ids:
      layer = model.layers[str(lid)]
      normed = tr.rmsnorm(h, layer.input_layernorm.weight, layer.eps)
      attn = layer.self_attn
      B = h.shape[0]
      c = attn.cfg
      qkv = attn.qkv_proj(normed)
      q, k, v = torch.split(qkv, [c.n_heads * c.head_dim, c.n_kv_heads * c.head_dim, c.n_kv_heads * c.head_dim], -1)
      q = q.reshape(B, S, c.n_heads, c.head_dim)
      k = k.reshape(B, S, c.n_kv_heads, c.head_dim)
      v = v.reshape(B, S, c.n_kv_heads, c.head_dim)
      q, k = tr.rope_apply(q, k, model.rope_cos, model.rope_sin, positions)
      rep = c.n_heads // c.n_kv_heads
      out = torch.nn.functional.scaled_dot_product_attention(
        q.transpose(1, 2), k.transpose(1, 2).repeat_interleave(rep, 1),
        v.transpose(1, 2).repeat_interleave(rep, 1), is_causal=True)
      h = h + attn.o_proj(out.transpose(1, 2).reshape(B, S, -1))
      normed2 = tr.rmsnorm(h, layer.post_attention_layernorm.weight, layer.eps)
      h = h + layer.mlp(normed2)
    h = tr.rmsnorm(h, model.norm.weight, model.cfg.norm_eps)
    return torch.nn.functional.linear(h, model.head_weight())
  return fn


def test_lora_wrap_and_merge():
  from xotorch_amd.train.lora import apply_lora, lora_parameters, lora_state_dict, load_lora_state_dict
  m, cfg = tiny_model()
  wrapped = apply_lora(m, rank=4)
  assert len(wrapped) == 2 * 4  # 2 layers x (qkv, o, gate_up, down)
  trainable = lora_parameters(m)
  assert trainable and all(p.requires_grad for p in trainable)
  base = [p for n, p in m.named_parameters() if "lora_" not in n]
  assert all(not p.requires_grad for p in base)
  # with lora_b zero-init, output == base output; after merge, same too
  x = torch.randint(0, 128, (1, 8))
  f = fwd_fn(m)
  with torch.no_grad():
    y0 = f(x)
  lin = m.layers["0"].self_attn.qkv_proj
  with torch.no_grad():
    lin.lora_b.normal_(0, 0.05)
    y1 = f(x)
    lin.merge()
    y2 = f(x)
    lin.unmerge()
    y3 = f(x)
  assert not torch.allclose(y0, y1)
  assert torch.allclose(y1, y2, atol=1e-4)
  assert torch.allclose(y1, y3, atol=1e-5)
  sd = lora_state_dict(m)
  assert load_lora_state_dict(m, sd) == len(sd)


def test_lora_training_reduces_loss():
  from xotorch_amd.train.lora import apply_lora, lora_parameters
  from xotorch_amd.train.trainer import DPTrainer
  torch.manual_seed(0)
  m, cfg = tiny_model()
  apply_lora(m, rank=4)
  trainer = DPTrainer(m, lr=1e-2, trainable_params=lora_parameters(m))
  tokens = torch.randint(0, 128, (4, 12))
  targets = torch.roll(tokens, -1, 1)
  f = fwd_fn(m)
  first = trainer.step(f, tokens, targets)
  for _ in range(12):
    last = trainer.step(f, tokens, targets)
  assert last < first


def _dp_worker(rank, world, port, out_dir):
  os.environ.update(RANK=str(rank), WORLD_SIZE=str(world),
                    MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port))
  import torch.distributed as dist
  dist.init_process_group("gloo", rank=rank, world_size=world)
  try:
    from xotorch_amd.train.lora import apply_lora, lora_parameters
    from xotorch_amd.train.trainer import DPTrainer
    torch.manual_seed(100)  # same init on both ranks via crc32-seeded random_init
    m, cfg = tiny_model()
    apply_lora(m, rank=4)
    trainer = DPTrainer(m, lr=1e-2, trainable_params=lora_parameters(m), bucket_bytes=4096)
    trainer.sync_initial_state()
    g = torch.Generator().manual_seed(42 + rank)  # different data per rank
    tokens = torch.randint(0, 128, (2, 10), generator=g)
    targets = torch.roll(tokens, -1, 1)
    f = fwd_fn(m)
    losses = [trainer.step(f, tokens, targets) for _ in range(4)]
    # after DP steps, replicas must hold IDENTICAL adapter weights
    sd = {n: p.detach() for n, p in m.named_parameters() if "lora_" in n}
    flat = torch.cat([v.reshape(-1) for _, v in sorted(sd.items())])
    Path(out_dir, f"rank{rank}.pt").write_bytes(b"")
    torch.save({"flat": flat, "losses": losses}, Path(out_dir, f"rank{rank}.pt"))
  finally:
    dist.destroy_process_group()


@pytest.mark.timeout(240)
def test_dp_allreduce_replicas_stay_identical(tmp_path):
  from xotorch_amd.helpers import find_available_port
  port = find_available_port("127.0.0.1")
  mp.spawn(_dp_worker, args=(2, port, str(tmp_path)), nprocs=2, join=True)
  r0 = torch.load(tmp_path / "rank0.pt")
  r1 = torch.load(tmp_path / "rank1.pt")
  assert torch.allclose(r0["flat"], r1["flat"], atol=1e-6), "replicas diverged"
  assert all(np.isfinite(r0["losses"]))


def test_dataset_loading(tmp_path):
  from xotorch_amd.train.dataset import iterate_batches, load_dataset
  data = [{"text": f"sample number {i} with words"} for i in range(10)]
  (tmp_path / "train.jsonl").write_text("\n".join(json.dumps(d) for d in data))
  (tmp_path / "valid.jsonl").write_text(json.dumps({"text": "valid sample"}))
  enc = lambda t: [ord(c) % 100 for c in t]
  train, valid, test = load_dataset(str(tmp_path), enc)
  assert len(train) == 10 and len(valid) == 1 and test == []
  batches = list(iterate_batches(train, batch_size=4))
  assert len(batches) == 2
  inputs, targets, lengths = batches[0]
  assert inputs.shape == targets.shape
  assert (targets[:, :-1] == inputs[:, 1:]).all()


def test_train_forward_is_inference_forward():
  """The cache-free training forward and the cached inference prefill are
  the SAME model code — logits must match exactly (fp32, dummy model),
  with and without activation checkpointing."""
  import os
  import torch
  from xotorch_amd.engine.kvcache import ShardKVCache
  from xotorch_amd.models.config import config_from_hf
  from xotorch_amd.models.llama import ShardedModel
  from xotorch_amd.models.registry import builtin_config
  from xotorch_amd.models.weights import random_init
  from xotorch_amd.shard import Shard
  cfg = config_from_hf(builtin_config("dummy"), "dummy")
  shard = Shard("dummy", 0, cfg.n_layers - 1, cfg.n_layers)
  m = ShardedModel(cfg, shard).float()
  random_init(m)
  B, S = 2, 10
  toks = torch.randint(0, cfg.vocab_size, (B, S))
  pos = torch.arange(S, dtype=torch.int32)
  cache = ShardKVCache(cfg.n_layers, B, cfg.n_kv_heads, S + 2, cfg.head_dim, torch.float32, "cpu")
  m.eval()
  with torch.no_grad():
    ref = m(toks, caches=cache.caches, positions=pos, start_pos=0, last_only=False)
  m.train()
  for ckpt in ("0", "1"):
    os.environ["XOT_ACT_CKPT"] = ckpt
    out = m(toks, caches=None, positions=pos, start_pos=0, last_only=False)
    assert torch.allclose(out, ref, atol=1e-4, rtol=1e-4), (ckpt, (out - ref).abs().max())
    # and it is differentiable end to end
    out.float().pow(2).mean().backward()
  os.environ.pop("XOT_ACT_CKPT", None)
  grads = [p.grad for p in m.parameters() if p.requires_grad]
  assert any(g is not None and g.abs().sum() > 0 for g in grads)


def test_gemma2_train_forward_matches_inference():
  import torch
  from xotorch_amd.engine.kvcache import ShardKVCache
  from xotorch_amd.models.config import config_from_hf
  from xotorch_amd.models.gemma2 import Gemma2Model
  from xotorch_amd.models.weights import random_init
  from xotorch_amd.shard import Shard
  raw = dict(model_type="gemma2", vocab_size=101, hidden_size=64, intermediate_size=128,
             num_hidden_layers=4, num_attention_heads=4, num_key_value_heads=2, head_dim=16,
             sliding_window=8, attn_logit_softcapping=50.0, final_logit_softcapping=30.0,
             query_pre_attn_scalar=16, rms_norm_eps=1e-6, rope_theta=10000.0,
             max_position_embeddings=64, tie_word_embeddings=True)
  cfg = config_from_hf(raw, "gemma2-tiny")
  shard = Shard("gemma2-tiny", 0, 3, 4)
  m = Gemma2Model(cfg, shard).float()
  random_init(m)
  B, S = 2, 12
  toks = torch.randint(0, 101, (B, S))
  pos = torch.arange(S, dtype=torch.int32)
  cache = ShardKVCache(4, B, 2, S + 2, 16, torch.float32, "cpu")
  m.eval()
  with torch.no_grad():
    ref = m(toks, caches=cache.caches, positions=pos, start_pos=0, last_only=False)
  out = m(toks, caches=None, positions=pos, start_pos=0, last_only=False)
  assert torch.allclose(out, ref, atol=1e-4, rtol=1e-4), (out - ref).abs().max()
