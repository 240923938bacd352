"""TorchEngine end-to-end on CPU with a random-init tiny model (offline)."""
import asyncio

import numpy as np
import pytest
import torch

from xotorch_amd.engine.dummy import DummyEngine
from xotorch_amd.engine.torch_engine import TorchEngine
from xotorch_amd.models.registry import build_full_shard
from xotorch_amd.shard import Shard


def run(coro):
  return asyncio.new_event_loop().run_until_complete(coro)


def test_engine_prefill_decode_loop():
  async def go():
    eng = TorchEngine(device="cpu", dtype=torch.float32)
    shard = build_full_shard("dummy", "TorchEngine")
    out, state = await eng.infer_prompt("req1", shard, "hello world test prompt")
    assert out.shape[-1] == 256  # dummy vocab
    tok = await eng.sample(out, temp=0.0)
    # 3 decode steps
    for _ in range(3):
      out, state = await eng.infer_tensor("req1", shard, tok.reshape(1, 1), state)
      assert np.isfinite(out).all()
      tok = await eng.sample(out, temp=0.0)
    assert state["curr_pos"] > 4
    return True
  assert run(go())


def test_engine_sharded_ring_in_process():
  """Two shard engines chained in-process: hidden state hop between them."""
  async def go():
    n_layers = 4
    sh0 = Shard("dummy", 0, 1, n_layers)
    sh1 = Shard("dummy", 2, 3, n_layers)
    e0 = TorchEngine(device="cpu", dtype=torch.float32)
    e1 = TorchEngine(device="cpu", dtype=torch.float32)
    full = TorchEngine(device="cpu", dtype=torch.float32)
    prompt = "the quick brown fox"
    h, s0 = await e0.infer_prompt("r", sh0, prompt)
    assert h.ndim == 3  # hidden state
    logits, s1 = await e1.infer_tensor("r", sh1, h, s0)
    ref, _ = await full.infer_prompt("r", build_full_shard("dummy", "TorchEngine"), prompt)
    assert np.allclose(logits, ref, atol=1e-4)
    return True
  assert run(go())


def test_engine_train_last_shard():
  async def go():
    eng = TorchEngine(device="cpu", dtype=torch.float32)
    shard = build_full_shard("dummy", "TorchEngine")
    B, S = 2, 8
    inputs = np.random.randint(0, 200, (B, S))
    targets = np.roll(inputs, -1, axis=1)
    lengths = np.array([S, S - 2])
    loss1, _ = await eng.train("t", shard, inputs, targets, lengths)
    loss2, _ = await eng.evaluate("t", shard, inputs, targets, lengths)
    assert np.isfinite(loss1) and np.isfinite(loss2)
    # a few steps reduce loss on a fixed batch
    for _ in range(8):
      loss3, _ = await eng.train("t", shard, inputs, targets, lengths)
    assert loss3 < loss1
    return True
  assert run(go())


def test_engine_ring_train_back_gradient():
  """Pipeline training: last shard CE + gradient hop back to the first shard."""
  async def go():
    n_layers = 4
    sh0 = Shard("dummy", 0, 1, n_layers)
    sh1 = Shard("dummy", 2, 3, n_layers)
    e0 = TorchEngine(device="cpu", dtype=torch.float32)
    e1 = TorchEngine(device="cpu", dtype=torch.float32)
    B, S = 2, 6
    inputs = np.random.randint(0, 200, (B, S))
    targets = np.roll(inputs, -1, axis=1)
    lengths = np.array([S, S])
    # forward through shard0 (training-mode forward returns hidden)
    h, _ = await e0.infer_tensor("t", sh0, inputs, {"curr_pos": 0})
    loss, back_grad = await e1.train("t", sh1, h, targets, lengths)
    assert np.isfinite(loss) and back_grad is not None
    assert back_grad.shape == h.shape
    loss0, _ = await e0.train("t", sh0, inputs, back_grad, lengths, loss="back_gradient")
    return True
  assert run(go())


def test_engine_checkpoint_roundtrip(tmp_path):
  async def go():
    eng = TorchEngine(device="cpu", dtype=torch.float32)
    shard = build_full_shard("dummy", "TorchEngine")
    await eng.ensure_shard(shard)
    p = str(tmp_path / "ckpt.safetensors")
    await eng.save_checkpoint(shard, p)
    w_before = eng.model.layers["0"].self_attn.qkv_proj.weight.clone()
    with torch.no_grad():
      eng.model.layers["0"].self_attn.qkv_proj.weight.add_(1.0)
    await eng.load_checkpoint(shard, p)
    assert torch.allclose(eng.model.layers["0"].self_attn.qkv_proj.weight, w_before)
    return True
  assert run(go())


def test_dummy_engine_plumbing():
  async def go():
    eng = DummyEngine()
    shard = build_full_shard("dummy", "DummyEngine")
    out, state = await eng.infer_prompt("r", shard, "hi")
    tok = await eng.sample(out)
    assert tok.shape[0] == 1
    return True
  assert run(go())


def test_sessions_cleaned_after_finish_and_error():
  """Per-request KV sessions must not leak: cleared on normal finish and on
  context exhaustion (288 GB fills fast if sessions accumulate)."""
  import asyncio

  async def go():
    from xotorch_amd.engine.torch_engine import TorchEngine
    from xotorch_amd.models.registry import build_base_shard
    from xotorch_amd.orchestration.node import Node
    eng = TorchEngine(device="cpu", dtype=torch.float32)
    node = Node("sess-test", None, eng, None, max_generate_tokens=6)
    await node.start(0)
    try:
      shard = build_base_shard("dummy", "TorchEngine")
      done = asyncio.Event()
      node.on_token.register("s").on_next(lambda rid, t, fin: done.set() if fin else None)
      await node.process_prompt(shard, "hello", "sess-1")
      await asyncio.wait_for(done.wait(), 60)
      await asyncio.sleep(0.1)  # let the cleanup task run
      assert "sess-1" not in eng.sessions, list(eng.sessions)
      # context-exhaustion path also cleans up
      node.max_generate_tokens = 100000
      done2 = asyncio.Event()
      node.on_token.register("s2").on_next(lambda rid, t, fin: done2.set() if fin else None)
      await node.process_prompt(shard, "again", "sess-2")
      await asyncio.wait_for(done2.wait(), 120)
      await asyncio.sleep(0.1)
      assert "sess-2" not in eng.sessions, list(eng.sessions)
    finally:
      await node.stop()
  asyncio.run(go())


def test_engine_preserves_passthrough_state():
  """Engines must MERGE (not replace) inference_state: request metadata like
  max_tokens / traceparent has to survive every ring hop."""
  import asyncio

  async def go():
    from xotorch_amd.engine.torch_engine import TorchEngine
    from xotorch_amd.models.registry import build_full_shard
    import numpy as np
    eng = TorchEngine(device="cpu", dtype=torch.float32)
    shard = build_full_shard("dummy", "TorchEngine")
    toks = np.array([[1, 2, 3]], dtype=np.int64)
    state_in = {"max_tokens": 7, "traceparent": "00-abc-def-01"}
    _, state_out = await eng.infer_tensor("pt-1", shard, toks, state_in)
    assert state_out["max_tokens"] == 7
    assert state_out["traceparent"] == "00-abc-def-01"
    assert state_out["curr_pos"] == 3
  asyncio.run(go())


def test_softmax_sample_property_fuzz():
  """Hypothesis fuzz of the sampling reference: samples always come from
  the top-k set, never from -inf-masked tokens, and top-p keeps at least
  the argmax; temperature 0 is exact argmax."""
  import torch
  from hypothesis import given, settings, strategies as st
  from xotorch_amd.ops.torch_ref import softmax_sample

  @settings(max_examples=150, deadline=None)
  @given(
    seed=st.integers(min_value=0, max_value=2**31 - 1),
    vocab=st.integers(min_value=2, max_value=300),
    batch=st.integers(min_value=1, max_value=4),
    temp=st.floats(min_value=0.0, max_value=3.0, allow_nan=False),
    top_k=st.integers(min_value=0, max_value=310),
    top_p=st.floats(min_value=0.0, max_value=1.0, allow_nan=False),
  )
  def check(seed, vocab, batch, temp, top_k, top_p):
    g = torch.Generator().manual_seed(seed)
    logits = torch.randn(batch, vocab, generator=g) * 3
    toks = softmax_sample(logits, temperature=temp, top_k=top_k,
                          generator=g, top_p=top_p)
    assert toks.shape == (batch,)
    assert ((toks >= 0) & (toks < vocab)).all()
    if temp <= 1e-4:  # sub-epsilon temperature collapses to greedy
      assert (toks == logits.argmax(-1)).all()
      return
    if top_k and 0 < top_k < vocab:
      kth = torch.topk(logits, top_k, dim=-1).values[..., -1]
      picked = logits.gather(-1, toks[:, None]).squeeze(-1)
      assert (picked >= kth).all(), "sampled outside the top-k set"

  check()


def test_rope_rotation_invariants_fuzz():
  """RoPE is a rotation: each (i, i+hd/2) pair keeps its 2-norm, position 0
  is identity, and q.k dot products depend only on relative position."""
  import torch
  from hypothesis import given, settings, strategies as st
  from xotorch_amd.ops.torch_ref import rope_apply, rope_cos_sin

  @settings(max_examples=60, deadline=None)
  @given(seed=st.integers(min_value=0, max_value=2**31 - 1),
         hd=st.sampled_from([8, 32, 64, 128]),
         pos=st.integers(min_value=0, max_value=100))
  def check(seed, hd, pos):
    g = torch.Generator().manual_seed(seed)
    cos, sin = rope_cos_sin(hd, 256, 10000.0)
    q = torch.randn(1, 1, 2, hd, generator=g)
    k = torch.randn(1, 1, 1, hd, generator=g)
    qp, kp = rope_apply(q, k, cos, sin, torch.tensor([pos]))
    half = hd // 2
    pn = lambda x: (x[..., :half] ** 2 + x[..., half:] ** 2)
    torch.testing.assert_close(pn(qp), pn(q), rtol=1e-4, atol=1e-5)
    if pos == 0:
      torch.testing.assert_close(qp, q)
    # relative-position property: <rope(q,p+d), rope(k,p)> independent of p
    d = 7
    qa, ka = rope_apply(q, k, cos, sin, torch.tensor([pos + d]))
    _, kb = rope_apply(q, k, cos, sin, torch.tensor([pos]))
    dot1 = (qa[0, 0, 0] * kb[0, 0, 0]).sum()
    qc, _ = rope_apply(q, k, cos, sin, torch.tensor([d]))
    _, kd = rope_apply(q, k, cos, sin, torch.tensor([0]))
    dot2 = (qc[0, 0, 0] * kd[0, 0, 0]).sum()
    torch.testing.assert_close(dot1, dot2, rtol=1e-3, atol=1e-3)

  check()


def test_attn_decode_convexity_and_ragged_mask_fuzz():
  """Decode attention output is a convex combination of the first seq_len
  value rows — per-slot ragged lengths must never leak later cache rows
  (the property continuous batching rides on)."""
  import torch
  from hypothesis import given, settings, strategies as st
  from xotorch_amd.ops.torch_ref import attn_decode

  @settings(max_examples=60, deadline=None)
  @given(seed=st.integers(min_value=0, max_value=2**31 - 1),
         B=st.integers(min_value=1, max_value=4),
         T=st.integers(min_value=2, max_value=24))
  def check(seed, B, T):
    g = torch.Generator().manual_seed(seed)
    H, KVH, hd = 4, 2, 16
    q = torch.randn(B, 1, H, hd, generator=g)
    k = torch.randn(B, KVH, T, hd, generator=g)
    v = torch.randn(B, KVH, T, hd, generator=g)
    seq_lens = torch.randint(1, T + 1, (B,), generator=g)
    # poison the rows past each slot's seq_len: they must not matter
    v2 = v.clone()
    k2 = k.clone()
    for b in range(B):
      v2[b, :, seq_lens[b]:] = 1e6
      k2[b, :, seq_lens[b]:] = 1e6
    out = attn_decode(q, k, v, seq_lens)
    out2 = attn_decode(q, k2, v2, seq_lens)
    torch.testing.assert_close(out, out2, rtol=1e-5, atol=1e-6)
    # convexity: within [min, max] of the visible value rows per slot/head
    for b in range(B):
      vis = v[b].repeat_interleave(H // KVH, dim=0)[:, :seq_lens[b]]  # [H,L,hd]
      lo = vis.min(dim=1).values - 1e-4
      hi = vis.max(dim=1).values + 1e-4
      o = out[b, 0]
      assert (o >= lo).all() and (o <= hi).all()

  check()


def test_attn_prefill_vs_bruteforce_fuzz():
  """Cross-check the prefill attention reference (the oracle the HIP kernel
  is validated against) with an independent brute-force loop, including
  chunk offsets, sliding windows and softcap."""
  import torch
  from hypothesis import given, settings, strategies as st
  from xotorch_amd.ops.torch_ref import attn_prefill

  @settings(max_examples=40, deadline=None)
  @given(seed=st.integers(min_value=0, max_value=2**31 - 1),
         S=st.integers(min_value=1, max_value=6),
         start=st.integers(min_value=0, max_value=6),
         window=st.sampled_from([0, 2, 4]),
         softcap=st.sampled_from([0.0, 5.0]))
  def check(seed, S, start, window, softcap):
    g = torch.Generator().manual_seed(seed)
    B, H, KVH, hd = 2, 4, 2, 8
    T = start + S
    q = torch.randn(B, S, H, hd, generator=g)
    kc = torch.randn(B, KVH, T + 2, hd, generator=g)
    vc = torch.randn(B, KVH, T + 2, hd, generator=g)
    out = attn_prefill(q, kc, vc, start, S, window=window, softcap=softcap)
    scale = hd ** -0.5
    for b in range(B):
      for h in range(H):
        kv = h // (H // KVH)
        for i in range(S):
          pos = start + i
          lo = max(0, pos - window + 1) if window else 0
          ks = kc[b, kv, lo:pos + 1]
          sc = (q[b, i, h] @ ks.T) * scale
          if softcap:
            sc = torch.tanh(sc / softcap) * softcap
          ref = torch.softmax(sc, dim=-1) @ vc[b, kv, lo:pos + 1]
          torch.testing.assert_close(out[b, i, h], ref, rtol=2e-4, atol=2e-5)

  check()


def test_attn_decode_vs_bruteforce_fuzz():
  """Same independent cross-check for the decode attention oracle: ragged
  per-slot lengths, sliding window, softcap."""
  import torch
  from hypothesis import given, settings, strategies as st
  from xotorch_amd.ops.torch_ref import attn_decode

  @settings(max_examples=40, deadline=None)
  @given(seed=st.integers(min_value=0, max_value=2**31 - 1),
         T=st.integers(min_value=1, max_value=12),
         window=st.sampled_from([0, 3]),
         softcap=st.sampled_from([0.0, 4.0]))
  def check(seed, T, window, softcap):
    g = torch.Generator().manual_seed(seed)
    B, H, KVH, hd = 2, 4, 2, 8
    q = torch.randn(B, 1, H, hd, generator=g)
    kc = torch.randn(B, KVH, T + 3, hd, generator=g)
    vc = torch.randn(B, KVH, T + 3, hd, generator=g)
    seq_lens = torch.randint(1, T + 1, (B,), generator=g)
    out = attn_decode(q, kc, vc, seq_lens, window=window, softcap=softcap)
    scale = hd ** -0.5
    for b in range(B):
      L = int(seq_lens[b])
      lo = max(0, L - window) if window else 0
      for h in range(H):
        kv = h // (H // KVH)
        sc = (q[b, 0, h] @ kc[b, kv, lo:L].T) * scale
        if softcap:
          sc = torch.tanh(sc / softcap) * softcap
        ref = torch.softmax(sc, dim=-1) @ vc[b, kv, lo:L]
        torch.testing.assert_close(out[b, 0, h], ref, rtol=2e-4, atol=2e-5)

  check()
