"""Numerics tests for the CDNA4 HIP kernels vs the plain-PyTorch fp32 reference
(xotorch_amd/ops/torch_ref.py). Each op: random bf16 inputs, HIP output
compared against the fp32 reference cast to bf16."""
import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def hip():
  from xotorch_amd import ops
  assert ops.hip_available(), f"HIP ext must be built: {ops._hip_load_error}"
  return ops


def bt(*shape, scale=1.0, seed=0):
  g = torch.Generator(device="cuda").manual_seed(seed)
  return (torch.randn(*shape, generator=g, device="cuda", dtype=torch.float32) * scale).to(torch.bfloat16)


@pytest.mark.parametrize("shape", [(2, 1, 2048), (4, 7, 4096), (1, 3, 8192), (2, 2, 16384)])
def test_rmsnorm(hip, shape):
  from xotorch_amd.ops import torch_ref
  x = bt(*shape)
  w = bt(shape[-1], seed=1)
  out = hip.rmsnorm(x, w, 1e-5)
  ref = torch_ref.rmsnorm(x, w, 1e-5)
  assert torch.allclose(out.float(), ref.float(), atol=3e-2, rtol=3e-2)


def test_rmsnorm_residual(hip):
  from xotorch_amd.ops import torch_ref
  x, r = bt(3, 2, 4096), bt(3, 2, 4096, seed=2)
  w = bt(4096, seed=3)
  out, res = hip.rmsnorm_residual(x, r, w, 1e-5)
  ref_out, ref_res = torch_ref.rmsnorm_residual(x, r, w, 1e-5)
  assert torch.allclose(res.float(), ref_res.float(), atol=2e-2, rtol=2e-2)
  assert torch.allclose(out.float(), ref_out.float(), atol=3e-2, rtol=3e-2)


@pytest.mark.parametrize("hd,H,KVH,S", [(128, 8, 2, 5), (64, 4, 4, 1), (128, 64, 8, 3)])
def test_rope_qkv_append(hip, hd, H, KVH, S):
  from xotorch_amd.ops import torch_ref
  B, T, start = 2, 32, 11
  qkv = bt(B, S, (H + 2 * KVH) * hd)
  cos, sin = torch_ref.rope_cos_sin(hd, T, 500000.0, device="cuda")
  kc = torch.zeros(B, KVH, T, hd, dtype=torch.bfloat16, device="cuda")
  vc = torch.zeros_like(kc)
  kc_ref, vc_ref = kc.clone(), vc.clone()
  positions = torch.arange(start, start + S, dtype=torch.int32, device="cuda")
  qkv_ref = qkv.clone()
  torch_ref.rope_qkv_append(qkv_ref, cos, sin, positions, kc_ref, vc_ref, H, KVH, hd)
  hip.rope_qkv_append(qkv, cos, sin, positions, kc, vc, H, KVH, hd)
  assert torch.allclose(qkv.float(), qkv_ref.float(), atol=2e-2, rtol=2e-2)
  assert torch.allclose(kc.float(), kc_ref.float(), atol=2e-2, rtol=2e-2)
  assert torch.equal(vc, vc_ref)


@pytest.mark.parametrize("hd,H,KVH,B,T,sl", [
  (128, 64, 8, 4, 512, 301),   # llama-70b shape
  (128, 32, 8, 2, 2048, 2048), # llama-8b long
  (64, 32, 8, 3, 256, 77),     # llama-1b hd=64
  (128, 14, 2, 2, 128, 128),   # qwen-0.5-like rep=7
  (128, 96, 8, 1, 128, 65),    # rep=12 (two head-slices)
])
def test_attn_decode(hip, hd, H, KVH, B, T, sl):
  from xotorch_amd.ops import torch_ref
  q = bt(B, 1, H, hd)
  kc = bt(B, KVH, T, hd, seed=6)
  vc = bt(B, KVH, T, hd, seed=7)
  seq_lens = torch.full((B,), sl, dtype=torch.int32, device="cuda")
  out = hip.attn_decode(q, kc, vc, seq_lens)
  ref = torch_ref.attn_decode(q, kc, vc, sl)
  assert torch.allclose(out.float(), ref.float(), atol=3e-2, rtol=3e-2), \
    f"max abs diff {(out.float()-ref.float()).abs().max().item()}"


def test_attn_decode_ragged_lengths(hip):
  """Per-sequence lengths must be honored (flash-decoding split edges)."""
  from xotorch_amd.ops import torch_ref
  B, H, KVH, hd, T = 4, 16, 4, 128, 640
  q = bt(B, 1, H, hd)
  kc, vc = bt(B, KVH, T, hd, seed=8), bt(B, KVH, T, hd, seed=9)
  lens = torch.tensor([1, 17, 333, 640], dtype=torch.int32, device="cuda")
  out = hip.attn_decode(q, kc, vc, lens)
  for b in range(B):
    ref_b = torch_ref.attn_decode(q[b:b+1], kc[b:b+1], vc[b:b+1], int(lens[b]))
    d = (out[b:b+1].float() - ref_b.float()).abs()
    assert torch.allclose(out[b:b+1].float(), ref_b.float(), atol=3e-2, rtol=3e-2), \
      f"b={b} sl={int(lens[b])} maxdiff={d.max().item()} nan_out={out.float().isnan().any().item()}"


def test_swiglu(hip):
  from xotorch_amd.ops import torch_ref
  g, u = bt(4, 3, 14336), bt(4, 3, 14336, seed=10)
  out = hip.swiglu(g, u)
  ref = torch_ref.swiglu(g, u)
  assert torch.allclose(out.float(), ref.float(), atol=2e-2, rtol=2e-2)


def test_swiglu_packed(hip):
  from xotorch_amd.ops import torch_ref
  gu = bt(5, 2, 2 * 4864, seed=11)
  out = hip.swiglu_packed(gu)
  ref = torch_ref.swiglu_packed(gu)
  assert out.shape == (5, 2, 4864)
  assert torch.allclose(out.float(), ref.float(), atol=2e-2, rtol=2e-2)


def test_attn_decode_strided_q(hip):
  """q as a strided view into a packed qkv row (the fused-GEMM layout)."""
  from xotorch_amd.ops import torch_ref
  B, H, KVH, hd, T, sl = 2, 16, 4, 128, 256, 200
  qkv = bt(B, 1, (H + 2 * KVH) * hd, seed=12)
  q = qkv[:, :, : H * hd].view(B, 1, H, hd)
  kc, vc = bt(B, KVH, T, hd, seed=13), bt(B, KVH, T, hd, seed=14)
  lens = torch.full((B,), sl, dtype=torch.int32, device="cuda")
  out = hip.attn_decode(q, kc, vc, lens)
  ref = torch_ref.attn_decode(q.contiguous(), kc, vc, sl)
  assert torch.allclose(out.float(), ref.float(), atol=3e-2, rtol=3e-2)


def test_model_gpu_vs_cpu():
  """Tiny model, bf16 HIP path on GPU vs fp32 torch path on CPU."""
  from xotorch_amd.engine.kvcache import ShardKVCache
  from xotorch_amd.models.config import config_from_hf
  from xotorch_amd.models.llama import ShardedModel
  from xotorch_amd.models.weights import random_init
  from xotorch_amd.shard import Shard
  tiny = {
    "model_type": "llama", "hidden_size": 256, "num_hidden_layers": 4,
    "num_attention_heads": 4, "num_key_value_heads": 2, "intermediate_size": 512,
    "vocab_size": 199, "rope_theta": 10000.0, "rms_norm_eps": 1e-5,
    "max_position_embeddings": 64, "torch_dtype": "float32",
  }
  cfg = config_from_hf(tiny, "tiny")
  shard = Shard("tiny", 0, 3, 4)
  m_cpu = ShardedModel(cfg, shard).float()
  random_init(m_cpu)
  m_cpu.eval()
  m_gpu = ShardedModel(cfg, shard).to("cuda", torch.bfloat16)
  m_gpu.load_state_dict({k: v.to("cuda", torch.bfloat16) for k, v in m_cpu.state_dict().items()})
  m_gpu.rope_cos.copy_(m_cpu.rope_cos.cuda())
  m_gpu.rope_sin.copy_(m_cpu.rope_sin.cuda())
  m_gpu.eval()
  B, S = 2, 9
  tokens = torch.randint(0, 199, (B, S))
  cache_cpu = ShardKVCache(4, B, 2, S + 4, cfg.head_dim, torch.float32, "cpu")
  cache_gpu = ShardKVCache(4, B, 2, S + 4, cfg.head_dim, torch.bfloat16, "cuda")
  with torch.inference_mode():
    pos = torch.arange(S)
    lc = m_cpu(tokens, caches=cache_cpu.caches, positions=pos, start_pos=0)
    lg = m_gpu(tokens.cuda(), caches=cache_gpu.caches, positions=pos.to("cuda", torch.int32), start_pos=0)
    # prefill logits agree within bf16 tolerance
    assert torch.allclose(lc, lg.float().cpu(), atol=0.1, rtol=0.1)
    # one decode step
    nxt = lc.argmax(dim=-1, keepdim=True)
    sl = torch.full((B,), S + 1, dtype=torch.int32, device="cuda")
    lc2 = m_cpu(nxt, caches=cache_cpu.caches, positions=torch.tensor([S]), start_pos=S, is_decode=True)
    lg2 = m_gpu(nxt.cuda(), caches=cache_gpu.caches, positions=torch.tensor([S], dtype=torch.int32, device="cuda"),
                start_pos=S, is_decode=True, seq_lens=sl)
    assert torch.allclose(lc2, lg2.float().cpu(), atol=0.15, rtol=0.15)


@pytest.mark.parametrize("M,K,N,bias", [
  (32, 2048, 256, False),
  (64, 8192, 10240, False),
  (64, 28672, 8192, False),
  (128, 8192, 1152, True),   # qwen-7b qkv shape class (bias)
  (256, 4096, 14336, False),
  (96, 1024, 128, True),     # single n-tile, K not pow2-split friendly
  (64, 8192, 128256, False), # lm_head (large N, splitk=1 path)
])
def test_skinny_gemm(hip, M, K, N, bias):
  from xotorch_amd.ops import _hip_ops
  x = bt(M, K, scale=0.5, seed=M + N)
  w = bt(N, K, scale=0.02, seed=K)
  b = bt(N, seed=5) if bias else None
  got = _hip_ops.skinny_gemm(x, w, b).float()
  ref = torch.nn.functional.linear(x.float(), w.float(), b.float() if bias else None)
  denom = ref.abs().max().item() + 1e-9
  assert (got - ref).abs().max().item() / denom < 2e-2, \
    f"max abs err {(got - ref).abs().max().item()} denom {denom}"


def test_ops_linear_dispatch_matches(hip):
  """ops.linear must agree with F.linear on an eligible decode shape."""
  from xotorch_amd import ops
  x = bt(64, 1, 4096, scale=0.5).view(64, 1, 4096)
  w = bt(512, 4096, scale=0.02, seed=9)
  got = ops.linear(x, w).float()
  ref = torch.nn.functional.linear(x.float(), w.float())
  assert (got - ref).abs().max().item() / (ref.abs().max().item() + 1e-9) < 2e-2


@pytest.mark.parametrize("M,K,N,bias", [
  (64, 8192, 1280, True),
  (64, 28672, 8192, False),
  (256, 4096, 14336, False),
  (32, 1024, 128, False),
])
def test_skinny_gemm_packed(hip, M, K, N, bias):
  from xotorch_amd import ops as xops
  from xotorch_amd.ops import _hip_ops
  x = bt(M, K, scale=0.5, seed=M + N + 1)
  w = bt(N, K, scale=0.02, seed=K + 1)
  b = bt(N, seed=7) if bias else None
  wp = xops.pack_decode_weight(w)
  got = _hip_ops.skinny_gemm_packed(x, wp, N, b).float()
  ref = torch.nn.functional.linear(x.float(), w.float(), b.float() if bias else None)
  denom = ref.abs().max().item() + 1e-9
  assert (got - ref).abs().max().item() / denom < 2e-2


def test_xotlinear_packed_matches_unpacked(hip):
  from xotorch_amd.models.llama import XotLinear
  m = XotLinear(4096, 512, bias=False).to("cuda").to(torch.bfloat16)
  x = bt(64, 1, 4096, scale=0.5)
  with torch.inference_mode():
    y0 = m(x).float()
    m.pack_decode()
    assert m.weight_packed is not None
    y1 = m(x).float()
  assert (y0 - y1).abs().max().item() / (y0.abs().max().item() + 1e-9) < 2e-2


@pytest.mark.parametrize("E,C,K,N", [(8, 64, 512, 256), (128, 32, 256, 128)])
def test_skinny_gemm_grouped(hip, E, C, K, N):
  from xotorch_amd import ops as xops
  from xotorch_amd.ops import _hip_ops
  x = bt(E, C, K, scale=0.5, seed=11)
  ws = [bt(N, K, scale=0.05, seed=100 + e) for e in range(E)]
  wp = torch.stack([xops.pack_decode_weight(w) for w in ws]).contiguous()
  got = _hip_ops.skinny_gemm_grouped(x, wp, E, N).float()
  for e in range(E):
    ref = torch.nn.functional.linear(x[e].float(), ws[e].float())
    err = (got[e] - ref).abs().max().item() / (ref.abs().max().item() + 1e-9)
    assert err < 2e-2, (e, err)


def test_moe_grouped_gpu_matches_loop():
  """Mixtral MoE block: grouped MFMA kernel path vs the dynamic loop."""
  from xotorch_amd.models.config import config_from_hf
  from xotorch_amd.models.registry import builtin_config
  from xotorch_amd.models.llama import MoEMLP
  raw = dict(builtin_config("mixtral-8x7b"))
  raw.update(hidden_size=256, intermediate_size=512)
  cfg = config_from_hf(raw, "mixtral-gpu-tiny")
  torch.manual_seed(2)
  moe = MoEMLP(cfg).to("cuda").to(torch.bfloat16).eval()
  x = (torch.randn(64, 1, 256, device="cuda") * 0.5).to(torch.bfloat16)
  with torch.inference_mode():
    y_unpacked = moe(x).float()  # routed path, per-expert XotLinear (unpacked)
    for e in moe.experts:
      e.gate_up_proj.pack_decode()
      e.down_proj.pack_decode()
    moe.pack_grouped()
    assert moe.wp_gate_up is not None
    y_grouped = moe(x).float()
  err = (y_unpacked - y_grouped).abs().max().item() / (y_unpacked.abs().max().item() + 1e-9)
  assert err < 3e-2, err


def _pack_k(k, t32):
  """Python reference packer for the MFMA K cache layout (tests only)."""
  B, KVH, T, hd = k.shape
  kk = torch.zeros(B, KVH, t32, hd, dtype=k.dtype, device=k.device)
  kk[:, :, :T] = k
  # hd = c*32 + qt*8 + j ; lane = qt*16 + pos16
  v = kk.view(B, KVH, t32 // 16, 16, 4, 4, 8)
  return v.permute(0, 1, 2, 4, 5, 3, 6).contiguous().view(B, KVH, t32 // 16, 4, 64, 8)


def _pack_v(v, t32):
  B, KVH, T, hd = v.shape
  vv = torch.zeros(B, KVH, t32, hd, dtype=v.dtype, device=v.device)
  vv[:, :, :T] = v
  # pos = tp*32 + qt*8 + j ; hd = g*16 + c16 ; lane = qt*16 + c16
  w = vv.view(B, KVH, t32 // 32, 4, 8, 8, 16)
  return w.permute(0, 1, 5, 2, 3, 6, 4).contiguous().view(B, KVH, 8, t32 // 32, 64, 8)


def test_mfma16_probe(hip):
  """Validate the assumed v_mfma_f32_16x16x32_bf16 fragment layout (asymmetric B)."""
  from xotorch_amd.ops import _hip_ops
  a = bt(16, 32, scale=0.5, seed=31)
  b = bt(32, 16, scale=0.5, seed=32)
  d = _hip_ops.mfma16_probe(a, b)
  ref = a.float() @ b.float()
  assert torch.allclose(d, ref, atol=5e-2, rtol=5e-2), (d - ref).abs().max()


@pytest.mark.parametrize("B,H,KVH,T,sl", [(2, 8, 2, 128, 100), (3, 32, 8, 640, 576), (1, 16, 16, 64, 33)])
def test_attn_decode_mfma(hip, B, H, KVH, T, sl):
  from xotorch_amd.ops import _hip_ops, torch_ref
  hd = 128
  q = bt(B, 1, H, hd, seed=41)
  k = bt(B, KVH, T, hd, seed=42)
  v = bt(B, KVH, T, hd, seed=43)
  t32 = (T + 31) // 32 * 32
  kp, vp = _pack_k(k, t32), _pack_v(v, t32)
  sl_t = torch.full((B,), sl, dtype=torch.int32, device="cuda")
  out = _hip_ops.attn_decode_mfma(q, kp, vp, sl_t, T).float()
  ref = torch_ref.attn_decode(q, k, v, sl).float()
  assert torch.allclose(out, ref, atol=3e-2, rtol=3e-2), (out - ref).abs().max()


def test_attn_decode_mfma_ragged(hip):
  from xotorch_amd.ops import _hip_ops, torch_ref
  B, H, KVH, T, hd = 4, 8, 4, 256, 128
  q = bt(B, 1, H, hd, seed=51)
  k = bt(B, KVH, T, hd, seed=52)
  v = bt(B, KVH, T, hd, seed=53)
  t32 = (T + 31) // 32 * 32
  kp, vp = _pack_k(k, t32), _pack_v(v, t32)
  sls = [7, 64, 130, 256]
  sl_t = torch.tensor(sls, dtype=torch.int32, device="cuda")
  out = _hip_ops.attn_decode_mfma(q, kp, vp, sl_t, T).float()
  for b, s in enumerate(sls):
    ref = torch_ref.attn_decode(q[b:b + 1], k[b:b + 1, :, :s], v[b:b + 1, :, :s], s).float()
    assert torch.allclose(out[b:b + 1], ref, atol=3e-2, rtol=3e-2), (b, (out[b:b+1] - ref).abs().max())


def test_rope_append_writes_packed(hip):
  """The fused append kernel's packed-cache writes must match the reference packer."""
  from xotorch_amd.ops import _hip_ops
  B, S, H, KVH, hd, T = 2, 48, 8, 2, 128, 64
  t32 = 64
  qkv = bt(B, S, (H + 2 * KVH) * hd, seed=61)
  cos = torch.randn(T, hd // 2, device="cuda")
  sin = torch.randn(T, hd // 2, device="cuda")
  kc = torch.zeros(B, KVH, T, hd, dtype=torch.bfloat16, device="cuda")
  vc = torch.zeros_like(kc)
  kp = torch.zeros(B, KVH, t32 // 16, 4, 64, 8, dtype=torch.bfloat16, device="cuda")
  vp = torch.zeros(B, KVH, 8, t32 // 32, 64, 8, dtype=torch.bfloat16, device="cuda")
  pos = torch.arange(S, dtype=torch.int32, device="cuda")
  _hip_ops.rope_qkv_append(qkv, cos, sin, pos, kc, vc, H, KVH, hd, kp, vp)
  assert torch.equal(kp, _pack_k(kc, t32))
  assert torch.equal(vp, _pack_v(vc, t32))


def test_quant_fp8_rows(hip):
  from xotorch_amd.ops import _hip_ops
  x = bt(16, 512, scale=3.0, seed=71)
  x8, sx = _hip_ops.quant_fp8_rows(x)
  deq = x8.view(torch.float8_e4m3fn).float() * sx[:, None]
  err = (deq - x.float()).abs().max().item() / x.float().abs().max().item()
  assert err < 0.08, err  # e4m3 relative step ~ 2^-3 near max


@pytest.mark.parametrize("M,K,N,bias", [(64, 8192, 1280, True), (64, 28672, 8192, False),
                                        (128, 4096, 14336, False)])
def test_skinny_gemm_fp8(hip, M, K, N, bias):
  """W8A8 kernel vs the quantize-dequantize fp32 reference (tight tolerance:
  the quantization error itself is excluded by dequantizing the same values)."""
  from xotorch_amd import ops as xops
  from xotorch_amd.ops import _hip_ops
  x = bt(M, K, scale=0.5, seed=M + N + 3)
  w = bt(N, K, scale=0.02, seed=K + 3)
  b = bt(N, seed=8) if bias else None
  wp8, sw = xops.pack_decode_weight_fp8(w)
  x8, sx = _hip_ops.quant_fp8_rows(x)
  got = _hip_ops.skinny_gemm_fp8(x8, sx, wp8, sw, 1, N, b).float()
  w_deq = (w.float() / sw[:, None]).clamp(-448, 448).to(torch.float8_e4m3fn).float() * sw[:, None]
  x_deq = x8.view(torch.float8_e4m3fn).float() * sx[:, None]
  ref = torch.nn.functional.linear(x_deq, w_deq, b.float() if bias else None)
  err = (got - ref).abs().max().item() / (ref.abs().max().item() + 1e-9)
  assert err < 2e-2, err
  # and end-to-end (including quantization error) stays in the W8A8 class
  ref_full = torch.nn.functional.linear(x.float(), w.float(), b.float() if bias else None)
  e2e = (got - ref_full).abs().max().item() / (ref_full.abs().max().item() + 1e-9)
  assert e2e < 0.12, e2e


def test_xotlinear_fp8_mode(hip, monkeypatch):
  from xotorch_amd.models.llama import XotLinear
  monkeypatch.setenv("XOT_FP8_GEMM", "1")
  m = XotLinear(4096, 512, bias=True).to("cuda").to(torch.bfloat16)
  x = bt(64, 1, 4096, scale=0.5)
  with torch.inference_mode():
    y_bf16 = m(x).float()
    m.pack_decode()
    assert m.weight_packed_fp8 is not None
    y_fp8 = m(x).float()
  err = (y_bf16 - y_fp8).abs().max().item() / (y_bf16.abs().max().item() + 1e-9)
  assert err < 0.12, err


@pytest.mark.parametrize("B,H,KVH,S,start", [(2, 8, 2, 128, 0), (1, 32, 8, 512, 0),
                                             (2, 16, 4, 96, 0), (1, 8, 8, 64, 40)])
def test_attn_prefill_mfma(hip, B, H, KVH, S, start):
  """Causal MFMA prefill vs the fp32 torch reference (full prefix + causal
  block; start > 0 = chat continuation with an existing prefix)."""
  from xotorch_amd.ops import _hip_ops, torch_ref
  hd = 128
  total = start + S
  t32 = (total + 31) // 32 * 32
  q = bt(B, S, H, hd, seed=81)
  k = bt(B, KVH, total, hd, seed=82)
  v = bt(B, KVH, total, hd, seed=83)
  kp, vp = _pack_k(k, t32), _pack_v(v, t32)
  out = _hip_ops.attn_prefill_mfma(q, kp, vp, start).float()
  ref = torch_ref.attn_prefill(q, k, v, start, S).float()
  assert torch.allclose(out, ref, atol=3e-2, rtol=3e-2), (out - ref).abs().max()


def test_attn_prefill_mfma_strided_q(hip):
  """q as the strided view into the packed qkv row (the fused-GEMM layout)."""
  from xotorch_amd.ops import _hip_ops, torch_ref
  B, S, H, KVH, hd = 2, 80, 16, 4, 128
  t32 = (S + 31) // 32 * 32
  qkv = bt(B, S, (H + 2 * KVH) * hd, seed=91)
  q = qkv[:, :, : H * hd].view(B, S, H, hd)
  k = bt(B, KVH, S, hd, seed=92)
  v = bt(B, KVH, S, hd, seed=93)
  kp, vp = _pack_k(k, t32), _pack_v(v, t32)
  out = _hip_ops.attn_prefill_mfma(q, kp, vp, 0).float()
  ref = torch_ref.attn_prefill(q.contiguous(), k, v, 0, S).float()
  assert torch.allclose(out, ref, atol=3e-2, rtol=3e-2), (out - ref).abs().max()


def test_moe_fp8_matches_bf16(hip, monkeypatch):
  """fp8 grouped expert path vs the bf16 routed path (W8A8 tolerance)."""
  from xotorch_amd.models.config import config_from_hf
  from xotorch_amd.models.registry import builtin_config
  from xotorch_amd.models.llama import MoEMLP
  raw = dict(builtin_config("mixtral-8x7b"))
  raw.update(hidden_size=256, intermediate_size=512)
  cfg = config_from_hf(raw, "mixtral-fp8-tiny")
  torch.manual_seed(3)
  moe = MoEMLP(cfg).to("cuda").to(torch.bfloat16).eval()
  x = (torch.randn(64, 1, 256, device="cuda") * 0.5).to(torch.bfloat16)
  with torch.inference_mode():
    y_bf16 = moe(x).float()
    monkeypatch.setenv("XOT_FP8_GEMM", "1")
    for e in moe.experts:
      e.gate_up_proj.pack_decode()
      e.down_proj.pack_decode()
    moe.pack_grouped()
    assert moe.wp_gate_up_fp8 is not None
    y_fp8 = moe(x).float()
  err = (y_bf16 - y_fp8).abs().max().item() / (y_bf16.abs().max().item() + 1e-9)
  assert err < 0.15, err


@pytest.mark.parametrize("hd,H,KVH,S", [(128, 8, 2, 4), (64, 4, 2, 1)])
def test_rope_qkv_append_qk_norm(hip, hd, H, KVH, S):
  """Fused qwen3 per-head q/k RMSNorm inside the rope+append kernel."""
  from xotorch_amd.ops import torch_ref
  B, T, start = 2, 32, 5
  qkv = bt(B, S, (H + 2 * KVH) * hd, seed=101)
  cos, sin = torch_ref.rope_cos_sin(hd, T, 1000000.0, device="cuda")
  kc = torch.zeros(B, KVH, T, hd, dtype=torch.bfloat16, device="cuda")
  vc = torch.zeros_like(kc)
  kc_ref, vc_ref = kc.clone(), vc.clone()
  qn = (torch.randn(hd, device="cuda") * 0.2 + 1.0).to(torch.bfloat16)
  kn = (torch.randn(hd, device="cuda") * 0.2 + 1.0).to(torch.bfloat16)
  positions = torch.arange(start, start + S, dtype=torch.int32, device="cuda")
  qkv_ref = qkv.clone()
  torch_ref.rope_qkv_append(qkv_ref, cos, sin, positions, kc_ref, vc_ref, H, KVH, hd, qn, kn, 1e-6)
  hip.rope_qkv_append(qkv, cos, sin, positions, kc, vc, H, KVH, hd, None, None, qn, kn, 1e-6)
  assert torch.allclose(qkv.float(), qkv_ref.float(), atol=3e-2, rtol=3e-2), \
    (qkv.float() - qkv_ref.float()).abs().max()
  assert torch.allclose(kc.float(), kc_ref.float(), atol=3e-2, rtol=3e-2)
  assert torch.equal(vc, vc_ref)


# ---------------- round 2: xreg GEMM + gemma2 attention semantics ----------------

@pytest.mark.parametrize("M,K,N", [(64, 512, 256), (128, 4096, 1280), (128, 2048, 512), (256, 1024, 384)])
def test_skinny_gemm_packed_xreg(hip, M, K, N):
  """v3 register-resident-X kernel vs torch matmul."""
  from xotorch_amd import ops
  from xotorch_amd.ops import _hip_ops
  w = bt(N, K, seed=71, scale=1.0 / K ** 0.5)
  x = bt(M, K, seed=72, scale=0.25)
  wp = ops.pack_decode_weight(w)
  y = _hip_ops.skinny_gemm_packed_xreg(x, wp, N, None).float()
  ref = (x.float() @ w.float().T)
  assert torch.allclose(y, ref, atol=5e-2, rtol=5e-2), (y - ref).abs().max()


def test_skinny_gemm_packed_xreg_bias(hip):
  from xotorch_amd import ops
  from xotorch_amd.ops import _hip_ops
  M, K, N = 128, 1024, 256
  w = bt(N, K, seed=73, scale=1.0 / K ** 0.5)
  b = bt(N, seed=74)
  x = bt(M, K, seed=75, scale=0.25)
  wp = ops.pack_decode_weight(w)
  y = _hip_ops.skinny_gemm_packed_xreg(x, wp, N, b).float()
  ref = (x.float() @ w.float().T) + b.float()
  assert torch.allclose(y, ref, atol=5e-2, rtol=5e-2), (y - ref).abs().max()


@pytest.mark.parametrize("softcap,window", [(50.0, 0), (0.0, 40), (30.0, 40)])
def test_attn_decode_mfma_softcap_window(hip, softcap, window):
  """gemma2 semantics in the MFMA decode kernel vs the torch reference."""
  from xotorch_amd.ops import _hip_ops, torch_ref
  B, H, KVH, T, hd, sl = 2, 8, 4, 128, 128, 100
  scale = 1.0 / 16.0  # query_pre_attn_scalar-style custom scale
  q = bt(B, 1, H, hd, seed=81)
  k = bt(B, KVH, T, hd, seed=82)
  v = bt(B, KVH, T, hd, seed=83)
  t32 = (T + 31) // 32 * 32
  kp, vp = _pack_k(k, t32), _pack_v(v, t32)
  sl_t = torch.full((B,), sl, dtype=torch.int32, device="cuda")
  out = _hip_ops.attn_decode_mfma(q, kp, vp, sl_t, T, scale, softcap, window).float()
  ref = torch_ref.attn_decode(q, k, v, sl, scale, softcap, window).float()
  assert torch.allclose(out, ref, atol=3e-2, rtol=3e-2), (out - ref).abs().max()


@pytest.mark.parametrize("softcap,window,start", [(50.0, 0, 0), (0.0, 48, 0), (30.0, 48, 32)])
def test_attn_prefill_mfma_softcap_window(hip, softcap, window, start):
  from xotorch_amd.ops import _hip_ops, torch_ref
  B, H, KVH, S, hd = 2, 4, 2, 96, 128
  scale = 1.0 / 16.0
  total = start + S
  q = bt(B, S, H, hd, seed=91)
  k = bt(B, KVH, total, hd, seed=92)
  v = bt(B, KVH, total, hd, seed=93)
  t32 = (total + 31) // 32 * 32
  kp, vp = _pack_k(k, t32), _pack_v(v, t32)
  out = _hip_ops.attn_prefill_mfma(q, kp, vp, start, scale, softcap, window).float()
  ref = torch_ref.attn_prefill(q, k, v, start, S, scale, softcap, window).float()
  assert torch.allclose(out, ref, atol=3e-2, rtol=3e-2), (out - ref).abs().max()


def test_rmsnorm_gemma_bias(hip):
  from xotorch_amd.ops import _hip_ops, torch_ref
  x = bt(4, 33, 256, seed=95)
  w = bt(256, seed=96, scale=0.1)
  out = _hip_ops.rmsnorm(x, w, 1e-6, 1.0).float()
  ref = torch_ref.rmsnorm(x.float(), w.float(), 1e-6, 1.0)
  assert torch.allclose(out, ref, atol=2e-2, rtol=2e-2), (out - ref).abs().max()


def test_geglu_packed(hip):
  from xotorch_amd.ops import _hip_ops, torch_ref
  gu = bt(8, 512, seed=97)
  out = _hip_ops.geglu_packed(gu).float()
  ref = torch_ref.geglu_packed(gu).float()
  assert torch.allclose(out, ref, atol=2e-2, rtol=2e-2), (out - ref).abs().max()


def test_gemma2_gpu_hip_path_matches_eager():
  """gemma2 (hd=128 tiny config) on the HIP fast path vs the eager oracle
  path — proves softcap + sliding window + geglu + gemma-norm run the CDNA4
  kernels end-to-end with matching numerics."""
  import os
  from xotorch_amd.engine.kvcache import ShardKVCache
  from xotorch_amd.models.config import config_from_hf
  from xotorch_amd.models.gemma2 import Gemma2Model
  from xotorch_amd.models.weights import random_init
  from xotorch_amd.shard import Shard
  raw = dict(model_type="gemma2", vocab_size=256, hidden_size=256, intermediate_size=512,
             num_hidden_layers=4, num_attention_heads=4, num_key_value_heads=2, head_dim=128,
             sliding_window=32, attn_logit_softcapping=50.0, final_logit_softcapping=30.0,
             query_pre_attn_scalar=256, rms_norm_eps=1e-6, rope_theta=10000.0,
             max_position_embeddings=128, tie_word_embeddings=True)
  cfg = config_from_hf(raw, "gemma2-tiny128")
  shard = Shard("gemma2-tiny128", 0, 3, 4)
  torch.manual_seed(5)
  m = Gemma2Model(cfg, shard).to("cuda").to(torch.bfloat16)
  random_init(m)
  m.reset_rope()
  m.eval()
  B, S = 2, 48
  toks = torch.randint(0, 256, (B, S), device="cuda")
  cache_hip = ShardKVCache(4, B, 2, S + 8, 128, torch.bfloat16, "cuda")
  cache_ref = ShardKVCache(4, B, 2, S + 8, 128, torch.float32, "cpu")
  mc = Gemma2Model(cfg, shard).float()
  mc.load_state_dict({k: v.float().cpu() for k, v in m.state_dict().items()}, strict=False)
  mc.reset_rope()
  mc.eval()
  with torch.inference_mode():
    pos = torch.arange(S, dtype=torch.int32, device="cuda")
    lg = m(toks, caches=cache_hip.caches, positions=pos, start_pos=0)
    lr = mc(toks.cpu(), caches=cache_ref.caches, positions=torch.arange(S), start_pos=0)
    assert (lg.float().cpu().argmax(-1) == lr.argmax(-1)).float().mean() > 0.9
    assert torch.allclose(lg.float().cpu(), lr.float(), atol=0.6, rtol=0.1), (lg.float().cpu() - lr.float()).abs().max()
    # one decode step
    nxt = lg.argmax(-1, keepdim=True)
    sl = torch.full((B,), S + 1, dtype=torch.int32, device="cuda")
    lg2 = m(nxt, caches=cache_hip.caches, positions=torch.tensor([S], dtype=torch.int32, device="cuda"),
            start_pos=S, is_decode=True, seq_lens=sl)
    lr2 = mc(nxt.cpu(), caches=cache_ref.caches, positions=torch.tensor([S]), start_pos=S, is_decode=True)
    assert (lg2.float().cpu().argmax(-1) == lr2.argmax(-1)).all()


# ---------------- MLA (DeepSeek) absorbed-latent decode ----------------

def _mla_pack_ref(lat, rot, T32):
  """Python reference packer for the MLA fragment layouts."""
  B, T = lat.shape[0], lat.shape[1]
  full = torch.cat([lat, rot], dim=-1)  # [B, T, 576]
  kp = torch.zeros(B, T32 // 16, 18, 64, 8, dtype=lat.dtype, device=lat.device)
  vp = torch.zeros(B, 32, T32 // 32, 64, 8, dtype=lat.dtype, device=lat.device)
  for pos in range(T):
    for d in range(576):
      kp[:, pos >> 4, d >> 5, ((d & 31) >> 3) * 16 + (pos & 15), d & 7] = full[:, pos, d]
      if d < 512:
        vp[:, d >> 4, pos >> 5, ((pos & 31) >> 3) * 16 + (d & 15), pos & 7] = full[:, pos, d]
  return kp, vp


def test_mla_append_layout(hip):
  from xotorch_amd.ops import _hip_ops
  B, T = 2, 48
  t32 = 64
  lat = bt(B, T, 512, seed=101)
  rot = bt(B, T, 64, seed=102)
  kp = torch.zeros(B, t32 // 16, 18, 64, 8, dtype=torch.bfloat16, device="cuda")
  vp = torch.zeros(B, 32, t32 // 32, 64, 8, dtype=torch.bfloat16, device="cuda")
  pos = torch.arange(T, dtype=torch.int32, device="cuda")
  _hip_ops.mla_append(lat, rot, pos, kp, vp)
  kp_ref, vp_ref = _mla_pack_ref(lat, rot, t32)
  assert torch.equal(kp, kp_ref)
  assert torch.equal(vp, vp_ref)


@pytest.mark.parametrize("B,H,T,sl", [(2, 8, 64, 50), (1, 16, 128, 128), (3, 4, 96, 33)])
def test_attn_decode_mla_vs_ref(hip, B, H, T, sl):
  from xotorch_amd.ops import _hip_ops
  torch.manual_seed(0)
  lat = bt(B, T, 512, seed=111, scale=0.5)
  rot = bt(B, T, 64, seed=112, scale=0.5)
  q = bt(B, H, 576, seed=113, scale=0.2)
  t32 = (T + 31) // 32 * 32
  kp = torch.zeros(B, t32 // 16, 18, 64, 8, dtype=torch.bfloat16, device="cuda")
  vp = torch.zeros(B, 32, t32 // 32, 64, 8, dtype=torch.bfloat16, device="cuda")
  _hip_ops.mla_append(lat, rot, torch.arange(T, dtype=torch.int32, device="cuda"), kp, vp)
  sl_t = torch.full((B,), sl, dtype=torch.int32, device="cuda")
  scale = 576 ** -0.5
  out = _hip_ops.attn_decode_mla(q, kp, vp, sl_t, scale).float()
  # reference: MQA over [lat | rot], PV over lat
  full = torch.cat([lat, rot], dim=-1).float()[:, :sl]       # [B, sl, 576]
  scores = torch.einsum("bhd,btd->bht", q.float(), full) * scale
  p = torch.softmax(scores, dim=-1)
  ref = torch.einsum("bht,btl->bhl", p, lat.float()[:, :sl])
  assert torch.allclose(out, ref, atol=3e-2, rtol=3e-2), (out - ref).abs().max()


def test_deepseek_gpu_decode_matches_eager():
  """Tiny MLA model (real 512/64 latent dims): GPU HIP decode path vs the
  CPU fp32 eager oracle — prefill + cached decode steps."""
  from xotorch_amd.engine.kvcache import ShardKVCache
  from xotorch_amd.models.config import config_from_hf
  from xotorch_amd.models.deepseek_v3 import DeepseekV3Model
  from xotorch_amd.models.weights import random_init
  from xotorch_amd.shard import Shard
  raw = dict(model_type="deepseek_v3", vocab_size=512, hidden_size=256, intermediate_size=512,
             num_hidden_layers=2, num_attention_heads=8, num_key_value_heads=8,
             kv_lora_rank=512, qk_rope_head_dim=64, qk_nope_head_dim=128, v_head_dim=128,
             q_lora_rank=0, first_k_dense_replace=2, n_routed_experts=0,
             rms_norm_eps=1e-6, rope_theta=10000.0, max_position_embeddings=128)
  cfg = config_from_hf(raw, "ds-tiny")
  shard = Shard("ds-tiny", 0, 1, 2)
  torch.manual_seed(3)
  m = DeepseekV3Model(cfg, shard).to("cuda").to(torch.bfloat16)
  random_init(m)
  m.reset_rope()
  m.eval()
  mc = DeepseekV3Model(cfg, shard).float()
  mc.load_state_dict({k: v.float().cpu() for k, v in m.state_dict().items()}, strict=False)
  mc.reset_rope()
  mc.eval()
  B, S = 2, 33
  toks = torch.randint(0, 512, (B, S), device="cuda")
  heads, kd, vd = cfg.kv_cache_dims()
  cg = ShardKVCache(2, B, heads, S + 8, kd, torch.bfloat16, "cuda", v_dim=vd)
  cc = ShardKVCache(2, B, heads, S + 8, kd, torch.float32, "cpu", v_dim=vd)
  assert cg.caches[0].kp is not None, "MLA packed cache copies missing"
  with torch.inference_mode():
    pos = torch.arange(S, dtype=torch.int32, device="cuda")
    lg = m(toks, caches=cg.caches, positions=pos, start_pos=0)
    lr = mc(toks.cpu(), caches=cc.caches, positions=torch.arange(S), start_pos=0)
    assert (lg.float().cpu().argmax(-1) == lr.argmax(-1)).all()
    nxt = lg.argmax(-1, keepdim=True)
    for step in range(3):
      p = S + step
      lg = m(nxt, caches=cg.caches, positions=torch.tensor([p], dtype=torch.int32, device="cuda"),
             start_pos=p, is_decode=True)
      lr = mc(nxt.cpu(), caches=cc.caches, positions=torch.tensor([p]), start_pos=p, is_decode=True)
      agree = (lg.float().cpu().argmax(-1) == lr.argmax(-1)).float().mean()
      assert agree >= 0.5, (step, agree)
      assert torch.allclose(lg.float().cpu(), lr.float(), atol=0.5, rtol=0.1), \
        (step, (lg.float().cpu() - lr.float()).abs().max())
      nxt = lg.argmax(-1, keepdim=True)


def test_dsmoe_grouped_matches_eager_loop():
  """DeepSeek MoE static grouped decode path vs the dynamic eager loop."""
  from xotorch_amd.models.config import config_from_hf
  from xotorch_amd.models.deepseek_v3 import DsMoE
  raw = dict(model_type="deepseek_v2", vocab_size=64, hidden_size=256, intermediate_size=512,
             moe_intermediate_size=256, n_routed_experts=8, num_experts_per_tok=2,
             n_shared_experts=1, n_group=1, topk_group=1, routed_scaling_factor=1.0,
             norm_topk_prob=False, first_k_dense_replace=0,
             num_attention_heads=4, num_key_value_heads=4, kv_lora_rank=512,
             qk_rope_head_dim=64, qk_nope_head_dim=64, v_head_dim=64,
             rms_norm_eps=1e-6, max_position_embeddings=64)
  cfg = config_from_hf(raw, "dsmoe-tiny")
  torch.manual_seed(11)
  moe = DsMoE(cfg).to("cuda").to(torch.bfloat16)
  with torch.no_grad():
    moe.gate_weight.normal_(0, 0.5)
    for e in moe.experts:
      for p in e.parameters():
        p.normal_(0, 0.05)
    for p in moe.shared_experts.parameters():
      p.normal_(0, 0.05)
  x = (torch.randn(1, 48, 256, device="cuda") * 0.5).to(torch.bfloat16)
  with torch.inference_mode():
    ref = moe(x).float()          # eager dynamic loop (wp not packed yet)
    moe.pack_grouped()
    assert moe.wp_gate_up is not None
    out = moe(x).float()          # static grouped path
  assert torch.allclose(out, ref, atol=5e-2, rtol=5e-2), (out - ref).abs().max()


def test_moe_build_combine_kernels():
  """moe_build/moe_combine vs the torch glue semantics."""
  from xotorch_amd.ops import _hip_ops
  torch.manual_seed(4)
  T, k, E = 48, 4, 16
  C = 64
  idx = torch.randint(0, E, (T, k), dtype=torch.int32, device="cuda")
  w = torch.rand(T, k, dtype=torch.float32, device="cuda")
  gather_tok, inv_pos = _hip_ops.moe_build(idx, E, C)
  # every (t, j) pair landed in its expert's slot range and maps back
  for t in range(T):
    for j in range(k):
      pos = int(inv_pos[t, j])
      e = int(idx[t, j])
      assert e * C <= pos < (e + 1) * C
      assert int(gather_tok[pos]) == t
  # per-expert slot ranks are unique
  assert len(set(int(v) for v in inv_pos.reshape(-1))) == T * k
  # combine = weighted sum over the pair rows
  D = 128
  y = (torch.randn(E * C, D, device="cuda") * 0.5).to(torch.bfloat16)
  out = _hip_ops.moe_combine(y, inv_pos, w).float()
  ref = torch.zeros(T, D, device="cuda")
  for t in range(T):
    for j in range(k):
      ref[t] += w[t, j] * y[int(inv_pos[t, j])].float()
  assert torch.allclose(out, ref, atol=2e-2, rtol=2e-2), (out - ref).abs().max()


@pytest.mark.parametrize("M,I,N", [(64, 512, 256), (128, 1024, 384), (96, 768, 128)])
def test_skinny_gemm_packed_swiglu(hip, M, I, N):
  """Fused silu(gate)*up + down-GEMM vs the two-step reference."""
  from xotorch_amd import ops
  from xotorch_amd.ops import _hip_ops, torch_ref
  gu = bt(M, 2 * I, seed=121, scale=0.5)
  w = bt(N, I, seed=122, scale=1.0 / I ** 0.5)
  wp = ops.pack_decode_weight(w)
  y = _hip_ops.skinny_gemm_packed_swiglu(gu, wp, N, None).float()
  h = torch_ref.swiglu_packed(gu.float())
  ref = h.float() @ w.float().T
  assert torch.allclose(y, ref, atol=5e-2, rtol=5e-2), (y - ref).abs().max()


def test_mla_prep_append_matches_torch():
  """Fused rms+rope+dual-cache-append vs the torch reference path."""
  from xotorch_amd.ops import _hip_ops
  from xotorch_amd.models.deepseek_v3 import _rms, _rope
  from xotorch_amd.ops.torch_ref import rope_cos_sin
  torch.manual_seed(8)
  B, S, T = 2, 20, 32
  t32 = 32
  ckv = bt(B, S, 576, seed=131, scale=0.5)
  w = bt(512, seed=132, scale=0.3) + 1.0
  cos, sin = rope_cos_sin(64, 64, 10000.0, device="cuda")
  lat_c = torch.zeros(B, 1, T, 512, dtype=torch.bfloat16, device="cuda")
  rot_c = torch.zeros(B, 1, T, 64, dtype=torch.bfloat16, device="cuda")
  kp = torch.zeros(B, t32 // 16, 18, 64, 8, dtype=torch.bfloat16, device="cuda")
  vp = torch.zeros(B, 32, t32 // 32, 64, 8, dtype=torch.bfloat16, device="cuda")
  pos = torch.arange(S, dtype=torch.int32, device="cuda")
  for interleave in (False, True):
    _hip_ops.mla_prep_append(ckv, w.contiguous(), cos, sin, pos, lat_c, rot_c, kp, vp,
                             1e-6, interleave)
    lat_ref = _rms(ckv[..., :512], w, 1e-6)
    rot_ref = _rope(ckv[..., 512:].view(B, S, 1, 64), cos[pos.long()], sin[pos.long()],
                    interleave)[:, :, 0, :]
    assert torch.allclose(lat_c[:, 0, :S].float(), lat_ref.float(), atol=2e-2), \
      (lat_c[:, 0, :S].float() - lat_ref.float()).abs().max()
    assert torch.allclose(rot_c[:, 0, :S].float(), rot_ref.float(), atol=2e-2), \
      (rot_c[:, 0, :S].float() - rot_ref.float()).abs().max()
    # packed copies match the reference packer of (lat, rot)
    kp_ref, vp_ref = _mla_pack_ref(lat_c[:, 0, :S], rot_c[:, 0, :S], t32)
    assert torch.equal(kp, kp_ref)
    assert torch.equal(vp, vp_ref)
    kp.zero_(); vp.zero_(); lat_c.zero_(); rot_c.zero_()


def test_moe_route_softmax_mode(hip):
  from xotorch_amd.ops import _hip_ops
  torch.manual_seed(5)
  T, E, k = 48, 64, 2
  logits = torch.randn(T, E, device="cuda", dtype=torch.float32)
  idx, w = _hip_ops.moe_route(logits, None, k, 0)
  ref_w, ref_idx = torch.topk(torch.softmax(logits, dim=-1), k, dim=-1)
  ref_w = ref_w / ref_w.sum(dim=-1, keepdim=True)
  assert torch.equal(idx.long(), ref_idx)
  assert torch.allclose(w, ref_w, atol=1e-5), (w - ref_w).abs().max()


@pytest.mark.parametrize("E,ng,tg,k,norm", [(64, 1, 1, 6, False), (256, 8, 4, 8, True)])
def test_moe_route_deepseek_mode(hip, E, ng, tg, k, norm):
  from xotorch_amd.ops import _hip_ops
  torch.manual_seed(6)
  T = 40
  logits = torch.randn(T, E, device="cuda", dtype=torch.float32)
  bias = torch.randn(E, device="cuda", dtype=torch.float32) * 0.1
  scale = 2.5
  idx, w = _hip_ops.moe_route(logits, bias, k, 1, ng, tg, scale, norm)
  # torch reference (DsMoE.route semantics)
  scores = logits.sigmoid()
  choice = scores + bias
  group_scores = choice.view(T, ng, E // ng).topk(min(2, E // ng), dim=-1)[0].sum(-1)
  gidx = torch.topk(group_scores, k=tg, dim=-1, sorted=False)[1]
  gmask = torch.zeros_like(group_scores).scatter_(1, gidx, 1)
  smask = gmask[:, :, None].expand(-1, ng, E // ng).reshape(T, E)
  masked = choice.masked_fill(~smask.bool(), float("-inf"))
  ref_idx = torch.topk(masked, k=k, dim=-1, sorted=True)[1]
  ref_w = scores.gather(1, ref_idx)
  if norm:
    ref_w = ref_w / (ref_w.sum(dim=-1, keepdim=True) + 1e-20)
  ref_w = ref_w * scale
  # compare as (expert -> weight) sets per row (tie order may differ)
  for t in range(T):
    a = dict(zip(idx[t].tolist(), w[t].tolist()))
    b = dict(zip(ref_idx[t].tolist(), ref_w[t].tolist()))
    assert set(a) == set(b), (t, a, b)
    for e in a:
      assert abs(a[e] - b[e]) < 1e-4, (t, e, a[e], b[e])


# ---------------- fp8 KV cache (opt-in XOT_FP8_KV) ----------------

def test_fp8_kv_append_and_decode(hip):
  """fp8 packed-cache append (e4m3 + per-row scales) and the fp8 decode
  attention vs the bf16 reference path."""
  from xotorch_amd.ops import _hip_ops, torch_ref
  torch.manual_seed(3)
  B, S, H, KVH, hd, T = 2, 40, 8, 4, 128, 64
  t32 = 64
  qkv = bt(B, S, (H + 2 * KVH) * hd, seed=141, scale=0.5)
  qkv8 = qkv.clone()
  kc = torch.zeros(B, KVH, T, hd, dtype=torch.bfloat16, device="cuda")
  vc = torch.zeros_like(kc)
  kc8, vc8 = kc.clone(), vc.clone()
  kp = torch.zeros(B, KVH, t32 // 16, 4, 64, 8, dtype=torch.bfloat16, device="cuda")
  vp = torch.zeros(B, KVH, 8, t32 // 32, 64, 8, dtype=torch.bfloat16, device="cuda")
  kp8 = torch.zeros(B, KVH, t32 // 16, 4, 64, 8, dtype=torch.uint8, device="cuda")
  vp8 = torch.zeros(B, KVH, 8, t32 // 32, 64, 8, dtype=torch.uint8, device="cuda")
  ksc = torch.ones(B, KVH, t32, dtype=torch.float32, device="cuda")
  vsc = torch.ones_like(ksc)
  from xotorch_amd.ops.torch_ref import rope_cos_sin
  cos, sin = rope_cos_sin(hd, T, 10000.0, device="cuda")
  pos = torch.arange(S, dtype=torch.int32, device="cuda")
  _hip_ops.rope_qkv_append(qkv, cos, sin, pos, kc, vc, H, KVH, hd, kp, vp)
  _hip_ops.rope_qkv_append(qkv8, cos, sin, pos, kc8, vc8, H, KVH, hd, kp8, vp8,
                           k_scale=ksc, v_scale=vsc)
  # plain caches identical; fp8 copies dequantize close to the bf16 rows
  assert torch.equal(kc, kc8) and torch.equal(vc, vc8)
  assert (ksc[:, :, :S] != 1.0).any() and (vsc[:, :, :S] != 1.0).any()
  # decode: fp8 path vs bf16 path vs torch reference
  q = qkv[:, -1:, : H * hd].view(B, 1, H, hd).contiguous()
  sl = torch.full((B,), S, dtype=torch.int32, device="cuda")
  out_bf = _hip_ops.attn_decode_mfma(q, kp, vp, sl, T).float()
  out_f8 = _hip_ops.attn_decode_mfma(q, kp8, vp8, sl, T, k_scale=ksc, v_scale=vsc).float()
  ref = torch_ref.attn_decode(q, kc, vc, S).float()
  assert torch.allclose(out_bf, ref, atol=3e-2, rtol=3e-2)
  # fp8 carries e4m3 quantization error (~6% per element): compare loosely
  err = (out_f8 - ref).abs().max().item()
  scale_ref = ref.abs().max().item()
  assert err < 0.12 * max(scale_ref, 1.0) + 0.05, (err, scale_ref)


@pytest.mark.gpu
def test_fp8_kv_end_to_end_model(monkeypatch):
  """XOT_FP8_KV=1 end to end on a tiny hd=128 llama: argmax agreement with
  the bf16-cache run."""
  monkeypatch.setenv("XOT_FP8_KV", "1")
  from xotorch_amd.engine.kvcache import ShardKVCache
  from xotorch_amd.models.config import config_from_hf
  from xotorch_amd.models.llama import ShardedModel
  from xotorch_amd.models.weights import random_init
  from xotorch_amd.shard import Shard
  raw = dict(model_type="llama", vocab_size=512, hidden_size=256, intermediate_size=512,
             num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
             head_dim=128, rms_norm_eps=1e-6, rope_theta=10000.0,
             max_position_embeddings=128)
  cfg = config_from_hf(raw, "fp8kv-tiny")
  shard = Shard("fp8kv-tiny", 0, 1, 2)
  torch.manual_seed(9)
  m = ShardedModel(cfg, shard).to("cuda").to(torch.bfloat16)
  random_init(m)
  m.reset_rope()
  m.eval()
  B, S = 2, 24
  toks = torch.randint(0, 512, (B, S), device="cuda")
  c8 = ShardKVCache(2, B, 2, S + 8, 128, torch.bfloat16, "cuda")
  assert c8.caches[0].kp.dtype == torch.uint8 and c8.caches[0].ksc is not None
  monkeypatch.setenv("XOT_FP8_KV", "0")
  cb = ShardKVCache(2, B, 2, S + 8, 128, torch.bfloat16, "cuda")
  assert cb.caches[0].kp.dtype == torch.bfloat16
  with torch.inference_mode():
    pos = torch.arange(S, dtype=torch.int32, device="cuda")
    l8 = m(toks, caches=c8.caches, positions=pos, start_pos=0)
    lb = m(toks, caches=cb.caches, positions=pos, start_pos=0)
    assert (l8.argmax(-1) == lb.argmax(-1)).all()  # prefill identical (plain cache)
    nxt = lb.argmax(-1, keepdim=True)
    sl = torch.full((B,), S + 1, dtype=torch.int32, device="cuda")
    p1 = torch.tensor([S], dtype=torch.int32, device="cuda")
    d8 = m(nxt, caches=c8.caches, positions=p1, start_pos=S, is_decode=True, seq_lens=sl)
    db = m(nxt, caches=cb.caches, positions=p1, start_pos=S, is_decode=True, seq_lens=sl)
    assert torch.allclose(d8.float(), db.float(), atol=0.5, rtol=0.1), \
      (d8.float() - db.float()).abs().max()


def test_fp8_prefill_scaled_mm(monkeypatch):
  """XOT_FP8_PREFILL: XotLinear prefill GEMMs via e4m3 scaled_mm within
  quantization tolerance of the bf16 path."""
  from xotorch_amd.models.llama import XotLinear
  torch.manual_seed(12)
  lin = XotLinear(512, 1024, bias=True).to("cuda").to(torch.bfloat16)
  x = (torch.randn(2, 300, 512, device="cuda") * 0.3).to(torch.bfloat16)
  with torch.inference_mode():
    ref = lin(x).float()
    monkeypatch.setenv("XOT_FP8_PREFILL", "1")
    out = lin(x).float()
  rel = (out - ref).abs().max().item() / max(ref.abs().max().item(), 1.0)
  assert rel < 0.08, rel
  # decode shapes (M <= 256) stay on the bf16 path
  xd = (torch.randn(64, 512, device="cuda") * 0.3).to(torch.bfloat16)
  with torch.inference_mode():
    d1 = lin(xd)
    monkeypatch.setenv("XOT_FP8_PREFILL", "0")
    d0 = lin(xd)
  assert torch.equal(d1, d0)


def test_mla_q_prep(hip):
  from xotorch_amd.ops import _hip_ops
  from xotorch_amd.models.deepseek_v3 import _rope
  from xotorch_amd.ops.torch_ref import rope_cos_sin
  torch.manual_seed(13)
  B, H, nope = 3, 8, 128
  q = bt(B, 1, H, nope + 64, seed=151, scale=0.4)
  q_lat = bt(B, 1, H, 512, seed=152, scale=0.4)
  cos, sin = rope_cos_sin(64, 128, 10000.0, device="cuda")
  for npos, pos in ((1, torch.tensor([37], dtype=torch.int32, device="cuda")),
                    (B, torch.tensor([5, 90, 44], dtype=torch.int32, device="cuda"))):
    for interleave in (False, True):
      qf = _hip_ops.mla_q_prep(q, q_lat, cos, sin, pos, nope, interleave)[0]
      assert torch.equal(qf[:, :, :512], q_lat.view(B, H, 512))
      cs = cos[pos.long()]
      sn = sin[pos.long()]
      if npos == B:
        cs, sn = cs.view(B, 1, -1), sn.view(B, 1, -1)
      ref = _rope(q[:, :, :, nope:], cs, sn, interleave)[:, 0]  # [B, H, 64]
      assert torch.allclose(qf[:, :, 512:].float(), ref.float(), atol=2e-2), \
        (npos, interleave, (qf[:, :, 512:].float() - ref.float()).abs().max())


def test_mla_fp8_decode(hip, monkeypatch):
  """fp8 MLA latent cache: fused prep quantizes (single per-token scale),
  fp8 absorbed decode within quantization tolerance of the bf16 path, and
  the tiny end-to-end model agrees under XOT_FP8_KV=1."""
  from xotorch_amd.ops import _hip_ops
  torch.manual_seed(21)
  B, T, S = 2, 64, 40
  ckv = bt(B, S, 576, seed=161, scale=0.5)
  w = bt(512, seed=162, scale=0.3) + 1.0
  from xotorch_amd.ops.torch_ref import rope_cos_sin
  cos, sin = rope_cos_sin(64, T, 10000.0, device="cuda")
  pos = torch.arange(S, dtype=torch.int32, device="cuda")
  def mk(fp8):
    lat_c = torch.zeros(B, 1, T, 512, dtype=torch.bfloat16, device="cuda")
    rot_c = torch.zeros(B, 1, T, 64, dtype=torch.bfloat16, device="cuda")
    dt = torch.uint8 if fp8 else torch.bfloat16
    kp = torch.zeros(B, T // 16, 18, 64, 8, dtype=dt, device="cuda")
    vp = torch.zeros(B, 32, T // 32, 64, 8, dtype=dt, device="cuda")
    ks = torch.ones(B, T, dtype=torch.float32, device="cuda") if fp8 else None
    _hip_ops.mla_prep_append(ckv, w.contiguous(), cos, sin, pos, lat_c, rot_c, kp, vp,
                             1e-6, False, ks)
    return lat_c, rot_c, kp, vp, ks
  lat_b, rot_b, kp_b, vp_b, _ = mk(False)
  lat_8, rot_8, kp_8, vp_8, ks = mk(True)
  assert torch.equal(lat_b, lat_8) and torch.equal(rot_b, rot_8)  # plain caches identical
  assert (ks[:, :S] != 1.0).any()
  q = bt(B, 8, 576, seed=163, scale=0.3)
  sl = torch.full((B,), S, dtype=torch.int32, device="cuda")
  out_b = _hip_ops.attn_decode_mla(q, kp_b, vp_b, sl, 576 ** -0.5).float()
  # quantize q the same way the model path does (mla_q_prep fp8): here by hand
  sq = q.float().abs().amax(dim=-1).clamp(min=1e-12) / 448.0
  q8 = (q.float() / sq[..., None]).clamp(-448, 448).to(torch.float8_e4m3fn).view(torch.uint8)
  out_8 = _hip_ops.attn_decode_mla(q8, kp_8, vp_8, sl, 576 ** -0.5,
                                   sq.contiguous(), ks).float()
  err = (out_8 - out_b).abs().max().item()
  ref_scale = out_b.abs().max().item()
  assert err < 0.12 * max(ref_scale, 1.0) + 0.05, (err, ref_scale)


@pytest.mark.gpu
def test_spec_self_draft_full_acceptance():
  """Draft == target (same seed -> identical weights): acceptance must stay
  ~1.0 across MANY rounds. Catches the draft-cache hole regression (a fully
  accepted round used to leave one unwritten KV row in the draft cache,
  collapsing acceptance to ~0.06 from round 2 on)."""
  import torch
  from xotorch_amd.engine.spec import SpeculativeDecoder
  sd = SpeculativeDecoder.from_model_ids("llama-3-8b", "llama-3-8b",
                                         device="cuda", dtype=torch.bfloat16, gamma=4)
  g = torch.Generator().manual_seed(5)
  prompt = torch.randint(0, 32000, (1, 64), generator=g)
  toks, stats = sd.generate(prompt, max_new=64)
  assert stats.rounds >= 10
  assert stats.accept_rate > 0.9, f"accept {stats.accept_rate:.2f} ({stats.accepted}/{stats.proposed})"


@pytest.mark.gpu
def test_deepseek_yarn_gpu_matches_eager():
  """YaRN rope scaling flows through the HIP MLA path (host cos/sin tables +
  softmax-scale param reach mla_prep_append / mla_q_prep / attn_decode_mla):
  GPU bf16 decode vs the CPU fp32 eager oracle with the same yarn config."""
  from xotorch_amd.engine.kvcache import ShardKVCache
  from xotorch_amd.models.config import config_from_hf
  from xotorch_amd.models.deepseek_v3 import DeepseekV3Model
  from xotorch_amd.models.weights import random_init
  from xotorch_amd.shard import Shard
  raw = dict(model_type="deepseek_v3", vocab_size=512, hidden_size=256, intermediate_size=512,
             num_hidden_layers=2, num_attention_heads=8, num_key_value_heads=8,
             kv_lora_rank=512, qk_rope_head_dim=64, qk_nope_head_dim=128, v_head_dim=128,
             q_lora_rank=0, first_k_dense_replace=2, n_routed_experts=0,
             rms_norm_eps=1e-6, rope_theta=10000.0, max_position_embeddings=256,
             rope_scaling={"rope_type": "yarn", "factor": 8.0, "beta_fast": 32,
                           "beta_slow": 1, "mscale": 0.707, "mscale_all_dim": 0.707,
                           "original_max_position_embeddings": 32})
  cfg = config_from_hf(raw, "ds-yarn")
  assert cfg.rope_scaling is not None and cfg.rope_scaling.rope_type == "yarn"
  shard = Shard("ds-yarn", 0, 1, 2)
  torch.manual_seed(5)
  m = DeepseekV3Model(cfg, shard).to("cuda").to(torch.bfloat16)
  random_init(m)
  m.reset_rope()
  m.eval()
  base_scale = (cfg.qk_nope_head_dim + cfg.qk_rope_head_dim) ** -0.5
  assert m.layers["0"].self_attn.scale > base_scale  # mscale^2 correction applied
  mc = DeepseekV3Model(cfg, shard).float()
  mc.load_state_dict({k: v.float().cpu() for k, v in m.state_dict().items()}, strict=False)
  mc.reset_rope()
  mc.eval()
  B, S = 2, 48  # prefill past the 32-token pretraining window
  toks = torch.randint(0, 512, (B, S), device="cuda")
  heads, kd, vd = cfg.kv_cache_dims()
  cg = ShardKVCache(2, B, heads, S + 8, kd, torch.bfloat16, "cuda", v_dim=vd)
  cc = ShardKVCache(2, B, heads, S + 8, kd, torch.float32, "cpu", v_dim=vd)
  with torch.inference_mode():
    pos = torch.arange(S, dtype=torch.int32, device="cuda")
    lg = m(toks, caches=cg.caches, positions=pos, start_pos=0)
    lr = mc(toks.cpu(), caches=cc.caches, positions=torch.arange(S), start_pos=0)
    assert (lg.float().cpu().argmax(-1) == lr.argmax(-1)).all()
    nxt = lg.argmax(-1, keepdim=True)
    for step in range(3):
      p = S + step
      lg = m(nxt, caches=cg.caches, positions=torch.tensor([p], dtype=torch.int32, device="cuda"),
             start_pos=p, is_decode=True)
      lr = mc(nxt.cpu(), caches=cc.caches, positions=torch.tensor([p]), start_pos=p, is_decode=True)
      assert torch.allclose(lg.float().cpu(), lr.float(), atol=0.5, rtol=0.1), \
        (step, (lg.float().cpu() - lr.float()).abs().max())
      nxt = lg.argmax(-1, keepdim=True)
