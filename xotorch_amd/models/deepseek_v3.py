"""Sharded DeepSeek-V3 / R1 decoder: Multi-head Latent Attention + MoE.

A separate decoder (like models/gemma2.py) — MLA shares nothing with the GQA
hot path: queries go through a LoRA bottleneck (q_a → norm → q_b), keys and
values are EXPANDED from a compressed latent (kv_lora_rank + a single shared
rope head), and only the LATENT is cached — the KV cache is
(kv_lora_rank + qk_rope_head_dim) elements per token (576 for the real V3)
instead of n_kv_heads*head_dim, MLA's whole point. MoE adds sigmoid routing
with e-score correction bias, group-limited top-k (n_group/topk_group),
routed scaling and always-on shared experts; the first k layers are dense.

The reference lists deepseek-v3 / r1 cards its torchtune GQA assembly cannot
run; here the family runs the absorbed-latent
MLA MFMA decode path on GPU (hip_ops.hip: mla_prep_append / mla_q_prep /
attn_decode_mla) and a plain-torch path on CPU, both validated against
transformers' DeepseekV3ForCausalLM on tiny random configs (the full 671B
needs the 8-GPU ring). YaRN long-context rope scaling is implemented
(NTK-by-parts frequency blend + mscale corrections, validated against
transformers).

Cache contract: caches[idx] is a LayerKV whose k tensor stores the kv_nope
latent [B, 1, T, kv_lora_rank] and v tensor stores the roped shared key
[B, 1, T, qk_rope_head_dim] (see ModelConfig.kv_cache_dims()).
"""
from __future__ import annotations

from typing import List

import torch
import torch.nn as nn

from xotorch_amd.models.config import ModelConfig
from xotorch_amd.models.llama import XotLinear
from xotorch_amd.ops.torch_ref import rope_cos_sin
from xotorch_amd.shard import Shard


def _rms(x: torch.Tensor, w: torch.Tensor, eps: float) -> torch.Tensor:
  if x.is_cuda and x.dtype == torch.bfloat16 and w.dtype == torch.bfloat16 \
     and x.shape[-1] % 8 == 0 and x.shape[-1] <= 16384:
    from xotorch_amd import ops
    return ops.rmsnorm(x.contiguous(), w, eps)  # one HIP launch vs ~4 torch ops
  xf = x.float()
  out = xf * torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + eps)
  return (w.float() * out).to(x.dtype)


def _rope(x: torch.Tensor, cos: torch.Tensor, sin: torch.Tensor, interleave: bool) -> torch.Tensor:
  """x: [B, S, H, D]; cos/sin: [S, D/2] shared across batch, or [B, S, D/2]
  per-row (continuous-batching slots decode at different positions)."""
  if cos.dim() == 3:
    c = cos[:, :, None, :].float()
    s = sin[:, :, None, :].float()
  else:
    c = cos[None, :, None, :].float()
    s = sin[None, :, None, :].float()
  xf = x.float()
  if interleave:
    x1, x2 = xf[..., 0::2], xf[..., 1::2]
  else:
    half = x.shape[-1] // 2
    x1, x2 = xf[..., :half], xf[..., half:]
  return torch.cat([x1 * c - x2 * s, x2 * c + x1 * s], dim=-1).to(x.dtype)


class MLAttention(nn.Module):
  def __init__(self, cfg: ModelConfig):
    super().__init__()
    self.cfg = cfg
    D, H = cfg.dim, cfg.n_heads
    self.qk_head_dim = cfg.qk_nope_head_dim + cfg.qk_rope_head_dim
    if cfg.q_lora_rank:
      self.q_a_proj = nn.Linear(D, cfg.q_lora_rank, bias=cfg.attn_bias)
      self.q_a_layernorm = nn.Parameter(torch.ones(cfg.q_lora_rank))
      self.q_b_proj = XotLinear(cfg.q_lora_rank, H * self.qk_head_dim, bias=False)
    else:
      self.q_proj = XotLinear(D, H * self.qk_head_dim, bias=False)
    self.kv_a_proj_with_mqa = nn.Linear(D, cfg.kv_lora_rank + cfg.qk_rope_head_dim, bias=cfg.attn_bias)
    self.kv_a_layernorm = nn.Parameter(torch.ones(cfg.kv_lora_rank))
    self.kv_b_proj = nn.Linear(cfg.kv_lora_rank, H * (cfg.qk_nope_head_dim + cfg.v_head_dim), bias=False)
    self.o_proj = XotLinear(H * cfg.v_head_dim, D, bias=cfg.attn_bias)
    self.scale = self.qk_head_dim ** -0.5
    rs = cfg.rope_scaling
    if rs is not None and rs.rope_type == "yarn" and rs.mscale_all_dim and rs.factor > 1:
      # yarn softmax-scale correction (HF yarn_apply_mscale)
      import math
      m = 0.1 * rs.mscale_all_dim * math.log(rs.factor) + 1.0
      self.scale *= m * m

  def forward(self, x, cos, sin, positions, kv, start_pos: int,
              is_decode: bool = False, seq_lens=None):
    cfg = self.cfg
    B, S, _ = x.shape
    H = cfg.n_heads
    nope, rope_d, vd = cfg.qk_nope_head_dim, cfg.qk_rope_head_dim, cfg.v_head_dim
    if cfg.q_lora_rank:
      q = self.q_b_proj(_rms(self.q_a_proj(x), self.q_a_layernorm, cfg.norm_eps))
    else:
      q = self.q_proj(x)
    q = q.view(B, S, H, self.qk_head_dim)
    q_pass, q_rot = q[..., :nope], q[..., nope:]

    ckv = self.kv_a_proj_with_mqa(x)

    kp = kv[2] if len(kv) > 2 else None
    hip = None
    if kp is not None and x.is_cuda and x.dtype == torch.bfloat16:
      from xotorch_amd.ops import _load_hip
      hip = _load_hip()
    decode = is_decode or (S == 1 and (start_pos > 0 or start_pos < 0))
    if hip is not None:
      # fused KV-side prep: latent RMSNorm + shared-key rope + plain AND
      # fragment-packed cache writes in ONE launch (device positions —
      # graph-capturable; replaces ~10 torch launches per layer)
      hip.mla_prep_append(ckv.contiguous(), self.kv_a_layernorm, cos, sin,
                          positions.to(torch.int32).contiguous(), kv[0], kv[1],
                          kp, kv[3], cfg.norm_eps, cfg.rope_interleave,
                          kv[4] if len(kv) > 4 else None)
      if decode and not torch.is_grad_enabled():
        if seq_lens is None:
          seq_lens = torch.full((B,), start_pos + 1, dtype=torch.int32, device=x.device)
        return self._decode_mfma(x, q, q_pass, kv, seq_lens, hip, positions, cos, sin)
      if start_pos < 0:
        start_pos = int(positions.reshape(-1)[0])
      if not torch.is_grad_enabled():
        # GPU prefill: query-chunked bf16 attention (the eager path below
        # materializes [B,H,S,T] fp32 scores — 67 GB transient per layer
        # at B=64, S=512)
        q_rot = self._rope_q(q_rot, cos, sin, positions, B, S)
        return self._prefill_absorbed(x, q_pass, q_rot, kv, start_pos, S)
      q_rot = self._rope_q(q_rot, cos, sin, positions, B, S)
    else:
      q_rot = self._rope_q(q_rot, cos, sin, positions, B, S)
      pidx = positions.reshape(-1).long()
      cs, sn = cos[pidx], sin[pidx]
      if pidx.numel() == B * S and pidx.numel() != S:
        cs = cs.view(B, S, -1)
        sn = sn.view(B, S, -1)
      kv_nope, k_rot = ckv[..., : cfg.kv_lora_rank], ckv[..., cfg.kv_lora_rank:]
      kv_nope = _rms(kv_nope, self.kv_a_layernorm, cfg.norm_eps)
      k_rot = _rope(k_rot.view(B, S, 1, rope_d), cs, sn, cfg.rope_interleave)
      if start_pos < 0:  # ring/serve decode contract: derive from positions
        prow = positions.reshape(-1)
        if prow.numel() == B * S and S == 1 and B > 1 and int(prow.min()) != int(prow.max()):
          # ragged per-row positions on the EAGER path (continuous-batching
          # slots on CPU / no packed cache): correctness-first row loop
          outs = []
          for b in range(B):
            outs.append(self.forward(x[b:b + 1], cos, sin, prow[b:b + 1],
                                     tuple(t[b:b + 1] if t is not None else None for t in kv),
                                     int(prow[b]), is_decode, None))
          return torch.cat(outs, dim=0)
        start_pos = int(prow[0])
      # latent cache: k tensor <- kv_nope [B,1,T,kv_lora], v tensor <- roped
      # shared key [B,1,T,rope_d]
      lat_c, rot_c = kv[0], kv[1]
      lat_c[:, 0, start_pos: start_pos + S] = kv_nope
      rot_c[:, 0, start_pos: start_pos + S] = k_rot[:, :, 0, :]
    lat_c, rot_c = kv[0], kv[1]
    total = start_pos + S
    lat = lat_c[:, 0, :total]                                   # [B, T, kv_lora]
    krot = rot_c[:, 0, :total]                                  # [B, T, rope_d]

    # expand latent -> per-head k_nope & v
    kvx = self.kv_b_proj(lat).view(B, total, H, nope + vd)
    k_nope, v = kvx[..., :nope], kvx[..., nope:]

    qf = torch.cat([q_pass, q_rot], dim=-1).float()             # [B, S, H, qk]
    kf = torch.cat([k_nope.float(), krot[:, :, None, :].expand(B, total, H, rope_d).float()], dim=-1)
    scores = torch.einsum("bshd,bthd->bhst", qf, kf) * self.scale
    qpos = torch.arange(start_pos, total, device=x.device)[:, None]
    kpos = torch.arange(0, total, device=x.device)[None, :]
    scores = scores.masked_fill((kpos > qpos)[None, None], float("-inf"))
    probs = torch.softmax(scores, dim=-1)
    out = torch.einsum("bhst,bthd->bshd", probs, v.float()).to(x.dtype)
    return self.o_proj(out.reshape(B, S, H * vd))


  def _rope_q(self, q_rot, cos, sin, positions, B, S):
    pidx = positions.reshape(-1).long()
    cs, sn = cos[pidx], sin[pidx]
    if pidx.numel() == B * S and pidx.numel() != S:
      cs = cs.view(B, S, -1)
      sn = sn.view(B, S, -1)
    return _rope(q_rot, cs, sn, self.cfg.rope_interleave)

  def _prefill_absorbed(self, x, q_pass, q_rot, kv, start_pos: int, S: int):
    """GPU prefill: expand the latent cache through kv_b into per-head
    K/V (qk dim 192 -> flash-eligible) and run sdpa causal; when the
    query block is offset inside a longer cache (chunked serve prefill),
    run query-chunked bf16 matmuls with an fp32 softmax so the O(S*T)
    score tile never materializes in fp32 for the whole batch (measured
    879->726 ms whole-batch TTFT at B=64 S=512 vs sdpa, whose flash
    backend refuses head_dim 192 on ROCm and falls to math)."""
    cfg = self.cfg
    B, H = x.shape[0], cfg.n_heads
    nope, vd = cfg.qk_nope_head_dim, cfg.v_head_dim
    total = start_pos + S
    lat = kv[0][:, 0, :total]                                   # [B,T,512]
    rot = kv[1][:, 0, :total]                                   # [B,T,64]
    kvx = self.kv_b_proj(lat).view(B, total, H, nope + vd)
    k = torch.cat([kvx[..., :nope],
                   rot[:, :, None, :].expand(B, total, H, -1)], dim=-1)
    k = k.permute(0, 2, 1, 3)                                   # [B,H,T,192]
    v = kvx[..., nope:].permute(0, 2, 1, 3)                     # [B,H,T,128]
    q = torch.cat([q_pass.to(x.dtype), q_rot.to(x.dtype)], dim=-1)
    q = q.permute(0, 2, 1, 3)                                   # [B,H,S,192]
    outs = []
    for s0 in range(0, S, 128):
      qc = q[:, :, s0:s0 + 128]
      t_end = start_pos + s0 + qc.shape[2]
      sc = torch.matmul(qc, k[:, :, :t_end].transpose(-1, -2)).float() * self.scale
      qi = torch.arange(start_pos + s0, t_end, device=x.device)
      sc.masked_fill_(qi[:, None] < torch.arange(t_end, device=x.device)[None, :], float("-inf"))
      outs.append(torch.matmul(sc.softmax(-1).to(x.dtype), v[:, :, :t_end]))
    out = torch.cat(outs, dim=2)
    out = out.permute(0, 2, 1, 3).reshape(B, S, H * vd)
    return self.o_proj(out)

  def _decode_mfma(self, x, q_raw, q_pass, kv, seq_lens, hip, positions, cos, sin):
    """Absorbed-latent MFMA decode: kv_b is folded into q and out, so
    attention runs as MQA over the packed 1152 B/token latent stream
    (hip_ops.hip attn_decode_mla); the q tail rope + assembly is one
    kernel (mla_q_prep)."""
    cfg = self.cfg
    B, S, H = q_pass.shape[0], q_pass.shape[1], cfg.n_heads
    nope, vd, lat = cfg.qk_nope_head_dim, cfg.v_head_dim, cfg.kv_lora_rank
    if not hasattr(self, "_w_k"):
      W = self.kv_b_proj.weight.view(H, nope + vd, lat)
      self._w_k = W[:, :nope, :].contiguous()   # [H, nope, lat]
      self._w_v = W[:, nope:, :].contiguous()   # [H, vd, lat]
    q_lat = torch.einsum("bshn,hnl->bshl", q_pass.to(x.dtype), self._w_k)
    fp8 = kv[2].dtype == torch.uint8
    prep = hip.mla_q_prep(q_raw.contiguous(), q_lat.contiguous(), cos, sin,
                          positions.to(torch.int32).contiguous(),
                          nope, cfg.rope_interleave, fp8)
    if fp8:
      out_lat = hip.attn_decode_mla(prep[0], kv[2], kv[3], seq_lens, self.scale,
                                    prep[1], kv[4])
    else:
      out_lat = hip.attn_decode_mla(prep[0], kv[2], kv[3], seq_lens, self.scale)
    out = torch.einsum("bhl,hdl->bhd", out_lat, self._w_v).to(x.dtype)
    return self.o_proj(out.reshape(B, 1, H * vd))


class DsMLP(nn.Module):
  def __init__(self, cfg: ModelConfig, intermediate: int):
    super().__init__()
    D = cfg.dim
    self.gate_proj = XotLinear(D, intermediate, bias=False)
    self.up_proj = XotLinear(D, intermediate, bias=False)
    self.down_proj = XotLinear(intermediate, D, bias=False)

  def forward(self, x):
    return self.down_proj(nn.functional.silu(self.gate_proj(x)) * self.up_proj(x))


class DsMoE(nn.Module):
  """Sigmoid-routed MoE with e-score correction bias, group-limited top-k,
  routed scaling and shared experts (transformers DeepseekV3MoE semantics)."""

  def __init__(self, cfg: ModelConfig):
    super().__init__()
    self.cfg = cfg
    E = cfg.n_experts
    self.gate_weight = nn.Parameter(torch.zeros(E, cfg.dim))
    self.register_buffer("e_score_correction_bias", torch.zeros(E))
    self.experts = nn.ModuleList([DsMLP(cfg, cfg.moe_intermediate_dim) for _ in range(E)])
    self.shared_experts = DsMLP(cfg, cfg.moe_intermediate_dim * max(1, cfg.n_shared_experts))

  def route(self, flat):
    cfg = self.cfg
    E, k = cfg.n_experts, cfg.n_experts_per_tok
    logits = nn.functional.linear(flat.float(), self.gate_weight.float())
    scores = logits.sigmoid()
    choice = scores + self.e_score_correction_bias.float()
    ng = max(1, cfg.n_group)
    group_scores = choice.view(-1, ng, E // ng).topk(min(2, E // ng), dim=-1)[0].sum(-1)
    gidx = torch.topk(group_scores, k=max(1, cfg.topk_group), dim=-1, sorted=False)[1]
    gmask = torch.zeros_like(group_scores).scatter_(1, gidx, 1)
    smask = gmask[:, :, None].expand(-1, ng, E // ng).reshape(-1, E)
    choice = choice.masked_fill(~smask.bool(), float("-inf"))
    idx = torch.topk(choice, k=k, dim=-1, sorted=False)[1]
    w = scores.gather(1, idx)
    if cfg.norm_topk_prob:
      w = w / (w.sum(dim=-1, keepdim=True) + 1e-20)
    return idx, w * cfg.routed_scaling_factor

  wp_gate_up = None
  wp_down = None

  def pack_grouped(self):
    """Stack per-expert [gate|up] and down prepacks for the single-launch
    grouped MFMA decode GEMM (same machinery as llama MoEMLP)."""
    if self.wp_gate_up is not None:
      return
    first = self.experts[0].gate_proj.weight
    if not (first.is_cuda and first.dtype == torch.bfloat16):
      return
    from xotorch_amd import ops
    gu, dn = [], []
    for e in self.experts:
      w = torch.cat([e.gate_proj.weight.detach(), e.up_proj.weight.detach()], dim=0)
      gu.append(ops.pack_decode_weight(w))
      dn.append(ops.pack_decode_weight(e.down_proj.weight.detach()))
    self.wp_gate_up = torch.stack(gu).contiguous()
    self.wp_down = torch.stack(dn).contiguous()

  def _forward_decode(self, flat):
    """Static-shape routed path (graph-capturable): sort token-expert pairs,
    pad to a fixed per-expert capacity, one grouped MFMA launch per GEMM."""
    from xotorch_amd import ops
    from xotorch_amd.ops import _load_hip
    hip = _load_hip()
    cfg = self.cfg
    T, D = flat.shape
    E, k = cfg.n_experts, cfg.n_experts_per_tok
    dev = flat.device
    cfg = self.cfg
    logits = nn.functional.linear(flat.float(), self.gate_weight.float())
    idx, w = hip.moe_route(logits.contiguous(), self.e_score_correction_bias.float().contiguous(),
                           cfg.n_experts_per_tok, 1, max(1, cfg.n_group),
                           max(1, cfg.topk_group), cfg.routed_scaling_factor,
                           bool(cfg.norm_topk_prob))
    C = max(32, -(-T // 32) * 32)
    # counting-sort routing + deterministic combine as single HIP launches
    # (the torch glue — argsort/cumsum/index_add — was the measured top
    # decode cost, profiles/r02_new_decoders_profile.md)
    gather_tok, inv_pos = hip.moe_build(idx.to(torch.int32).contiguous(), E, C)
    xg = flat[gather_tok.long()].to(flat.dtype)               # [E*C, D]
    I = cfg.moe_intermediate_dim
    gu = hip.skinny_gemm_grouped(xg.view(E, C, D).contiguous(), self.wp_gate_up, E, 2 * I)
    h = ops.swiglu_packed(gu.view(E * C, 2 * I))
    y = hip.skinny_gemm_grouped(h.view(E, C, I), self.wp_down, E, D).view(E * C, D)
    return hip.moe_combine(y, inv_pos, w.float().contiguous())

  def forward(self, x):
    B, S, D = x.shape
    flat = x.view(-1, D)
    T = flat.shape[0]
    if (self.wp_gate_up is not None and x.is_cuda and not torch.is_grad_enabled()
        and 1 <= T <= 256):
      out = self._forward_decode(flat) + self.shared_experts(flat)
      return out.to(x.dtype).view(B, S, D)
    if torch.cuda.is_available() and x.is_cuda and torch.cuda.is_current_stream_capturing():
      raise RuntimeError(
        "DsMoE dynamic routing is not graph-capturable: call "
        "model.pack_decode_weights() first (grouped static path), or bench "
        "with --no-graphs")
    idx, w = self.route(flat)
    if x.is_cuda and not torch.is_grad_enabled():
      # prefill: sorted contiguous per-expert slices (one argsort + gather
      # instead of E where/eq scans)
      k = self.cfg.n_experts_per_tok
      dev = flat.device
      expert_of = idx.reshape(-1)
      order = torch.argsort(expert_of)
      tok_sorted = torch.arange(T, device=dev).repeat_interleave(k)[order]
      w_sorted = w.reshape(-1)[order]
      counts = torch.bincount(expert_of, minlength=self.cfg.n_experts).cpu().tolist()
      xg = flat[tok_sorted]
      ys = []
      off = 0
      for e, c in enumerate(counts):
        if c:
          ys.append(self.experts[e](xg[off:off + c]))
        off += c
      y = torch.cat(ys, dim=0).float() * w_sorted[:, None].float()
      out = torch.zeros(T, D, dtype=torch.float32, device=dev)
      out.index_add_(0, tok_sorted, y)
      out = out.to(x.dtype) + self.shared_experts(flat)
      return out.view(B, S, D)
    out = torch.zeros_like(flat, dtype=torch.float32)
    for e in range(self.cfg.n_experts):
      tok, kk = torch.where(idx == e)
      if tok.numel() == 0:
        continue
      out.index_add_(0, tok, self.experts[e](flat[tok]).float() * w[tok, kk, None])
    out = out.to(x.dtype) + self.shared_experts(flat)
    return out.view(B, S, D)


class DsLayer(nn.Module):
  def __init__(self, cfg: ModelConfig, layer_idx: int):
    super().__init__()
    self.eps = cfg.norm_eps
    self.self_attn = MLAttention(cfg)
    dense = layer_idx < cfg.first_k_dense_replace or cfg.n_experts == 0
    self.mlp = DsMLP(cfg, cfg.intermediate_dim) if dense else DsMoE(cfg)
    self.input_layernorm = nn.Parameter(torch.ones(cfg.dim))
    self.post_attention_layernorm = nn.Parameter(torch.ones(cfg.dim))

  def forward(self, h, cos, sin, positions, kv, start_pos, is_decode=False, seq_lens=None):
    h = h + self.self_attn(_rms(h, self.input_layernorm, self.eps), cos, sin, positions, kv,
                           start_pos, is_decode, seq_lens)
    return h + self.mlp(_rms(h, self.post_attention_layernorm, self.eps))


class DeepseekV3Model(nn.Module):
  """Layer range [shard.start_layer .. shard.end_layer]; engine-contract
  forward (same as ShardedModel / Gemma2Model)."""

  def __init__(self, cfg: ModelConfig, shard: Shard):
    super().__init__()
    self.cfg = cfg
    self.shard = shard
    if shard.is_first_layer:
      self.embed_tokens = nn.Embedding(cfg.vocab_size, cfg.dim)
    self.layers = nn.ModuleDict(
      {str(i): DsLayer(cfg, i) for i in range(shard.start_layer, shard.end_layer + 1)}
    )
    if shard.is_last_layer:
      self.norm = nn.Parameter(torch.ones(cfg.dim))
      self.lm_head = nn.Linear(cfg.dim, cfg.vocab_size, bias=False)
    # per-pair rope tables at the ROPE head dim
    cos, sin = rope_cos_sin(cfg.qk_rope_head_dim, cfg.max_seq_len, cfg.rope_theta, cfg.rope_scaling)
    self.register_buffer("rope_cos", cos, persistent=False)
    self.register_buffer("rope_sin", sin, persistent=False)

  def reset_rope(self):
    cos, sin = rope_cos_sin(self.cfg.qk_rope_head_dim, self.cfg.max_seq_len, self.cfg.rope_theta,
                            self.cfg.rope_scaling, device=self.rope_cos.device)
    self.rope_cos, self.rope_sin = cos, sin

  @property
  def local_layer_ids(self) -> List[int]:
    return list(range(self.shard.start_layer, self.shard.end_layer + 1))

  def pack_decode_weights(self, reserve_bytes: int = 0) -> int:
    """MLA projections + dense/shared MLPs get the packed skinny-GEMM
    prepack (auto-picked per shape vs hipBLASLt); MoE experts get the
    stacked grouped prepack; routed per-expert linears are NOT individually
    packed (the grouped copy already covers them). MLA decode attention
    runs the absorbed-latent MFMA kernel via the packed cache."""
    packed = 0
    for name, mod in self.named_modules():
      if isinstance(mod, XotLinear) and mod.packable() and ".experts." not in name:
        mod.pack_decode()
        if mod.weight_packed is not None:
          packed += mod.weight_packed.numel() * 2
    for mod in self.modules():
      if isinstance(mod, DsMoE):
        mod.pack_grouped()
        if mod.wp_gate_up is not None:
          packed += mod.wp_gate_up.numel() * 2 + mod.wp_down.numel() * 2
    return packed

  def head_weight(self):
    return self.lm_head.weight

  def forward(self, x, caches, positions, start_pos: int, is_decode: bool = False,
              seq_lens=None, last_only: bool = True):
    cfg = self.cfg
    if caches is None:
      raise NotImplementedError("MLA training forward is not implemented (inference only)")
    h = self.embed_tokens(x) if x.dtype in (torch.int32, torch.int64) else x
    if positions.dim() == 0:
      positions = positions.reshape(1)
    for idx, lid in enumerate(self.local_layer_ids):
      h = self.layers[str(lid)](h, self.rope_cos, self.rope_sin, positions, caches[idx],
                                start_pos, is_decode, seq_lens)
    if not self.shard.is_last_layer:
      return h
    if last_only and h.shape[1] > 1:
      h = h[:, -1:, :]
    logits = torch.nn.functional.linear(_rms(h, self.norm, cfg.norm_eps),
                                        self.head_weight().to(h.dtype))
    if is_decode or last_only:
      return logits[:, -1, :]
    return logits


def hf_key_map_deepseek(shard: Shard, cfg: ModelConfig):
  """HF DeepseekV3ForCausalLM checkpoint keys → DeepseekV3Model keys."""
  mapping = {}
  if shard.is_first_layer:
    mapping["model.embed_tokens.weight"] = "embed_tokens.weight"
  if shard.is_last_layer:
    mapping["model.norm.weight"] = "norm"
    mapping["lm_head.weight"] = "lm_head.weight"
  for lid in range(shard.start_layer, shard.end_layer + 1):
    hf = f"model.layers.{lid}."
    ours = f"layers.{lid}."
    att = [("q_a_proj.weight", "q_a_proj.weight"), ("q_a_layernorm.weight", "q_a_layernorm"),
           ("q_b_proj.weight", "q_b_proj.weight")] if cfg.q_lora_rank else [("q_proj.weight", "q_proj.weight")]
    att += [("kv_a_proj_with_mqa.weight", "kv_a_proj_with_mqa.weight"),
            ("kv_a_layernorm.weight", "kv_a_layernorm"),
            ("kv_b_proj.weight", "kv_b_proj.weight"), ("o_proj.weight", "o_proj.weight")]
    for hk, ok in att:
      mapping[hf + "self_attn." + hk] = ours + "self_attn." + ok
    mapping[hf + "input_layernorm.weight"] = ours + "input_layernorm"
    mapping[hf + "post_attention_layernorm.weight"] = ours + "post_attention_layernorm"
    dense = lid < cfg.first_k_dense_replace or cfg.n_experts == 0
    if dense:
      for p in ("gate_proj", "up_proj", "down_proj"):
        mapping[hf + f"mlp.{p}.weight"] = ours + f"mlp.{p}.weight"
    else:
      mapping[hf + "mlp.gate.weight"] = ours + "mlp.gate_weight"
      mapping[hf + "mlp.gate.e_score_correction_bias"] = ours + "mlp.e_score_correction_bias"
      for e in range(cfg.n_experts):
        for p in ("gate_proj", "up_proj", "down_proj"):
          mapping[hf + f"mlp.experts.{e}.{p}.weight"] = ours + f"mlp.experts.{e}.{p}.weight"
      for p in ("gate_proj", "up_proj", "down_proj"):
        mapping[hf + f"mlp.shared_experts.{p}.weight"] = ours + f"mlp.shared_experts.{p}.weight"
  return mapping
