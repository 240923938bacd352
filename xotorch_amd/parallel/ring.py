"""Ring pipeline-parallel decode: one process per GPU, RCCL p2p over xGMI.

This is the MI355X-native replacement for the reference's per-hop gRPC
serialization: contiguous layer shards per rank (the ring
memory-weighted partitioning degenerates to equal layer counts on equal-HBM
GPUs), bf16 hidden states moved rank→rank with torch.distributed send/recv
(one xGMI link per hop, full bandwidth), sampled tokens looped back to stage
0 with async isend so the ring never deadlocks, and position/KV state kept
stage-local (never transmitted).

Pipelining: M = world_size micro-batches are kept in flight so every stage is
busy in steady state; one "step" advances every micro-batch by one token.
On GPU the per-micro-batch decode step (forward + sample + position advance)
is captured in a hipGraph and replayed — the launch-bound small-batch decode
path runs as one graph launch per (stage, micro-batch).
"""
from __future__ import annotations

import os
import time
from dataclasses import dataclass
from typing import List, Optional

import torch
import torch.distributed as dist

from xotorch_amd.engine.kvcache import ShardKVCache
from xotorch_amd.models import model_class_for
from xotorch_amd.models.config import ModelConfig, config_from_hf
from xotorch_amd.models.registry import builtin_config
from xotorch_amd.models.weights import fast_random_init_gpu, random_init
from xotorch_amd.parallel.comm import RingComm
from xotorch_amd.parallel.partitioning import Partition, map_partitions_to_shards
from xotorch_amd.shard import Shard


def equal_ring_shards(model_id: str, n_layers: int, world: int) -> List[Shard]:
  """Equal-memory ring partitions (8x identical MI355X) → balanced layer
  shards. Integer split (sizes differ by at most 1): the float-boundary
  partition map can hand one stage 2 extra layers on worlds that are
  binary-inexact (e.g. 99 layers / 11 ranks → a 10-layer and an 8-layer
  stage), and pipeline throughput is set by the slowest stage."""
  base, rem = divmod(n_layers, world)
  shards, start = [], 0
  for i in range(world):
    cnt = base + (1 if i < rem else 0)
    shards.append(Shard(model_id, start, start + cnt - 1, n_layers))
    start += cnt
  return shards


@dataclass
class RingStats:
  ttft_ms: List[float]  # per micro-batch TTFT (valid on the last stage / rank 0 after sync)


class RingPipeline:
  def __init__(
    self,
    model_id: str,
    rank: int,
    world: int,
    device: str = "cuda",
    dtype: torch.dtype = torch.bfloat16,
    mb_batch: int = 32,
    n_microbatches: Optional[int] = None,
    prompt_len: int = 512,
    max_gen: int = 128,
    use_graphs: bool = True,
    seed: int = 1234,
    cfg_override: Optional[dict] = None,
  ):
    self.rank, self.world = rank, world
    self.device, self.dtype = device, dtype
    self.mb_batch = mb_batch
    self.M = n_microbatches if n_microbatches is not None else max(1, world)
    self.prompt_len = prompt_len
    self.total_len = prompt_len + max_gen
    self.use_graphs = use_graphs and device == "cuda"
    raw = cfg_override or builtin_config(model_id)
    if raw is None:
      raise ValueError(f"no builtin config for {model_id}")
    self.cfg: ModelConfig = config_from_hf(raw, model_id)
    if self.total_len > self.cfg.max_seq_len:
      raise ValueError("prompt+gen exceeds model max_seq_len")
    shards = equal_ring_shards(model_id, self.cfg.n_layers, world)
    if len(shards) < world:
      raise ValueError(
        f"ring world {world} exceeds the model's shardable layers "
        f"({self.cfg.n_layers}): {len(shards)} non-empty shards"
      )
    self.shard: Shard = shards[rank]
    self.is_first = self.shard.is_first_layer
    self.is_last = self.shard.is_last_layer
    self.next_rank = (rank + 1) % world
    self.prev_rank = (rank - 1) % world
    self.comm = RingComm(device)

    # --- model ---
    # construct at the target dtype so to_empty materializes bf16 directly
    # (fp32-then-cast would peak at 2x the weight bytes and leave the
    # allocator's reserved pool at that level)
    model_cls = model_class_for(self.cfg)
    prev_dtype = torch.get_default_dtype()
    torch.set_default_dtype(dtype)
    try:
      with torch.device("meta"):
        model = model_cls(self.cfg, self.shard)
    finally:
      torch.set_default_dtype(prev_dtype)
    model = model.to_empty(device=device)
    model = model.to(dtype)
    if device == "cuda" and self.cfg.dim >= 2048:
      fast_random_init_gpu(model, seed)
    else:
      random_init(model, seed)
    model.reset_rope()  # to_empty left the tables uninitialized
    model.eval()
    self.model = model
    if device == "cuda":
      # decode-GEMM weight prepack (memory permitting; KV caches below still
      # need room — reserve their size + headroom before packing greedily)
      kv_bytes = (
        self.M * 2 * shards[rank].get_layer_count() * mb_batch * self.cfg.n_kv_heads
        * self.total_len * self.cfg.head_dim * (2 if dtype == torch.bfloat16 else 4)
      )
      if self.cfg.head_dim == 128 and dtype == torch.bfloat16 \
         and os.getenv("XOT_MFMA_ATTN", "1") == "1":
        kv_bytes *= 2  # the MFMA-packed cache copies double KV residency
      torch.cuda.empty_cache()  # release init-time cached blocks so the
      # pack policy's mem_get_info reflects actually-usable HBM
      model.pack_decode_weights(reserve_bytes=kv_bytes + (24 << 30))

    # --- per-micro-batch state ---
    B = mb_batch
    kv_heads, k_dim, v_dim = self.cfg.kv_cache_dims()
    self.caches = [
      ShardKVCache(self.shard.get_layer_count(), B, kv_heads, self.total_len,
                   k_dim, dtype, device, v_dim=v_dim)
      for _ in range(self.M)
    ]
    self.positions = [torch.zeros(1, dtype=torch.int32, device=device) for _ in range(self.M)]
    self.seq_lens = [torch.zeros(B, dtype=torch.int32, device=device) for _ in range(self.M)]
    # static I/O buffers (graph + recv targets)
    self.tok_buf = [torch.zeros(B, 1, dtype=torch.int64, device=device) for _ in range(self.M)]
    D = self.cfg.dim
    if not self.is_first:
      self.hid_buf = [torch.zeros(B, 1, D, dtype=dtype, device=device) for _ in range(self.M)]
    self.hid_out: List[Optional[torch.Tensor]] = [None] * self.M
    # async handles
    self._tok_recv_req = [None] * self.M
    self._tok_send_req = [None] * self.M
    self.generated: List[List[torch.Tensor]] = [[] for _ in range(self.M)]
    self.capture_tokens = False
    self._graphs: List[Optional[torch.cuda.CUDAGraph]] = [None] * self.M

  # ---------- single-micro-batch decode step (graph-capturable) ----------

  def _step_mb(self, mb: int):
    if self.is_first:
      x = self.tok_buf[mb]
    else:
      x = self.hid_buf[mb]
    out = self.model(
      x, caches=self.caches[mb].caches, positions=self.positions[mb],
      start_pos=-1, is_decode=True, seq_lens=self.seq_lens[mb],
    )
    if self.is_last:
      tok = out.argmax(dim=-1, keepdim=True)  # greedy (reference default temp=0)
      if self.world == 1:
        self.tok_buf[mb].copy_(tok)  # wrap-around stays local
      else:
        self._tok_out[mb].copy_(tok)
      self.hid_out[mb] = None
    else:
      self.hid_out[mb] = out
    self.positions[mb].add_(1)
    self.seq_lens[mb].add_(1)
    return self.hid_out[mb]

  def _build_graphs(self):
    """Capture the per-micro-batch decode step in hipGraphs (before prefill:
    warmup/capture writes land in cache slots that decode overwrites before
    reading, and positions are reset after prefill)."""
    if not self.use_graphs:
      return
    for mb in range(self.M):
      self.positions[mb].fill_(self.prompt_len)
      self.seq_lens[mb].fill_(self.prompt_len + 1)
      self.tok_buf[mb].random_(0, self.cfg.vocab_size)
    s = torch.cuda.Stream()
    s.wait_stream(torch.cuda.current_stream())
    with torch.cuda.stream(s):
      with torch.inference_mode():
        for mb in range(self.M):
          for _ in range(2):
            self._step_mb(mb)
    torch.cuda.current_stream().wait_stream(s)
    for mb in range(self.M):
      g = torch.cuda.CUDAGraph()
      with torch.inference_mode():
        with torch.cuda.graph(g):
          self._step_mb(mb)
      self._graphs[mb] = g

  # ---------- prefill ----------

  def _prefill_forward(self, mb: int, x: torch.Tensor, pos: torch.Tensor,
                       start_pos: int = 0) -> torch.Tensor:
    """Prefill one micro-batch through this shard in batch chunks: peak
    activation memory (the fused gate_up output is B*S*2I bf16 — 15 GB at
    B=256, S=512 on 70B) stays bounded at the chunk size regardless of B."""
    B = x.shape[0]
    bc = min(B, int(os.getenv("XOT_PREFILL_CHUNK", "128")))
    if bc >= B:
      return self.model(x, caches=self.caches[mb].caches, positions=pos, start_pos=start_pos)
    outs = []
    for c0 in range(0, B, bc):
      c1 = min(c0 + bc, B)
      sliced = []
      for layer in self.caches[mb].caches:
        if len(layer) > 2 and layer[2] is not None:
          sliced.append((layer[0][c0:c1], layer[1][c0:c1], layer[2][c0:c1], layer[3][c0:c1]))
        else:
          sliced.append((layer[0][c0:c1], layer[1][c0:c1]))
      outs.append(self.model(x[c0:c1], caches=sliced, positions=pos, start_pos=start_pos))
    return torch.cat(outs, dim=0)

  def prefill(self, prompts: Optional[List[torch.Tensor]] = None) -> RingStats:
    """Prefill all micro-batches through the ring; returns per-mb TTFT (ms)
    measured on the last stage from the common post-barrier start."""
    B, S = self.mb_batch, self.prompt_len
    if self.world > 1 and not hasattr(self, "_tok_out"):
      self._tok_out = [torch.zeros(B, 1, dtype=torch.int64, device=self.device) for _ in range(self.M)]
    if self.use_graphs:
      self._build_graphs()
    # reset state after graph warmup
    for mb in range(self.M):
      self.positions[mb].zero_()
      self.seq_lens[mb].zero_()
      self.generated[mb] = []
    ttfts = [0.0] * self.M
    dev = torch.device(self.device)
    if self.device == "cuda":
      torch.cuda.synchronize()
    if dist.is_initialized():
      dist.barrier()
    t0 = time.perf_counter()
    with torch.inference_mode():
      for mb in range(self.M):
        pos = torch.arange(0, S, dtype=torch.int32, device=self.device)
        if self.is_first:
          if prompts is not None:
            tokens = prompts[mb].to(self.device)
          else:
            g = torch.Generator(device="cpu").manual_seed(1000 + mb)
            tokens = torch.randint(0, self.cfg.vocab_size, (B, S), generator=g).to(self.device)
          h = self._prefill_forward(mb, tokens, pos)
        else:
          hbuf = torch.empty(B, S, self.cfg.dim, dtype=self.dtype, device=self.device)
          self.comm.recv(hbuf, self.prev_rank)
          h = self._prefill_forward(mb, hbuf, pos)
        if self.is_last:
          tok = h.argmax(dim=-1, keepdim=True)  # h is [B, V] logits
          if self.device == "cuda":
            torch.cuda.synchronize()
          ttfts[mb] = (time.perf_counter() - t0) * 1000.0
          if self.world == 1:
            self.tok_buf[mb].copy_(tok)
          else:
            self._tok_out[mb].copy_(tok)
            self._tok_send_req[mb] = self.comm.isend(self._tok_out[mb], self.next_rank)
          if self.capture_tokens:
            self.generated[mb].append(tok.clone())
        else:
          self.comm.send(h, self.next_rank)
        if self.is_first and self.world > 1:
          self._tok_recv_req[mb] = self.comm.irecv(self.tok_buf[mb], self.prev_rank)
        # advance to first decode position
        self.positions[mb].fill_(S)
        self.seq_lens[mb].fill_(S + 1)
    return RingStats(ttft_ms=ttfts)

  # ---------- decode ----------

  def decode_step(self):
    """Advance every micro-batch by one token (one whole-ring step)."""
    with torch.inference_mode():
      for mb in range(self.M):
        if self.world == 1:
          self._run_mb(mb)
          if self.capture_tokens:
            self.generated[mb].append(self.tok_buf[mb].clone())
          continue
        if self.is_first:
          self._tok_recv_req[mb].wait()
        else:
          self.comm.recv(self.hid_buf[mb], self.prev_rank)
        if self.is_last and self._tok_send_req[mb] is not None:
          self._tok_send_req[mb].wait()
        self._run_mb(mb)
        if self.is_last:
          self._tok_send_req[mb] = self.comm.isend(self._tok_out[mb], self.next_rank)
          if self.capture_tokens:
            self.generated[mb].append(self._tok_out[mb].clone())
        else:
          self.comm.send(self.hid_out[mb], self.next_rank)
        if self.is_first:
          self._tok_recv_req[mb] = self.comm.irecv(self.tok_buf[mb], self.prev_rank)

  def _run_mb(self, mb: int):
    if self._graphs[mb] is not None:
      self._graphs[mb].replay()
    else:
      self._step_mb(mb)

  def finish(self):
    """Drain outstanding async token traffic (call before teardown)."""
    for req in self._tok_recv_req + self._tok_send_req:
      if req is not None:
        try:
          req.wait()
        except Exception:
          pass
