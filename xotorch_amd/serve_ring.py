"""Single-node multi-GPU ring serving worker (launched via torchrun by
`xot serve --gpus N`).

Rank 0 hosts the ChatGPT-compatible HTTP API; every rank holds one layer
shard. Per request: rank 0 tokenizes and broadcasts {S, max_new} + prompt
tokens; the ring prefized+decodes with bf16 hidden send/recv over RCCL/xGMI;
the last stage's sampled token is broadcast to all ranks each step (stage 0
feeds it back in, rank 0 streams it to the client). This is the product
serving path the reference implements with per-hop gRPC+JSON
 — here the activation hop is one xGMI link.
"""
from __future__ import annotations

import argparse
import asyncio
import json
import threading
import time
import uuid
from queue import Queue
from typing import List, Optional

import torch
import torch.distributed as dist

from xotorch_amd.engine.kvcache import ShardKVCache
from xotorch_amd.models.config import config_from_hf
from xotorch_amd.models.llama import ShardedModel
from xotorch_amd.models.registry import builtin_config, get_repo
from xotorch_amd.models.weights import fast_random_init_gpu, load_shard_weights, random_init
from xotorch_amd.parallel.comm import init_distributed
from xotorch_amd.parallel.ring import equal_ring_shards

MAX_SEQ = 4096


class RingWorker:
  def __init__(self, model_id: str, rank: int, world: int, device="cuda",
               dtype=torch.bfloat16, model_dir: Optional[str] = None):
    self.rank, self.world = rank, world
    self.device, self.dtype = device, dtype
    self.cfg = config_from_hf(builtin_config(model_id) or {}, model_id)
    self.shard = equal_ring_shards(model_id, self.cfg.n_layers, world)[rank]
    prev_dtype = torch.get_default_dtype()
    torch.set_default_dtype(dtype)
    try:
      with torch.device("meta"):
        model = ShardedModel(self.cfg, self.shard)
    finally:
      torch.set_default_dtype(prev_dtype)
    model = model.to_empty(device=device).to(dtype)
    if model_dir:
      load_shard_weights(model, model_dir, device="cpu")
      model = model.to(device)
    elif device == "cuda" and self.cfg.dim >= 2048:
      fast_random_init_gpu(model)
    else:
      random_init(model)
    model.reset_rope()
    model.eval()
    if device == "cuda":
      torch.cuda.empty_cache()
      model.pack_decode_weights(reserve_bytes=24 << 30)
    self.model = model
    self.cache = ShardKVCache(self.shard.get_layer_count(), 1, self.cfg.n_kv_heads,
                              MAX_SEQ, self.cfg.head_dim, dtype, device)
    self.next_rank = (rank + 1) % world
    self.prev_rank = (rank - 1) % world

  def serve_request(self, tokens: Optional[torch.Tensor], max_new: int, temp: float,
                    on_token=None) -> List[int]:
    """Run one request through the ring. tokens given on rank 0 only."""
    dev = self.device
    hdr = torch.zeros(2, dtype=torch.int64, device=dev if self.world > 1 else "cpu")
    if self.rank == 0:
      if tokens.shape[1] >= MAX_SEQ:
        # reject BEFORE the ring is engaged (followers never see the request)
        raise ValueError(f"prompt length {tokens.shape[1]} exceeds context {MAX_SEQ}")
      hdr[0] = tokens.shape[1]
      hdr[1] = max_new
    if self.world > 1:
      dist.broadcast(hdr, 0)
    S, max_new = int(hdr[0]), int(hdr[1])
    # clamp so KV writes can never pass cache capacity (the HIP append
    # kernel trusts pos < T)
    max_new = min(max_new, MAX_SEQ - S)
    if self.world > 1:
      tok_bcast = torch.zeros(1, S, dtype=torch.int64, device=dev)
      if self.rank == 0:
        tok_bcast.copy_(tokens.to(dev))
      dist.broadcast(tok_bcast, 0)
      tokens = tok_bcast
    else:
      tokens = tokens.to(dev)
    total = min(MAX_SEQ, S + max_new)
    out_tokens: List[int] = []
    cur_tok = torch.zeros(1, 1, dtype=torch.int64, device=dev)
    with torch.inference_mode():
      for step in range(max_new):
        pos0 = S + step - 1 if step > 0 else 0
        s_cur = 1 if step > 0 else S
        positions = torch.arange(pos0, pos0 + s_cur, dtype=torch.int32, device=dev)
        seq_lens = torch.full((1,), pos0 + s_cur, dtype=torch.int32, device=dev)
        x = tokens if step == 0 else cur_tok
        if self.shard.is_first_layer:
          h = self.model(x, caches=self.cache.caches, positions=positions,
                         start_pos=pos0, is_decode=(step > 0), seq_lens=seq_lens)
        else:
          hbuf = torch.empty(1, s_cur, self.cfg.dim, dtype=self.dtype, device=dev)
          dist.recv(hbuf, self.prev_rank)
          h = self.model(hbuf, caches=self.cache.caches, positions=positions,
                         start_pos=pos0, is_decode=(step > 0), seq_lens=seq_lens)
        if self.shard.is_last_layer:
          from xotorch_amd import ops
          tok = ops.softmax_sample(h, temperature=temp, top_k=35 if temp > 0 else 0).view(1, 1)
          cur_tok.copy_(tok)
        elif self.world > 1:
          dist.send(h.contiguous(), self.next_rank)
        if self.world > 1:
          dist.broadcast(cur_tok, self.world - 1)
        t = int(cur_tok[0, 0])
        out_tokens.append(t)
        if on_token is not None:
          on_token(t)
        eos = self.cfg.eos_token_id
        if eos is not None and t == eos:
          break
    return out_tokens


def main():
  p = argparse.ArgumentParser()
  p.add_argument("--model", type=str, default="llama-3-8b")
  p.add_argument("--port", type=int, default=52415)
  p.add_argument("--model-dir", type=str, default=None)
  args = p.parse_args()
  rank, world = init_distributed()
  device = "cuda" if torch.cuda.is_available() else "cpu"
  dtype = torch.bfloat16 if device == "cuda" else torch.float32
  worker = RingWorker(args.model, rank, world, device, dtype, args.model_dir)

  if rank != 0:
    # follower loop: serve forever (header broadcast wakes us per request)
    while True:
      worker.serve_request(None, 0, 0.0)
    return

  # rank 0: HTTP API + ring driver
  from aiohttp import web
  from xotorch_amd.engine.tokenizers import resolve_tokenizer, DummyTokenizer

  loop = asyncio.new_event_loop()
  asyncio.set_event_loop(loop)
  try:
    tokenizer = loop.run_until_complete(resolve_tokenizer(args.model_dir or get_repo(args.model, "HIPEngine")))
  except Exception:
    tokenizer = DummyTokenizer()
  ring_lock = threading.Lock()

  async def completions(request):
    data = await request.json()
    messages = data.get("messages", [])
    stream = data.get("stream", False)
    temp = float(data.get("temperature", 0.0))
    max_new = int(data.get("max_tokens") or 256)
    try:
      prompt = tokenizer.apply_chat_template(conversation=messages, tokenize=False, add_generation_prompt=True)
    except Exception:
      prompt = "\n".join(m.get("content", "") for m in messages)
    ids = tokenizer.encode(prompt)
    tokens = torch.tensor([ids], dtype=torch.int64)
    rid = str(uuid.uuid4())
    q: Queue = Queue()

    def run_ring():
      with ring_lock:
        worker.serve_request(tokens, max_new, temp, on_token=lambda t: q.put(t))
      q.put(None)

    t = threading.Thread(target=run_ring, daemon=True)
    t.start()
    if stream:
      resp = web.StreamResponse(headers={"Content-Type": "text/event-stream"})
      await resp.prepare(request)
      while True:
        tok = await asyncio.get_running_loop().run_in_executor(None, q.get)
        if tok is None:
          break
        chunk = {"id": rid, "object": "chat.completion.chunk", "model": args.model,
                 "choices": [{"index": 0, "delta": {"content": tokenizer.decode([tok])}, "finish_reason": None}]}
        await resp.write(f"data: {json.dumps(chunk)}\n\n".encode())
      await resp.write(b"data: [DONE]\n\n")
      await resp.write_eof()
      return resp
    toks = []
    while True:
      tok = await asyncio.get_running_loop().run_in_executor(None, q.get)
      if tok is None:
        break
      toks.append(tok)
    return web.json_response({
      "id": rid, "object": "chat.completion", "model": args.model,
      "choices": [{"index": 0, "message": {"role": "assistant", "content": tokenizer.decode(toks)},
                   "finish_reason": "stop"}],
      "usage": {"completion_tokens": len(toks)},
    })

  async def health(request):
    return web.json_response({"status": "ok", "world": world, "model": args.model})

  app = web.Application()
  app.router.add_post("/v1/chat/completions", completions)
  app.router.add_get("/healthcheck", health)

  async def start():
    runner = web.AppRunner(app)
    await runner.setup()
    await web.TCPSite(runner, "0.0.0.0", args.port).start()
    print(f"ring server ({world} GPUs) on :{args.port}")
    await asyncio.Event().wait()

  loop.run_until_complete(start())


if __name__ == "__main__":
  main()
