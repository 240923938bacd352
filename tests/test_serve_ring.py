"""Slot-based continuous-batching ring server tests (CPU, world 1 and 2).

The serving path admits requests into free KV slots and advances every
active slot per decode tick — these tests pin: greedy tokens match the
single-request TorchEngine oracle, two concurrent requests interleave
without corrupting each other, slots are recycled, and the world=2 gloo
ring produces identical tokens to world=1.
"""
import asyncio
import os
import queue
import threading

import numpy as np
import pytest
import torch

from xotorch_amd.serve_ring import AdmitMsg, RingSlotWorker

TINY_ID = "dummy"  # builtin tiny llama config


def oracle_tokens(prompt_ids, max_new):
  """Greedy reference via the TorchEngine single-request path."""
  from xotorch_amd.engine.torch_engine import TorchEngine
  from xotorch_amd.models.registry import build_full_shard

  async def go():
    eng = TorchEngine(device="cpu", dtype=torch.float32)
    shard = build_full_shard(TINY_ID, "TorchEngine")
    toks = np.asarray(prompt_ids, dtype=np.int64).reshape(1, -1)
    out, state = await eng.infer_tensor("o", shard, toks, {"total_len": 256})
    res = []
    for _ in range(max_new):
      tok = int(out.argmax(-1).reshape(-1)[0])
      res.append(tok)
      out, state = await eng.infer_tensor("o", shard, np.asarray([[tok]]), state)
    return res
  return asyncio.new_event_loop().run_until_complete(go())


def serve_requests(worker, reqs, expect_tokens):
  """Drive serve_forever with a set of requests; returns {rid: [tokens]}."""
  q = queue.Queue()
  got = {}
  ttft = {}
  done = threading.Event()
  remaining = {r[0] for r in reqs}

  def emit(rid, tok, fin, meta):
    got.setdefault(rid, []).append(tok)
    if "ttft_s" in meta:
      ttft[rid] = meta["ttft_s"]
    if fin:
      remaining.discard(rid)
      if not remaining:
        done.set()

  for rid, ids, max_new in reqs:
    q.put(AdmitMsg(rid, torch.tensor([ids], dtype=torch.int64), max_new, 0.0))

  t = threading.Thread(target=worker.serve_forever, args=(q, emit), daemon=True)
  t.start()
  assert done.wait(120), f"requests did not finish: {remaining}"
  q.put(AdmitMsg("stop", None, 0, 0.0))  # shutdown sentinel
  t.join(timeout=30)
  return got, ttft


def test_slots_single_request_matches_oracle():
  torch.manual_seed(0)
  w = RingSlotWorker(TINY_ID, 0, 1, device="cpu", dtype=torch.float32,
                     slots=4, max_seq=128, use_graphs=False)
  ids = list(np.random.default_rng(3).integers(0, 200, 7))
  got, ttft = serve_requests(w, [("r1", ids, 5)], None)
  assert got["r1"] == oracle_tokens(ids, 5)
  assert ttft["r1"] > 0


def test_slots_concurrent_requests_interleave():
  """Two requests in flight share decode ticks; both match their oracles
  (per-slot positions keep their KV independent)."""
  w = RingSlotWorker(TINY_ID, 0, 1, device="cpu", dtype=torch.float32,
                     slots=4, max_seq=128, use_graphs=False)
  rng = np.random.default_rng(7)
  ids_a = list(rng.integers(0, 200, 9))
  ids_b = list(rng.integers(0, 200, 5))
  got, _ = serve_requests(w, [("a", ids_a, 6), ("b", ids_b, 6)], None)
  assert got["a"] == oracle_tokens(ids_a, 6)
  assert got["b"] == oracle_tokens(ids_b, 6)


def test_slots_recycled_across_requests():
  """More requests than slots: slots are released and reused."""
  w = RingSlotWorker(TINY_ID, 0, 1, device="cpu", dtype=torch.float32,
                     slots=2, max_seq=128, use_graphs=False)
  rng = np.random.default_rng(11)
  reqs = [(f"r{i}", list(rng.integers(0, 200, 4 + i)), 3) for i in range(5)]
  got, _ = serve_requests(w, reqs, None)
  for rid, ids, max_new in reqs:
    assert got[rid] == oracle_tokens(ids, max_new), rid


# ---------------- world=2 gloo ring ----------------

def _ring_worker(rank, world, port, out_dir, reqs):
  import json
  import torch.distributed as dist
  os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                    RANK=str(rank), WORLD_SIZE=str(world))
  dist.init_process_group("gloo", rank=rank, world_size=world)
  w = RingSlotWorker(TINY_ID, rank, world, device="cpu", dtype=torch.float32,
                     slots=2, max_seq=128, use_graphs=False)
  if rank == 0:
    got, _ = serve_requests(w, reqs, None)
    with open(os.path.join(out_dir, "tokens.json"), "w") as f:
      json.dump(got, f)
  else:
    w.serve_forever()
  dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_two_rank_ring_matches_single(tmp_path):
  import json
  import torch.multiprocessing as mp
  from xotorch_amd.helpers import find_available_port
  rng = np.random.default_rng(13)
  reqs = [("a", [int(v) for v in rng.integers(0, 200, 6)], 4),
          ("b", [int(v) for v in rng.integers(0, 200, 8)], 4)]
  port = find_available_port("127.0.0.1")
  mp.spawn(_ring_worker, args=(2, port, str(tmp_path), reqs), nprocs=2, join=True)
  got = json.loads((tmp_path / "tokens.json").read_text())
  for rid, ids, max_new in reqs:
    assert got[rid] == oracle_tokens(ids, max_new), rid


def test_cancel_releases_slot():
  """Cancelling a long request frees its slot for the next one."""
  import time as _t
  w = RingSlotWorker(TINY_ID, 0, 1, device="cpu", dtype=torch.float32,
                     slots=1, max_seq=2048, use_graphs=False)
  q = queue.Queue()
  got = {}
  done = threading.Event()

  def emit(rid, tok, fin, meta):
    got.setdefault(rid, []).append((tok, fin, meta))
    if fin and rid == "short":
      done.set()

  ids = list(np.random.default_rng(2).integers(0, 200, 6))
  q.put(AdmitMsg("long", torch.tensor([ids], dtype=torch.int64), 2000, 0.0))
  t = threading.Thread(target=w.serve_forever, args=(q, emit), daemon=True)
  t.start()
  _t.sleep(0.15)         # long request occupies the only slot
  w.cancel("long")
  q.put(AdmitMsg("short", torch.tensor([ids], dtype=torch.int64), 3, 0.0))
  assert done.wait(60), "slot was not freed by cancellation"
  q.put(AdmitMsg("stop", None, 0, 0.0))
  t.join(timeout=30)
  assert len(got["long"]) < 240  # cancelled well before the max_seq-clamped max_new
  assert got["long"][-1][2].get("cancelled") or got["long"][-1][1]


@pytest.mark.timeout(300)
def test_three_rank_ring_matches_single(tmp_path):
  """Odd-world serving ring (middle stage is neither first nor last)."""
  import json
  import torch.multiprocessing as mp
  from xotorch_amd.helpers import find_available_port
  rng = np.random.default_rng(17)
  reqs = [("a", [int(v) for v in rng.integers(0, 200, 5)], 4),
          ("b", [int(v) for v in rng.integers(0, 200, 9)], 4)]
  port = find_available_port("127.0.0.1")
  mp.spawn(_ring_worker, args=(3, port, str(tmp_path), reqs), nprocs=3, join=True)
  got = json.loads((tmp_path / "tokens.json").read_text())
  for rid, ids, max_new in reqs:
    assert got[rid] == oracle_tokens(ids, max_new), rid

def test_prefill_interleaves_with_decode(monkeypatch):
  """A long prompt prefills in chunks WHILE an active slot keeps emitting:
  tokens from the decoding request must appear between the second request's
  admission and its first token, and both must still match their oracles."""
  monkeypatch.setattr("xotorch_amd.serve_ring.PREFILL_CHUNK", 2)
  w = RingSlotWorker(TINY_ID, 0, 1, device="cpu", dtype=torch.float32,
                     slots=4, max_seq=128, use_graphs=False)
  rng = np.random.default_rng(23)
  ids_a = list(rng.integers(0, 200, 2))    # one-chunk prompt: activates first
  ids_b = list(rng.integers(0, 200, 10))   # five chunks: prefills while a decodes
  order = []
  q = queue.Queue()
  done = threading.Event()
  remaining = {"a", "b"}

  def emit(rid, tok, fin, meta):
    order.append((rid, tok))
    if fin:
      remaining.discard(rid)
      if not remaining:
        done.set()

  q.put(AdmitMsg("a", torch.tensor([ids_a], dtype=torch.int64), 12, 0.0))
  q.put(AdmitMsg("b", torch.tensor([ids_b], dtype=torch.int64), 4, 0.0))
  t = threading.Thread(target=w.serve_forever, args=(q, emit), daemon=True)
  t.start()
  assert done.wait(120)
  q.put(AdmitMsg("stop", None, 0, 0.0))
  t.join(timeout=30)
  got_a = [tok for rid, tok in order if rid == "a"]
  got_b = [tok for rid, tok in order if rid == "b"]
  assert got_a == oracle_tokens(ids_a, 12)
  assert got_b == oracle_tokens(ids_b, 4)
  first_b = next(i for i, (rid, _) in enumerate(order) if rid == "b")
  a_before_b = sum(1 for rid, _ in order[:first_b] if rid == "a")
  assert a_before_b >= 2, f"no interleaving: order head {order[:8]}"


def test_cancel_during_prefill(monkeypatch):
  """Cancelling a request whose prompt is still prefilling drops the
  in-progress chunked prefill and frees the slot."""
  import time as _t
  monkeypatch.setattr("xotorch_amd.serve_ring.PREFILL_CHUNK", 1)
  w = RingSlotWorker(TINY_ID, 0, 1, device="cpu", dtype=torch.float32,
                     slots=1, max_seq=128, use_graphs=False)
  q = queue.Queue()
  got = {}
  done = threading.Event()

  def emit(rid, tok, fin, meta):
    got.setdefault(rid, []).append((tok, fin, meta))
    if fin and rid == "short":
      done.set()

  rng = np.random.default_rng(31)
  long_ids = list(rng.integers(0, 200, 64))   # 64 one-token chunks
  short_ids = list(rng.integers(0, 200, 4))
  q.put(AdmitMsg("long", torch.tensor([long_ids], dtype=torch.int64), 8, 0.0))
  t = threading.Thread(target=w.serve_forever, args=(q, emit), daemon=True)
  t.start()
  _t.sleep(0.1)          # mid-prefill (64 chunks take a while on CPU)
  w.cancel("long")
  q.put(AdmitMsg("short", torch.tensor([short_ids], dtype=torch.int64), 3, 0.0))
  assert done.wait(60), "slot not freed by mid-prefill cancellation"
  q.put(AdmitMsg("stop", None, 0, 0.0))
  t.join(timeout=30)
  assert [tok for tok, _, _ in got["short"]] == oracle_tokens(short_ids, 3)
  assert got["long"][-1][2].get("cancelled")


@pytest.mark.timeout(300)
def test_two_rank_chunked_prefill_interleave(tmp_path, monkeypatch):
  """World-2 ring with multi-chunk prompts (chunk=2): OP_PREFILL hops each
  chunk through both stages while an active slot keeps decoding; tokens
  still match the single-request oracle."""
  import json
  import torch.multiprocessing as mp
  from xotorch_amd.helpers import find_available_port
  monkeypatch.setenv("XOT_SERVE_PREFILL_CHUNK", "2")  # inherited by spawned ranks
  rng = np.random.default_rng(41)
  reqs = [("a", [int(v) for v in rng.integers(0, 200, 3)], 8),   # activates first
          ("b", [int(v) for v in rng.integers(0, 200, 11)], 4)]  # 6 chunks
  port = find_available_port("127.0.0.1")
  mp.spawn(_ring_worker, args=(2, port, str(tmp_path), reqs), nprocs=2, join=True)
  got = json.loads((tmp_path / "tokens.json").read_text())
  for rid, ids, max_new in reqs:
    assert got[rid] == oracle_tokens(ids, max_new), rid


@pytest.mark.timeout(300)
def test_serve_chaos_random_admits_and_cancels(monkeypatch):
  """Randomized integration: 12 requests with random prompt lengths, budgets
  and admission times, a few mid-flight cancels, small prefill chunks, 3
  slots. Every request that ran to completion must match its oracle."""
  import time as _t
  monkeypatch.setattr("xotorch_amd.serve_ring.PREFILL_CHUNK", 3)
  rng = np.random.default_rng(97)
  w = RingSlotWorker(TINY_ID, 0, 1, device="cpu", dtype=torch.float32,
                     slots=3, max_seq=128, use_graphs=False)
  reqs = {}
  for i in range(12):
    rid = f"r{i}"
    plen = int(rng.integers(2, 20))
    reqs[rid] = (list(rng.integers(0, 200, plen)), int(rng.integers(1, 10)))
  cancel_ids = {"r3", "r7"}
  q = queue.Queue()
  got, done_flags = {}, {}
  all_done = threading.Event()

  def emit(rid, tok, fin, meta):
    if not meta.get("cancelled"):
      got.setdefault(rid, []).append(tok)
    if fin:
      done_flags[rid] = meta.get("cancelled", False)
      if len(done_flags) == len(reqs):
        all_done.set()

  t = threading.Thread(target=w.serve_forever, args=(q, emit), daemon=True)
  t.start()
  for i, (rid, (ids, max_new)) in enumerate(reqs.items()):
    q.put(AdmitMsg(rid, torch.tensor([ids], dtype=torch.int64), max_new, 0.0))
    if rid in cancel_ids:
      _t.sleep(0.02)
      w.cancel(rid)
    if i % 3 == 2:
      _t.sleep(0.05)  # stagger admissions across decode progress
  assert all_done.wait(180), f"unfinished: {set(reqs) - set(done_flags)}"
  q.put(AdmitMsg("stop", None, 0, 0.0))
  t.join(timeout=30)
  completed = [rid for rid, cancelled in done_flags.items() if not cancelled]
  assert len(completed) >= len(reqs) - len(cancel_ids)
  for rid in completed:
    ids, max_new = reqs[rid]
    assert got[rid] == oracle_tokens(ids, max_new), rid


@pytest.mark.timeout(300)
def test_serve_ring_main_http_end_to_end():
  """The `xot serve` entry point (serve_ring.py __main__): world-1 CPU
  worker + ChatGPT API in a subprocess; a completion request round-trips."""
  import json as _json
  import subprocess
  import sys
  import time as _t
  import urllib.request
  from pathlib import Path
  from xotorch_amd.helpers import find_available_port

  port = find_available_port("127.0.0.1")
  script = Path(__file__).resolve().parent.parent / "xotorch_amd" / "serve_ring.py"
  env = dict(os.environ, XOT_OFFLINE="1")
  proc = subprocess.Popen([sys.executable, str(script), "--model", "dummy",
                           "--port", str(port), "--no-graphs"],
                          stdout=subprocess.PIPE, stderr=subprocess.STDOUT, env=env)
  try:
    deadline = _t.time() + 60
    up = False
    while _t.time() < deadline:
      try:
        urllib.request.urlopen(f"http://127.0.0.1:{port}/healthcheck", timeout=2)
        up = True
        break
      except Exception:
        if proc.poll() is not None:
          out = proc.stdout.read().decode(errors="replace")
          raise AssertionError(f"server died early:\n{out[-2000:]}")
        _t.sleep(0.3)
    assert up, "server never came up"
    body = _json.dumps({"model": "dummy", "max_tokens": 4,
                        "messages": [{"role": "user", "content": "hi"}]}).encode()
    req = urllib.request.Request(f"http://127.0.0.1:{port}/v1/chat/completions",
                                 data=body, headers={"Content-Type": "application/json"})
    with urllib.request.urlopen(req, timeout=60) as r:
      data = _json.loads(r.read())
    assert data["object"] == "chat.completion"
    assert data["choices"][0]["message"]["content"]
  finally:
    proc.terminate()
    try:
      proc.wait(timeout=15)
    except subprocess.TimeoutExpired:
      proc.kill()
