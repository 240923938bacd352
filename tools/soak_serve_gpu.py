"""GPU serve soak: many requests through slot churn with graphs on."""
import queue, threading, time, sys
sys.path.insert(0, "/root/repo")
import numpy as np, torch
from xotorch_amd.serve_ring import AdmitMsg, RingSlotWorker

w = RingSlotWorker("llama-3.2-1b", 0, 1, device="cuda", dtype=torch.bfloat16,
                   slots=4, max_seq=512, use_graphs=True)
w._build_graph()
rng = np.random.default_rng(1)
q = queue.Queue()
got, remaining = {}, set()
done = threading.Event()
def emit(rid, tok, fin, meta):
  got.setdefault(rid, []).append(tok)
  if fin:
    remaining.discard(rid)
    if not remaining:
      done.set()
import os
N = int(os.getenv("SOAK_N", "24"))
for i in range(N):
  rid = f"r{i}"
  remaining.add(rid)
  ids = [int(v) for v in rng.integers(0, 32000, int(rng.integers(4, 40)))]
  q.put(AdmitMsg(rid, torch.tensor([ids], dtype=torch.int64), int(rng.integers(3, 20)), 0.0))
mem0 = torch.cuda.memory_allocated()
t = threading.Thread(target=w.serve_forever, args=(q, emit), daemon=True)
t0 = time.perf_counter()
t.start()
ok = done.wait(600)
q.put(AdmitMsg("stop", None, 0, 0.0))
t.join(timeout=30)
assert ok, f"soak failed, remaining: {remaining}"
mem1 = torch.cuda.memory_allocated()
growth = (mem1 - mem0) / 2**20
assert growth < 256, f"memory grew {growth:.0f} MiB over the soak (slot/KV leak?)"
print(f"serve soak: {N} requests, {sum(len(v) for v in got.values())} tokens in "
      f"{time.perf_counter()-t0:.1f}s, graphs={w._graph is not None}, "
      f"mem growth {growth:.1f} MiB, ok")
