"""Debug: verify decode-GEMM prepack engages in the ring pipeline (8B, fast)."""
import sys, time
from pathlib import Path
sys.path.insert(0, str(Path(__file__).resolve().parent.parent))
import torch
from xotorch_amd.parallel.ring import RingPipeline

ring = RingPipeline(model_id="llama-3-8b", rank=0, world=1, device="cuda",
                    dtype=torch.bfloat16, mb_batch=64, prompt_len=512, max_gen=40,
                    use_graphs=True)
m = ring.model
packed = sum(1 for _, mod in m.named_modules() if getattr(mod, "weight_packed", None) is not None)
total = sum(1 for _, mod in m.named_modules() if hasattr(mod, "weight_packed"))
print(f"packed {packed}/{total} XotLinear modules")
l0 = m.layers[str(m.local_layer_ids[0])]
print("down packed:", l0.mlp.down_proj.weight_packed is not None,
      "qkv packed:", l0.self_attn.qkv_proj.weight_packed is not None,
      "lm_head packed:", getattr(m, "lm_head", None) is not None and m.lm_head.weight_packed is not None)
ring.prefill()
for _ in range(4):
    ring.decode_step()
torch.cuda.synchronize()
t0 = time.perf_counter()
for _ in range(16):
    ring.decode_step()
torch.cuda.synchronize()
dt = (time.perf_counter() - t0) / 16
print(f"step {dt*1e3:.2f} ms  tok/s {64/dt:.0f}")
