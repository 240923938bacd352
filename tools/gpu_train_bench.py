"""LoRA fine-tune step benchmark (BASELINE config: Llama-3 8B LoRA, DP via
torchrun on multi-GPU; single-GPU here measures the per-GPU step)."""
import sys, time
from pathlib import Path
sys.path.insert(0, str(Path(__file__).resolve().parent.parent))
import torch

from xotorch_amd.models.config import config_from_hf
from xotorch_amd.models.registry import builtin_config
from xotorch_amd.models.llama import ShardedModel
from xotorch_amd.models.weights import fast_random_init_gpu
from xotorch_amd.shard import Shard
from xotorch_amd.train.lora import apply_lora, lora_parameters
from xotorch_amd.train.trainer import DPTrainer

from xotorch_amd.parallel.comm import init_distributed
rank, world = init_distributed()
model_id = sys.argv[1] if len(sys.argv) > 1 else "llama-3-8b"
B, S = 8, 512
cfg = config_from_hf(builtin_config(model_id), model_id)
shard = Shard(model_id, 0, cfg.n_layers - 1, cfg.n_layers)
prev = torch.get_default_dtype()
torch.set_default_dtype(torch.bfloat16)
with torch.device("meta"):
  model = ShardedModel(cfg, shard)
torch.set_default_dtype(prev)
model = model.to_empty(device="cuda").to(torch.bfloat16)
fast_random_init_gpu(model)
model.reset_rope()
n_lora = apply_lora(model, rank=16)
model.train()
trainer = DPTrainer(model, lr=1e-4)
trainer.sync_initial_state()
g = torch.Generator(device="cpu").manual_seed(100 + rank)  # per-rank data shard
tokens = torch.randint(0, cfg.vocab_size, (B, S + 1), generator=g).to("cuda")
inputs, targets = tokens[:, :-1], tokens[:, 1:]
lengths = torch.full((B,), S, device="cuda")

def fwd(x):
  # unified cache-free training forward (activation checkpointing on)
  pos = torch.arange(S, dtype=torch.int32, device="cuda")
  return model(x, caches=None, positions=pos, start_pos=0, last_only=False)

for _ in range(2):
  loss = trainer.step(fwd, inputs, targets, lengths)
torch.cuda.synchronize()
t0 = time.perf_counter()
N = 6
for _ in range(N):
  loss = trainer.step(fwd, inputs, targets, lengths)
torch.cuda.synchronize()
dt = (time.perf_counter() - t0) / N
import hashlib
import torch.distributed as dist
sig = hashlib.sha256()
for p_ in trainer.params:
  sig.update(p_.detach().float().cpu().numpy().tobytes())
print(f"[rank {rank}/{world}] lora train {model_id}: {len(n_lora)} adapters, loss {float(loss):.3f}, "
      f"{dt*1e3:.1f} ms/step, {B*S/dt:.0f} tokens/s/gpu, params sha256 {sig.hexdigest()[:16]}")
if dist.is_initialized():
  dist.destroy_process_group()
