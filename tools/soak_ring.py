import os, sys, json, tempfile
sys.path.insert(0, '/root/repo')
import torch
import torch.multiprocessing as mp
from pathlib import Path

TINY = {
  "model_type": "llama", "hidden_size": 64, "num_hidden_layers": 6,
  "num_attention_heads": 4, "num_key_value_heads": 2, "intermediate_size": 128,
  "vocab_size": 211, "rope_theta": 10000.0, "rms_norm_eps": 1e-5,
  "max_position_embeddings": 128, "torch_dtype": "float32",
}
STEPS = 40

def run_ring(world, out_dir):
    from xotorch_amd.parallel.ring import RingPipeline
    rank = int(os.environ.get("RANK", "0"))
    ring = RingPipeline(model_id="soak", rank=rank, world=world, device="cpu",
                        dtype=torch.float32, mb_batch=2, n_microbatches=max(2, world),
                        prompt_len=12, max_gen=STEPS + 2, use_graphs=False, cfg_override=TINY)
    ring.capture_tokens = True
    ring.prefill()
    for _ in range(STEPS):
        ring.decode_step()
    ring.finish()
    if ring.is_last:
        toks = [[t.tolist() for t in mb] for mb in ring.generated]
        Path(out_dir, f"tokens_w{world}.json").write_text(json.dumps(toks))

def worker(rank, world, out_dir, port):
    os.environ.update(RANK=str(rank), WORLD_SIZE=str(world),
                      MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port))
    import torch.distributed as dist
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        run_ring(world, out_dir)
    finally:
        dist.destroy_process_group()

if __name__ == "__main__":
    from xotorch_amd.helpers import find_available_port
    with tempfile.TemporaryDirectory() as d:
        os.environ.pop("RANK", None); os.environ.pop("WORLD_SIZE", None)
        run_ring(1, d)
        ref = json.loads(Path(d, "tokens_w1.json").read_text())
        for world in (2, 3, 6):
            port = find_available_port("127.0.0.1")
            mp.spawn(worker, args=(world, d, port), nprocs=world, join=True)
            got = json.loads(Path(d, f"tokens_w{world}.json").read_text())
            for mb in range(2):
                assert ref[mb] == got[mb], f"world={world} mb={mb} diverged"
            print(f"soak ok world={world}: {STEPS} steps, mb tokens identical")
