"""Debug sweep for attn_decode: per-length max diff vs torch reference."""
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))
import torch  # noqa: E402

from xotorch_amd.ops import _hip_ops as hip  # noqa: E402
from xotorch_amd.ops import torch_ref  # noqa: E402


def bt(*shape, seed=0):
  g = torch.Generator(device="cuda").manual_seed(seed)
  return torch.randn(*shape, generator=g, device="cuda", dtype=torch.float32).to(torch.bfloat16)


def main():
  B, H, KVH, hd, T = 4, 16, 4, 128, 640
  q = bt(B, 1, H, hd)
  kc, vc = bt(B, KVH, T, hd, seed=8), bt(B, KVH, T, hd, seed=9)
  for sl in [1, 2, 15, 16, 17, 31, 127, 128, 129, 333, 639, 640]:
    lens = torch.full((B,), sl, dtype=torch.int32, device="cuda")
    out = hip.attn_decode(q, kc, vc, lens)
    ref = torch_ref.attn_decode(q, kc, vc, sl)
    d = (out.float() - ref.float()).abs()
    print(f"sl={sl:4d} maxdiff={d.max().item():.6f} at={d.argmax().item()}")
  lens = torch.tensor([1, 17, 333, 640], dtype=torch.int32, device="cuda")
  out = hip.attn_decode(q, kc, vc, lens)
  for b in range(B):
    ref_b = torch_ref.attn_decode(q[b:b+1], kc[b:b+1], vc[b:b+1], int(lens[b]))
    d = (out[b:b+1].float() - ref_b.float()).abs()
    print(f"ragged b={b} sl={int(lens[b])} maxdiff={d.max().item():.6f}")


if __name__ == "__main__":
  main()
