import torch, time
M, K, N = 65536, 8192, 57344
x = (torch.randn(M, K, device="cuda") * 0.1).to(torch.bfloat16)
w = (torch.randn(N, K, device="cuda") * 0.02).to(torch.bfloat16)
# bf16 baseline
for _ in range(3): y = torch.nn.functional.linear(x, w)
torch.cuda.synchronize(); t0 = time.perf_counter()
for _ in range(5): y = torch.nn.functional.linear(x, w)
torch.cuda.synchronize(); t_bf = (time.perf_counter() - t0) / 5
# fp8 scaled_mm
sx = x.float().abs().amax(dim=1, keepdim=True).clamp(min=1e-12) / 448.0
x8 = (x.float() / sx).clamp(-448, 448).to(torch.float8_e4m3fn)
swt = w.float().abs().amax(dim=1, keepdim=True).clamp(min=1e-12) / 448.0
w8 = (w.float() / swt).clamp(-448, 448).to(torch.float8_e4m3fn)
try:
  for _ in range(3):
    y8 = torch._scaled_mm(x8, w8.t(), scale_a=sx, scale_b=swt.t(), out_dtype=torch.bfloat16)
  torch.cuda.synchronize(); t0 = time.perf_counter()
  for _ in range(5):
    y8 = torch._scaled_mm(x8, w8.t(), scale_a=sx, scale_b=swt.t(), out_dtype=torch.bfloat16)
  torch.cuda.synchronize(); t_f8 = (time.perf_counter() - t0) / 5
  err = (y8.float() - y.float()).abs().max().item() / y.float().abs().max().item()
  print(f"bf16 {t_bf*1e3:.2f} ms ({2*M*K*N/t_bf/1e15:.2f} PF); fp8 {t_f8*1e3:.2f} ms ({2*M*K*N/t_f8/1e15:.2f} PF); rel err {err:.3f}")
except Exception as e:
  print("scaled_mm failed:", type(e).__name__, str(e)[:300])
