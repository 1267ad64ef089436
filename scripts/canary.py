import torch, time
a = torch.randn(8192, 8192, device="cuda", dtype=torch.bfloat16)
b = torch.randn(8192, 8192, device="cuda", dtype=torch.bfloat16)
for _ in range(3): c = a @ b
torch.cuda.synchronize()
t = time.perf_counter()
for _ in range(20): c = a @ b
torch.cuda.synchronize()
dt = (time.perf_counter() - t) / 20
print(f"canary gemm: {2*8192**3/dt/1e12:.1f} TFLOP/s")
