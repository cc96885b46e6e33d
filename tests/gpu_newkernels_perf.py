"""Microbench: VeAdamW sweep, weight transpose, transpose-pad (evidence)."""
import sys, time
sys.path.insert(0, "/root/repo")
import torch
from veomni_amd.ops import hip_lib as L

dev = "cuda"
# ---- VeAdamW on a 30B-like param set
from veomni_amd.optim import VeAdamW
torch.manual_seed(0)
params = []
for _ in range(48):
    params.append(torch.randn(128, 1536, 2048, dtype=torch.bfloat16, device=dev).requires_grad_(True))
for _ in range(300):
    params.append(torch.randn(2048, 2048, dtype=torch.bfloat16, device=dev).requires_grad_(True))
total = sum(p.numel() for p in params)
opt = VeAdamW(params, lr=1e-5)
for p in params:
    p.grad = torch.randn_like(p)
opt.step()
torch.cuda.synchronize()
t0 = time.perf_counter()
for _ in range(5):
    opt.step()
torch.cuda.synchronize()
dt = (time.perf_counter() - t0) / 5
traffic = total * 14  # p,g,m,v read (8B) + p,m,v write (6B)
print(f"veadamw: {total/1e9:.1f}B params, {dt*1e3:.2f} ms/step, {traffic/dt/1e9:.0f} GB/s", flush=True)

# ---- weight transpose
b = torch.randn(128, 1536, 2048, dtype=torch.bfloat16, device=dev)
out = L.weight_transpose(b)
assert torch.equal(out, b.transpose(1, 2).contiguous())
torch.cuda.synchronize(); t0 = time.perf_counter()
for _ in range(10):
    L.weight_transpose(b)
torch.cuda.synchronize()
dt = (time.perf_counter() - t0) / 10
tr = 2 * b.numel() * 2
print(f"wtranspose [128,1536,2048]: {dt*1e3:.3f} ms, {tr/dt/1e9:.0f} GB/s", flush=True)
t0 = time.perf_counter()
for _ in range(10):
    b.transpose(1, 2).contiguous()
torch.cuda.synchronize()
dt2 = (time.perf_counter() - t0) / 10
print(f"torch copy same: {dt2*1e3:.3f} ms, {tr/dt2/1e9:.0f} GB/s", flush=True)
