import sys, os, math, time
sys.path.insert(0, "/root/repo")
import torch, torch.nn.functional as F
from veomni_amd.ops import hip_lib as L
dev = "cuda"
B, Hq, Hkv, S = 1, 32, 8, 4096
scale = 1.0 / math.sqrt(128)
torch.manual_seed(0)
q = (torch.randn(B, Hq, S, 128, device=dev) * 0.5).to(torch.bfloat16).requires_grad_(True)
kk = (torch.randn(B, Hq, S, 128, device=dev) * 0.5).to(torch.bfloat16).requires_grad_(True)
vv = (torch.randn(B, Hq, S, 128, device=dev) * 0.5).to(torch.bfloat16).requires_grad_(True)
do = torch.randn(B, Hq, S, 128, device=dev).to(torch.bfloat16)
fl = 2 * 2 * S * S * Hq * 128 * 0.5
flb = 3 * fl
def fwdbwd():
    o = F.scaled_dot_product_attention(q, kk, vv, is_causal=True, scale=scale)
    o.backward(do)
    q.grad = kk.grad = vv.grad = None
for _ in range(3): fwdbwd()
torch.cuda.synchronize()
# separate fwd and bwd timing via autograd.grad
o = F.scaled_dot_product_attention(q, kk, vv, is_causal=True, scale=scale)
torch.cuda.synchronize()
t0 = time.perf_counter()
for _ in range(10):
    g = torch.autograd.grad(o, (q, kk, vv), do, retain_graph=True)
torch.cuda.synchronize()
dt = (time.perf_counter() - t0) / 10
print(f"sdpa bwd-only: {dt*1e3:.3f} ms ({flb/dt/1e12:.0f} TF/s)", flush=True)
# our bwd for comparison
k = kk.detach()[:, :8].contiguous(); v = vv.detach()[:, :8].contiguous()
qd = q.detach().contiguous()
ob, lse = L.attn_fwd(qd, k, v, scale)
for _ in range(3): L.attn_bwd(qd, k, v, ob, lse, do, scale)
torch.cuda.synchronize()
t0 = time.perf_counter()
for _ in range(10): L.attn_bwd(qd, k, v, ob, lse, do, scale)
torch.cuda.synchronize()
dt2 = (time.perf_counter() - t0) / 10
print(f"our bwd (wrapper incl GQA sum): {dt2*1e3:.3f} ms ({flb/dt2/1e12:.0f} TF/s)", flush=True)
