import sys, time
sys.path.insert(0, "/root/repo")
import torch
from veomni_amd.ops import hip_lib as L
dev="cuda"
for (T,H) in [(16384,4096),(16384,2048),(8192,4096)]:
    x=(torch.randn(T,H,device=dev)*0.5).to(torch.bfloat16)
    dy=torch.randn_like(x)
    w=torch.randn(H,device=dev).to(torch.bfloat16)
    y,rstd=L.rmsnorm_fwd(x,w,1e-6)
    for _ in range(3): L.rmsnorm_bwd(dy,x,w,rstd)
    torch.cuda.synchronize(); t0=time.perf_counter()
    for _ in range(20): L.rmsnorm_bwd(dy,x,w,rstd)
    torch.cuda.synchronize(); dt=(time.perf_counter()-t0)/20
    alg=3*T*H*2
    print(f"rmsnorm_bwd T{T} H{H}: {dt*1e3:.3f} ms  {alg/dt/1e9:.0f} GB/s algorithmic", flush=True)

# qk-norm shape (per-head H=128)
for (T,H) in [(65536,128),(131072,128)]:
    x=(torch.randn(T,H,device=dev)*0.5).to(torch.bfloat16)
    dy=torch.randn_like(x)
    w=torch.randn(H,device=dev).to(torch.bfloat16)
    y,rstd=L.rmsnorm_fwd(x,w,1e-6)
    for _ in range(3): L.rmsnorm_bwd(dy,x,w,rstd)
    torch.cuda.synchronize(); t0=time.perf_counter()
    for _ in range(20): L.rmsnorm_bwd(dy,x,w,rstd)
    torch.cuda.synchronize(); dt=(time.perf_counter()-t0)/20
    alg=3*T*H*2
    print(f"rmsnorm_bwd T{T} H{H}: {dt*1e3:.3f} ms  {alg/dt/1e9:.0f} GB/s algorithmic", flush=True)
