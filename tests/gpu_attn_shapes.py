#!/usr/bin/env python3
"""Attention fwd/bwd timing, hip_flash vs torch SDPA, at the bench shapes."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import math
import time

import torch
import torch.nn.functional as F

from veomni_amd.ops import hip_lib as L
from veomni_amd.ops.kernels.attention import hip_flash_attention


def bench_shape(B, Hq, Hkv, S, iters=8):
    dev = "cuda"
    scale = 1.0 / math.sqrt(128)
    q = (torch.randn(B, Hq, S, 128, device=dev) * 0.5).to(torch.bfloat16)
    k = (torch.randn(B, Hkv, S, 128, device=dev) * 0.5).to(torch.bfloat16)
    v = (torch.randn(B, Hkv, S, 128, device=dev) * 0.5).to(torch.bfloat16)
    do = torch.randn_like(q)
    fl_f = 2 * 2 * S * S * Hq * 128 * 0.5 * B

    def timed(fn):
        for _ in range(3):
            fn()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(iters):
            fn()
        torch.cuda.synchronize()
        return (time.perf_counter() - t0) / iters

    # hip fwd / bwd
    o, lse = L.attn_fwd(q, k, v, scale)
    t_hf = timed(lambda: L.attn_fwd(q, k, v, scale))
    t_hb = timed(lambda: L.attn_bwd(q, k, v, o, lse, do, scale))

    # sdpa fwd / bwd (incl. the GQA repeat the model path pays)
    def sf():
        kk = k.repeat_interleave(Hq // Hkv, dim=1)
        vv = v.repeat_interleave(Hq // Hkv, dim=1)
        return F.scaled_dot_product_attention(q, kk, vv, is_causal=True, scale=scale)

    t_sf = timed(sf)
    qf = q.clone().requires_grad_(True)
    kf = k.clone().requires_grad_(True)
    vf = v.clone().requires_grad_(True)
    kk = kf.repeat_interleave(Hq // Hkv, dim=1)
    vv = vf.repeat_interleave(Hq // Hkv, dim=1)
    oo = F.scaled_dot_product_attention(qf, kk, vv, is_causal=True, scale=scale)

    def sb():
        oo.backward(do, retain_graph=True)
        qf.grad = kf.grad = vf.grad = None

    t_sb = timed(sb)
    print(f"(B{B},Hq{Hq},Hkv{Hkv},S{S}): hip fwd {t_hf*1e3:7.3f} ms ({fl_f/t_hf/1e12:4.0f} TF/s)"
          f"  bwd {t_hb*1e3:7.3f} ms | sdpa fwd {t_sf*1e3:7.3f}  bwd {t_sb*1e3:7.3f} ms"
          f" | hip/sdpa total {(t_hf+t_hb)/(t_sf+t_sb):.2f}x", flush=True)


def main():
    torch.manual_seed(0)
    bench_shape(1, 32, 8, 4096)    # llama microbench shape
    bench_shape(4, 32, 8, 4096)    # llama bench shape (mbs 4)
    bench_shape(8, 32, 4, 4096)    # qwen3-moe-30b bench shape (mbs 8)
    bench_shape(1, 32, 4, 4096)    # 30B shape at B=1


if __name__ == "__main__":
    main()
