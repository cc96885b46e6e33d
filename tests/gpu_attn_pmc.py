#!/usr/bin/env python3
"""Minimal attention-backward launcher for rocprofv3 PMC passes (llama shape)."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import math

import torch

from veomni_amd.ops import hip_lib as L


def main():
    torch.manual_seed(0)
    dev = "cuda"
    B, Hq, Hkv, S = 1, 32, 8, 4096
    scale = 1.0 / math.sqrt(128)
    q = (torch.randn(B, Hq, S, 128, device=dev) * 0.5).to(torch.bfloat16)
    k = (torch.randn(B, Hkv, S, 128, device=dev) * 0.5).to(torch.bfloat16)
    v = (torch.randn(B, Hkv, S, 128, device=dev) * 0.5).to(torch.bfloat16)
    do = torch.randn_like(q)
    o, lse = L.attn_fwd(q, k, v, scale)
    for _ in range(6):
        o, lse = L.attn_fwd(q, k, v, scale)
        L.attn_bwd(q, k, v, o, lse, do, scale)
    torch.cuda.synchronize()
    print("pmc run done")


if __name__ == "__main__":
    main()
