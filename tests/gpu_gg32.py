#!/usr/bin/env python3
"""nk256s32 (32x32x16 MFMA) vs nk256s: parity vs torch ref + timing."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import ctypes

import torch

from veomni_amd.ops import hip_lib as L

GG_ARGS = [ctypes.c_void_p] * 4 + [ctypes.c_int, ctypes.c_int64,
                                   ctypes.c_int64, ctypes.c_int64,
                                   ctypes.c_void_p]


def gg(fn, a, w, cumsum, G, N, K):
    c = torch.empty(a.shape[0], N, dtype=a.dtype, device=a.device)
    rc = fn(a.data_ptr(), w.data_ptr(), c.data_ptr(), cumsum.data_ptr(),
            G, N, K, a.shape[0], L.cur_stream())
    assert rc == 0
    torch.cuda.synchronize()
    return c


def main():
    lib = L.get_lib()
    for name in ("vh_group_gemm_nk256s_bf16", "vh_group_gemm_nk256s32_bf16"):
        fn = getattr(lib, name)
        fn.restype = ctypes.c_int
        fn.argtypes = GG_ARGS
    torch.manual_seed(0)
    dev = "cuda"
    # parity at a small ragged shape
    G, N, K = 8, 256, 128
    sizes = torch.tensor([64, 0, 300, 256, 17, 512, 1, 130], device=dev)
    cumsum = sizes.cumsum(0)
    rows = int(cumsum[-1])
    a = (torch.randn(rows, K, device=dev) * 0.3).to(torch.bfloat16)
    w = (torch.randn(G, N, K, device=dev) * 0.3).to(torch.bfloat16)
    c32 = gg(lib.vh_group_gemm_nk256s32_bf16, a, w, cumsum, G, N, K)
    start = 0
    for g in range(G):
        end = int(cumsum[g])
        if end > start:
            ref = (a[start:end].float() @ w[g].float().t())
            err = (c32[start:end].float() - ref).abs().max().item()
            rel = err / ref.abs().max().item()
            assert rel < 2e-2, (g, rel)
        start = end
    print("parity ok (ragged G=8)", flush=True)

    # timing at the bench fc1 shape and fc2 shape
    for (G, rows, N, K, tag) in ((128, 262144, 1536, 2048, "fc1"),
                                 (128, 262144, 2048, 768, "fc2")):
        cumsum = torch.full((G,), rows // G, device=dev).cumsum(0)
        a = (torch.randn(rows, K, device=dev) * 0.3).to(torch.bfloat16)
        w = (torch.randn(G, N, K, device=dev) * 0.3).to(torch.bfloat16)
        flops = 2.0 * rows * N * K
        for fn, name in ((lib.vh_group_gemm_nk256s_bf16, "nk256s"),
                         (lib.vh_group_gemm_nk256s32_bf16, "nk256s32"),
                         (lib.vh_group_gemm_nk256s_bf16, "nk256s-b"),
                         (lib.vh_group_gemm_nk256s32_bf16, "nk256s32-b")):
            for _ in range(3):
                gg(fn, a, w, cumsum, G, N, K)
            t0 = torch.cuda.Event(enable_timing=True)
            t1 = torch.cuda.Event(enable_timing=True)
            t0.record()
            for _ in range(10):
                gg(fn, a, w, cumsum, G, N, K)
            t1.record()
            torch.cuda.synchronize()
            ms = t0.elapsed_time(t1) / 10
            print(f"{tag} {name}: {ms:.3f} ms  {flops/ms/1e9:.0f} TF/s", flush=True)


if __name__ == "__main__":
    main()
