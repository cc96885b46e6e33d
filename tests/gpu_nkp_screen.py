#!/usr/bin/env python3
"""Race-screen + A/B for the 8-phase pipelined grouped GEMM (nkp).

Guide discipline for NEW sync templates: multi-run refcheck across shapes
(incl. ragged/skew splits) + within-probe A/B vs the unmodified kernels."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from veomni_amd.ops import hip_lib as L


def run_nkp(a, w, cumsum, N, K):
    lib = L.get_lib()
    c = torch.empty(a.shape[0], N, dtype=torch.bfloat16, device=a.device)
    rc = lib.vh_group_gemm_nkp_bf16(a.data_ptr(), w.data_ptr(), c.data_ptr(),
                                    cumsum.data_ptr(), w.shape[0], N, K,
                                    a.shape[0], L.cur_stream())
    assert rc == 0, lib.vh_last_error()
    return c


def refcheck(G, counts, N, K, seed):
    dev = "cuda"
    torch.manual_seed(seed)
    cumsum = torch.tensor(counts, device=dev).cumsum(0)
    rows = int(cumsum[-1])
    a = (torch.randn(rows, K, device=dev) * 0.3).to(torch.bfloat16)
    w = (torch.randn(G, N, K, device=dev) * 0.3).to(torch.bfloat16)
    c = run_nkp(a, w, cumsum, N, K)
    torch.cuda.synchronize()
    start = 0
    for g in range(G):
        end = int(cumsum[g])
        if end > start:
            ref = a[start:end].float() @ w[g].float().t()
            if not torch.allclose(c[start:end].float(), ref, rtol=2e-2, atol=2e-2):
                err = (c[start:end].float() - ref).abs().max().item()
                return f"FAIL G{G} N{N} K{K} counts{counts[:4]}.. group {g} err {err}"
        start = end
    return None


def main():
    shapes = [
        (16, [2048] * 16, 1536, 2048),
        (16, [2048] * 16, 2048, 768),
        (8, [0, 4096, 256, 1, 7937, 0, 510, 3584], 1536, 2048),   # ragged+empty
        (8, [16384, 0, 0, 0, 0, 0, 0, 0], 1536, 2048),            # full skew
        (4, [100, 200, 300, 400], 512, 256),                       # small
        (128, [512] * 128, 1536, 2048),                            # bench G
    ]
    for rep in range(3):   # multi-run race screen
        for i, (G, counts, N, K) in enumerate(shapes):
            r = refcheck(G, counts, N, K, seed=rep * 10 + i)
            if r:
                print(r, flush=True)
                return
    print("race-screen OK (3 runs x 6 shapes)", flush=True)

    # A/B timing
    def timeit(fn, iters=10):
        for _ in range(3):
            fn()
        torch.cuda.synchronize()
        ts = []
        for _ in range(iters):
            t0 = time.perf_counter()
            fn()
            torch.cuda.synchronize()
            ts.append(time.perf_counter() - t0)
        ts.sort()
        return ts[len(ts) // 2]

    dev = "cuda"
    lib = L.get_lib()
    for (G, rows, N, K, tag) in [(16, 32768, 1536, 2048, "fc1 EP8"),
                                 (16, 32768, 2048, 768, "fc2 EP8"),
                                 (128, 65536, 1536, 2048, "fc1 G128"),
                                 (128, 262144, 1536, 2048, "fc1 G128 mbs8")]:
        torch.manual_seed(0)
        cumsum = torch.full((G,), rows // G, device=dev).cumsum(0)
        a = (torch.randn(rows, K, device=dev) * 0.3).to(torch.bfloat16)
        w = (torch.randn(G, N, K, device=dev) * 0.3).to(torch.bfloat16)
        c = torch.empty(rows, N, dtype=torch.bfloat16, device=dev)
        fl = 2.0 * rows * N * K

        def nkp():
            rc = lib.vh_group_gemm_nkp_bf16(a.data_ptr(), w.data_ptr(), c.data_ptr(),
                                            cumsum.data_ptr(), G, N, K, rows, L.cur_stream())
            assert rc == 0

        def nk256s():
            rc = lib.vh_group_gemm_nk256s_bf16(a.data_ptr(), w.data_ptr(), c.data_ptr(),
                                               cumsum.data_ptr(), G, N, K, rows, L.cur_stream())
            assert rc == 0

        t1, t2 = timeit(nkp), timeit(nk256s)
        print(f"{tag} (M{rows} N{N} K{K} G{G}): nkp {t1*1e3:.2f} ms {fl/t1/1e12:.0f} TF/s"
              f" | nk256s {t2*1e3:.2f} ms {fl/t2/1e12:.0f} TF/s", flush=True)


if __name__ == "__main__":
    main()
