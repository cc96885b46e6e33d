#!/usr/bin/env python3
"""Microbench: grouped GEMM + memory-bound kernels at north-star shapes.

Shapes: EP=8 on Qwen3-30B-A3B -> per rank 16 local experts,
32768 scattered rows, fc1 [16,1536,2048], fc2 [16,2048,768].
Prints TF/s and GB/s per kernel (within-run repeats, median)."""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from veomni_amd.ops import hip_lib as L


def timeit(fn, iters=10, warmup=3):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    times = []
    for _ in range(iters):
        t0 = time.perf_counter()
        fn()
        torch.cuda.synchronize()
        times.append(time.perf_counter() - t0)
    times.sort()
    return times[len(times) // 2]


def main():
    dev = "cuda"
    torch.manual_seed(0)
    E, rows = 16, 32768
    H, I = 2048, 768
    counts = torch.full((E,), rows // E)
    cumsum = counts.cumsum(0).to(dev)

    a = (torch.randn(rows, H, device=dev) * 0.3).to(torch.bfloat16)
    w1 = (torch.randn(E, 2 * I, H, device=dev) * 0.3).to(torch.bfloat16)
    w2 = (torch.randn(E, H, I, device=dev) * 0.3).to(torch.bfloat16)
    act = (torch.randn(rows, I, device=dev) * 0.3).to(torch.bfloat16)

    # fc1 fwd: [rows,H] x [E,2I,H]^T
    fl = 2.0 * rows * 2 * I * H
    t = timeit(lambda: L.group_gemm_nk(a, w1, cumsum, trans_b=True))
    print(f"nk fc1 fwd  (M{rows} N{2*I} K{H} G{E}): {t*1e3:.2f} ms  {fl/t/1e12:.0f} TF/s")

    # fc2 fwd: [rows,I] x [E,H,I]^T
    fl = 2.0 * rows * H * I
    t = timeit(lambda: L.group_gemm_nk(act, w2, cumsum, trans_b=True))
    print(f"nk fc2 fwd  (M{rows} N{H} K{I} G{E}): {t*1e3:.2f} ms  {fl/t/1e12:.0f} TF/s")

    # dgrad fc2: [rows,H] x [E,H,I]
    g2 = (torch.randn(rows, H, device=dev) * 0.3).to(torch.bfloat16)
    fl = 2.0 * rows * H * I
    t = timeit(lambda: L.group_gemm_nk(g2, w2, cumsum, trans_b=False))
    print(f"nk fc2 dgrad(M{rows} N{I} K{H} G{E}): {t*1e3:.2f} ms  {fl/t/1e12:.0f} TF/s")

    # dgrad fc1: [rows,2I] x [E,2I,H]
    g1 = (torch.randn(rows, 2 * I, device=dev) * 0.3).to(torch.bfloat16)
    fl = 2.0 * rows * 2 * I * H
    t = timeit(lambda: L.group_gemm_nk(g1, w1, cumsum, trans_b=False))
    print(f"nk fc1 dgrad(M{rows} N{H} K{2*I} G{E}): {t*1e3:.2f} ms  {fl/t/1e12:.0f} TF/s")

    # wgrad fc2 via wrapper (transpose+wg256 path when eligible)
    fl = 2.0 * rows * H * I
    t = timeit(lambda: L.group_gemm_mn(g2, act, cumsum, E))
    print(f"mn fc2 wgrad(M{H} N{I} k{rows} G{E}): {t*1e3:.2f} ms  {fl/t/1e12:.0f} TF/s")

    # wgrad via transpose-pad + wg256 (the wrapper path at large shapes)
    fl = 2.0 * rows * 2 * I * H
    t = timeit(lambda: L.group_gemm_mn(g1, a, cumsum, E))
    print(f"wg256 fc1 wgrad (M{2*I} N{H} k{rows} G{E}): {t*1e3:.2f} ms  {fl/t/1e12:.0f} TF/s (incl transposes)")

    # nk256 variant (A/B vs the 128^2 fwd above)
    fl = 2.0 * rows * 2 * I * H
    lib2 = L.get_lib()
    c256 = torch.empty(rows, 2 * I, dtype=torch.bfloat16, device=dev)
    def run256():
        rc = lib2.vh_group_gemm_nk256_bf16(a.data_ptr(), w1.data_ptr(), c256.data_ptr(),
                                           cumsum.data_ptr(), E, 2 * I, H, rows, L.cur_stream())
        assert rc == 0, lib2.vh_last_error()
    t = timeit(run256)
    ref = L.group_gemm_nk(a, w1, cumsum, trans_b=True)
    ok = torch.allclose(c256.float(), ref.float(), rtol=2e-2, atol=2e-2)
    print(f"nk256 fc1 fwd (M{rows} N{2*I} K{H} G{E}): {t*1e3:.2f} ms  {fl/t/1e12:.0f} TF/s  correct={ok}")
    fl = 2.0 * rows * H * I
    c256b = torch.empty(rows, H, dtype=torch.bfloat16, device=dev)
    def run256b():
        rc = lib2.vh_group_gemm_nk256_bf16(act.data_ptr(), w2.data_ptr(), c256b.data_ptr(),
                                           cumsum.data_ptr(), E, H, I, rows, L.cur_stream())
        assert rc == 0, lib2.vh_last_error()
    t = timeit(run256b)
    print(f"nk256 fc2 fwd (M{rows} N{H} K{I} G{E}): {t*1e3:.2f} ms  {fl/t/1e12:.0f} TF/s")

    # nk256s (XCD-clustered persistent schedule) A/B at the same shapes
    def run256s():
        rc = lib2.vh_group_gemm_nk256s_bf16(a.data_ptr(), w1.data_ptr(), c256.data_ptr(),
                                            cumsum.data_ptr(), E, 2 * I, H, rows, L.cur_stream())
        assert rc == 0, lib2.vh_last_error()
    fl = 2.0 * rows * 2 * I * H
    t = timeit(run256s)
    ok = torch.allclose(c256.float(), ref.float(), rtol=2e-2, atol=2e-2)
    print(f"nk256s fc1 fwd (M{rows} N{2*I} K{H} G{E}): {t*1e3:.2f} ms  {fl/t/1e12:.0f} TF/s  correct={ok}")
    def run256sb():
        rc = lib2.vh_group_gemm_nk256s_bf16(act.data_ptr(), w2.data_ptr(), c256b.data_ptr(),
                                            cumsum.data_ptr(), E, H, I, rows, L.cur_stream())
        assert rc == 0, lib2.vh_last_error()
    fl = 2.0 * rows * H * I
    t = timeit(run256sb)
    print(f"nk256s fc2 fwd (M{rows} N{H} K{I} G{E}): {t*1e3:.2f} ms  {fl/t/1e12:.0f} TF/s")

    # nk8s (counted-vmcnt ring + XCD schedule) A/B
    def run8s():
        rc = lib2.vh_group_gemm_nk8s_bf16(a.data_ptr(), w1.data_ptr(), c256.data_ptr(),
                                          cumsum.data_ptr(), E, 2 * I, H, rows, L.cur_stream())
        assert rc == 0, lib2.vh_last_error()
    fl = 2.0 * rows * 2 * I * H
    t = timeit(run8s)
    ok = torch.allclose(c256.float(), ref.float(), rtol=2e-2, atol=2e-2)
    print(f"nk8s  fc1 fwd (M{rows} N{2*I} K{H} G{E}): {t*1e3:.2f} ms  {fl/t/1e12:.0f} TF/s  correct={ok}")
    def run8sb():
        rc = lib2.vh_group_gemm_nk8s_bf16(act.data_ptr(), w2.data_ptr(), c256b.data_ptr(),
                                          cumsum.data_ptr(), E, H, I, rows, L.cur_stream())
        assert rc == 0, lib2.vh_last_error()
    fl = 2.0 * rows * H * I
    t = timeit(run8sb)
    print(f"nk8s  fc2 fwd (M{rows} N{H} K{I} G{E}): {t*1e3:.2f} ms  {fl/t/1e12:.0f} TF/s")

    # non-EP bench shape: G=128 experts (the N=1 30B case), 64k rows sample
    E2, rows2 = 128, 65536
    cs2 = torch.full((E2,), rows2 // E2).cumsum(0).to(dev)
    a2 = (torch.randn(rows2, H, device=dev) * 0.3).to(torch.bfloat16)
    w1b = (torch.randn(E2, 2 * I, H, device=dev) * 0.3).to(torch.bfloat16)
    c2 = torch.empty(rows2, 2 * I, dtype=torch.bfloat16, device=dev)
    fl = 2.0 * rows2 * 2 * I * H
    def run256_g128():
        rc = lib2.vh_group_gemm_nk256_bf16(a2.data_ptr(), w1b.data_ptr(), c2.data_ptr(),
                                           cs2.data_ptr(), E2, 2 * I, H, rows2, L.cur_stream())
        assert rc == 0, lib2.vh_last_error()
    t = timeit(run256_g128)
    print(f"nk256  fc1 fwd (M{rows2} N{2*I} K{H} G{E2}): {t*1e3:.2f} ms  {fl/t/1e12:.0f} TF/s")
    def run256s_g128():
        rc = lib2.vh_group_gemm_nk256s_bf16(a2.data_ptr(), w1b.data_ptr(), c2.data_ptr(),
                                            cs2.data_ptr(), E2, 2 * I, H, rows2, L.cur_stream())
        assert rc == 0, lib2.vh_last_error()
    def run8s_g128():
        rc = lib2.vh_group_gemm_nk8s_bf16(a2.data_ptr(), w1b.data_ptr(), c2.data_ptr(),
                                          cs2.data_ptr(), E2, 2 * I, H, rows2, L.cur_stream())
        assert rc == 0, lib2.vh_last_error()
    t8 = timeit(run8s_g128)
    print(f"nk8s   fc1 fwd (M{rows2} N{2*I} K{H} G{E2}): {t8*1e3:.2f} ms  {fl/t8/1e12:.0f} TF/s")
    t = timeit(run256s_g128)
    gsz = rows2 // E2
    ok2 = all(
        torch.allclose(c2[g * gsz:(g + 1) * gsz].float(),
                       (a2[g * gsz:(g + 1) * gsz].float() @ w1b[g].float().t()),
                       rtol=2e-2, atol=2e-2)
        for g in (0, 1, 63, 127))
    print(f"nk256s fc1 fwd (M{rows2} N{2*I} K{H} G{E2}): {t*1e3:.2f} ms  {fl/t/1e12:.0f} TF/s  correct={ok2}")

    # hipBLASLt comparison: one dense bf16 GEMM of the fc1-fwd size
    bd = w1.reshape(E * 2 * I, H)
    t = timeit(lambda: torch.matmul(a, bd.t()))
    fl = 2.0 * rows * E * 2 * I * H
    print(f"hipBLASLt dense (M{rows} N{E*2*I} K{H}): {t*1e3:.2f} ms  {fl/t/1e12:.0f} TF/s  (upper ref)")

    # memory-bound kernels
    T = 4096
    x = (torch.randn(T, H, device=dev)).to(torch.bfloat16)
    w = (torch.randn(H, device=dev) * 0.1 + 1).to(torch.bfloat16)
    by = 2.0 * T * H * 2 + T * H * 2
    t = timeit(lambda: L.rmsnorm_fwd(x, w, 1e-6), iters=20)
    print(f"rmsnorm fwd (T{T} H{H}): {t*1e6:.0f} us  {by/t/1e9:.0f} GB/s")

    sel = torch.randint(0, E, (T, 8), device=dev)
    from veomni_amd.ops.kernels.moe import compute_expert_scatter_index

    _, sidx = compute_expert_scatter_index(sel)
    t = timeit(lambda: L.moe_scatter(x, sidx), iters=20)
    by = (1 + 8) * T * H * 2
    print(f"moe_scatter (T{T} topk8 H{H}): {t*1e6:.0f} us  {by/t/1e9:.0f} GB/s")

    big = (torch.randn(T * 8, H, device=dev)).to(torch.bfloat16)
    t = timeit(lambda: L.moe_gather(big, sidx), iters=20)
    print(f"moe_gather  (T{T} topk8 H{H}): {t*1e6:.0f} us  {by/t/1e9:.0f} GB/s")

    fc1 = (torch.randn(rows, 2 * I, device=dev)).to(torch.bfloat16)
    wr = torch.rand(rows, device=dev).to(torch.bfloat16)
    by = rows * (2 * I + I) * 2
    t = timeit(lambda: L.silu_mul_weighted(fc1, wr), iters=20)
    print(f"silu_mul_w  (rows{rows} I{I}): {t*1e6:.0f} us  {by/t/1e9:.0f} GB/s")

    V = 151936
    logits = (torch.randn(2048, V, device=dev) * 2).to(torch.bfloat16)
    labels = torch.randint(0, V, (2048,), device=dev)
    by = 3.0 * 2048 * V * 2
    t = timeit(lambda: L.ce_fwd(logits, labels, 1.0 / 2048), iters=5)
    print(f"ce_fwd      (rows2048 V{V}): {t*1e3:.2f} ms  {by/t/1e9:.0f} GB/s")

    # skewed expert load (realistic routing)
    sel_sk = torch.multinomial(torch.rand(E) + 0.1, rows, replacement=True)
    counts_sk = torch.bincount(sel_sk, minlength=E)
    cs_sk = counts_sk.cumsum(0).to(dev)
    fl = 2.0 * rows * 2 * I * H
    t = timeit(lambda: L.group_gemm_nk(a, w1, cs_sk, trans_b=True))
    print(f"nk fc1 fwd skewed (max {int(counts_sk.max())}): {t*1e3:.2f} ms  {fl/t/1e12:.0f} TF/s")


if __name__ == "__main__":
    main()
