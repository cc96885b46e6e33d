#!/usr/bin/env python3
"""fwd TRF (tr16 V image) probe vs dispatched forward: bit-parity + timing."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import ctypes
import math

import torch

from veomni_amd.ops import hip_lib as L


def main():
    lib = L.get_lib()
    fn = lib.vh_attn_fwd_probe_bf16
    fn.restype = ctypes.c_int
    fn.argtypes = [ctypes.c_void_p] * 5 + [ctypes.c_int] * 3 + [
        ctypes.c_int64, ctypes.c_float, ctypes.c_int, ctypes.c_void_p]

    def run(q, k, v, mode):
        B, Hq, S = q.shape[0], q.shape[1], q.shape[2]
        o = torch.zeros_like(q)
        lse = torch.zeros(B * Hq * S, dtype=torch.float32, device="cuda")
        rc = fn(q.data_ptr(), k.data_ptr(), v.data_ptr(), o.data_ptr(),
                lse.data_ptr(), B, Hq, k.shape[1], S,
                1.0 / math.sqrt(128), mode, L.cur_stream())
        assert rc == 0
        torch.cuda.synchronize()
        return o, lse

    for shp in ((1, 4, 2, 256), (2, 8, 2, 512)):
        torch.manual_seed(0)
        B, Hq, Hkv, S = shp
        q = (torch.randn(B, Hq, S, 128, device="cuda") * 0.5).to(torch.bfloat16)
        k = (torch.randn(B, Hkv, S, 128, device="cuda") * 0.5).to(torch.bfloat16)
        v = (torch.randn(B, Hkv, S, 128, device="cuda") * 0.5).to(torch.bfloat16)
        o0, l0 = run(q, k, v, 0)
        for mode, nm in ((20, "TRF"), (21, "TRF2")):
            o1, l1 = run(q, k, v, mode)
            eo = (o1.float() - o0.float()).abs().max().item()
            el = (l1 - l0).abs().max().item()
            print(f"shape {shp}: {nm}-vs-v0 |O|={eo:.4g} |LSE|={el:.4g}", flush=True)
            assert eo == 0.0 and el == 0.0

    torch.manual_seed(0)
    q = (torch.randn(1, 32, 8192, 128, device="cuda") * 0.5).to(torch.bfloat16)
    k = (torch.randn(1, 8, 8192, 128, device="cuda") * 0.5).to(torch.bfloat16)
    v = (torch.randn(1, 8, 8192, 128, device="cuda") * 0.5).to(torch.bfloat16)
    for mode, name in ((0, "v0"), (21, "TRF2"), (0, "v0b"), (21, "TRF2b")):
        for _ in range(3):
            run(q, k, v, mode)
        t0 = torch.cuda.Event(enable_timing=True)
        t1 = torch.cuda.Event(enable_timing=True)
        t0.record()
        for _ in range(10):
            run(q, k, v, mode)
        t1.record()
        torch.cuda.synchronize()
        print(f"{name}: {t0.elapsed_time(t1) / 10 * 1000:.0f} us", flush=True)


if __name__ == "__main__":
    main()
