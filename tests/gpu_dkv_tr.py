#!/usr/bin/env python3
"""dkv TR (tr16 transpose-read) probe vs dispatched v6: parity at a small
shape vs fp32 autograd, then timing at the microbench shape."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import ctypes
import math

import torch

from veomni_amd.ops import hip_lib as L


def bind(lib):
    fn = lib.vh_attn_bwd2_dkv6probe_bf16
    fn.restype = ctypes.c_int
    fn.argtypes = [ctypes.c_void_p] * 8 + [ctypes.c_int] * 3 + [
        ctypes.c_int64, ctypes.c_float, ctypes.c_int, ctypes.c_void_p]
    return fn


def run(fn, q, k, v, do, delta, lse2, scale, pref):
    B, Hq, S = q.shape[0], q.shape[1], q.shape[2]
    Hkv = k.shape[1]
    dk = torch.zeros(B, Hkv, S, 128, dtype=torch.bfloat16, device="cuda")
    dv = torch.zeros_like(dk)
    rc = fn(q.data_ptr(), k.data_ptr(), v.data_ptr(), do.data_ptr(),
            delta.data_ptr(), lse2.data_ptr(), dk.data_ptr(), dv.data_ptr(),
            B, Hq, Hkv, S, scale, pref, L.cur_stream())
    assert rc == 0
    torch.cuda.synchronize()
    return dk, dv


def prep(B, Hq, Hkv, S):
    torch.manual_seed(0)
    scale = 1.0 / math.sqrt(128)
    q = (torch.randn(B, Hq, S, 128, device="cuda") * 0.5).to(torch.bfloat16)
    k = (torch.randn(B, Hkv, S, 128, device="cuda") * 0.5).to(torch.bfloat16)
    v = (torch.randn(B, Hkv, S, 128, device="cuda") * 0.5).to(torch.bfloat16)
    do = (torch.randn(B, Hq, S, 128, device="cuda") * 0.5).to(torch.bfloat16)
    o, lse = L.attn_fwd(q, k, v, scale)
    rows = B * Hq * S
    delta = torch.empty(rows, dtype=torch.float32, device="cuda")
    lse2 = torch.empty(rows, dtype=torch.float32, device="cuda")
    lib = L.get_lib()
    L.check(lib.vh_attn_bwd_pre_bf16(L.dptr(do), L.dptr(o),
                                     L.dptr(lse.contiguous()), L.dptr(delta),
                                     L.dptr(lse2), rows, L.cur_stream()), "pre")
    return q, k, v, do, delta, lse2, scale


def bind_dq(lib):
    fn = lib.vh_attn_bwd2_dqprobe_bf16
    fn.restype = ctypes.c_int
    fn.argtypes = [ctypes.c_void_p] * 7 + [ctypes.c_int] * 3 + [
        ctypes.c_int64, ctypes.c_float, ctypes.c_int, ctypes.c_void_p]
    return fn


def run_dq(fn, q, k, v, do, delta, lse2, scale, mode):
    B, Hq, S = q.shape[0], q.shape[1], q.shape[2]
    dq = torch.zeros(B, Hq, S, 128, dtype=torch.bfloat16, device="cuda")
    rc = fn(q.data_ptr(), k.data_ptr(), v.data_ptr(), do.data_ptr(),
            delta.data_ptr(), lse2.data_ptr(), dq.data_ptr(),
            B, Hq, k.shape[1], S, scale, mode, L.cur_stream())
    assert rc == 0
    torch.cuda.synchronize()
    return dq


def main():
    fn = bind(L.get_lib())
    fnq = bind_dq(L.get_lib())

    # parity: TR vs the dispatched v6 (itself fp32-ref-verified) at two shapes
    for shp in ((1, 4, 2, 256), (2, 8, 2, 512)):
        q, k, v, do, delta, lse2, scale = prep(*shp)
        dk0, dv0 = run(fn, q, k, v, do, delta, lse2, scale, 0)
        for mode, nm in ((20, "TR"), (21, "TR2"), (22, "TR2G")):
            dk1, dv1 = run(fn, q, k, v, do, delta, lse2, scale, mode)
            ek = (dk1.float() - dk0.float()).abs().max().item()
            ev = (dv1.float() - dv0.float()).abs().max().item()
            print(f"shape {shp}: {nm}-vs-v6 |dK|={ek:.4g} |dV|={ev:.4g}", flush=True)
            assert ek == 0.0 and ev == 0.0, f"{nm} must be bit-identical"

    for shp in ((1, 4, 2, 256), (2, 8, 2, 512)):
        q, k, v, do, delta, lse2, scale = prep(*shp)
        dq0 = run_dq(fnq, q, k, v, do, delta, lse2, scale, 0)
        for mode, nm in ((20, "TRQ"), (21, "TRQ2"), (22, "TRQ3")):
            dq1 = run_dq(fnq, q, k, v, do, delta, lse2, scale, mode)
            eq = (dq1.float() - dq0.float()).abs().max().item()
            print(f"shape {shp}: dq {nm}-vs-v0 |dQ|={eq:.4g}", flush=True)
            assert eq == 0.0

    # timing at the microbench shape
    q, k, v, do, delta, lse2, scale = prep(1, 32, 8, 8192)
    for pref, name in ((0, "v6"), (21, "TR2"), (22, "TR2G"), (21, "TR2b"), (22, "TR2Gb")):
        for _ in range(3):
            run(fn, q, k, v, do, delta, lse2, scale, pref)
        t0 = torch.cuda.Event(enable_timing=True)
        t1 = torch.cuda.Event(enable_timing=True)
        t0.record()
        for _ in range(10):
            run(fn, q, k, v, do, delta, lse2, scale, pref)
        t1.record()
        torch.cuda.synchronize()
        print(f"{name}: {t0.elapsed_time(t1) / 10 * 1000:.0f} us", flush=True)
    for mode, name in ((0, "dq-v0"), (20, "dq-TRQ"), (22, "dq-TRQ3"), (20, "dq-TRQb"), (22, "dq-TRQ3b")):
        for _ in range(3):
            run_dq(fnq, q, k, v, do, delta, lse2, scale, mode)
        t0 = torch.cuda.Event(enable_timing=True)
        t1 = torch.cuda.Event(enable_timing=True)
        t0.record()
        for _ in range(10):
            run_dq(fnq, q, k, v, do, delta, lse2, scale, mode)
        t1.record()
        torch.cuda.synchronize()
        print(f"{name}: {t0.elapsed_time(t1) / 10 * 1000:.0f} us", flush=True)


if __name__ == "__main__":
    main()
