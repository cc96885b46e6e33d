#!/usr/bin/env python3
"""Flash-attention forward kernel vs fp32 torch reference (debug harness)."""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import ctypes
import math
import time

import torch

from veomni_amd.ops import hip_lib as L


def attn_fwd(q, k, v, scale):
    return L.attn_fwd(q.contiguous(), k.contiguous(), v.contiguous(), scale)


def ref_attn(q, k, v, scale):
    B, Hq, S, D = q.shape
    Hkv = k.shape[1]
    rep = Hq // Hkv
    kk = k.repeat_interleave(rep, dim=1).float()
    vv = v.repeat_interleave(rep, dim=1).float()
    s = torch.matmul(q.float(), kk.transpose(-1, -2)) * scale
    mask = torch.triu(torch.ones(S, S, dtype=torch.bool, device=q.device), 1)
    s = s.masked_fill(mask, float("-inf"))
    lse = torch.logsumexp(s, dim=-1)
    p = torch.softmax(s, dim=-1)
    return torch.matmul(p, vv), lse


def main():
    torch.manual_seed(0)
    dev = "cuda"
    for (B, Hq, Hkv, S) in [(1, 1, 1, 256), (2, 4, 2, 256), (1, 8, 2, 1024)]:
        q = (torch.randn(B, Hq, S, 128, device=dev) * 0.5).to(torch.bfloat16)
        k = (torch.randn(B, Hkv, S, 128, device=dev) * 0.5).to(torch.bfloat16)
        v = (torch.randn(B, Hkv, S, 128, device=dev) * 0.5).to(torch.bfloat16)
        scale = 1.0 / math.sqrt(128)
        o, lse = attn_fwd(q, k, v, scale)
        oref, lref = ref_attn(q, k, v, scale)
        err = (o.float() - oref).abs().max().item()
        lerr = (lse - lref).abs().max().item()
        ok = err < 3e-2 and lerr < 1e-3
        print(f"B{B} Hq{Hq} Hkv{Hkv} S{S}: max|dO|={err:.4g} max|dLSE|={lerr:.4g} ok={ok}", flush=True)
        if not ok:
            print("o[0,0,:2,:6]   ", o.float()[0, 0, :2, :6].tolist())
            print("oref[0,0,:2,:6]", oref[0, 0, :2, :6].tolist())
            print("o[0,0,64,:6]   ", o.float()[0, 0, 64, :6].tolist())
            print("oref[0,0,64,:6]", oref[0, 0, 64, :6].tolist())
            print("lse[0,0,:8]    ", lse[0, 0, :8].tolist())
            print("lref[0,0,:8]   ", lref[0, 0, :8].tolist())
            return

    # spike test: force a late max jump (guide rule 26)
    B, Hq, Hkv, S = 1, 2, 1, 512
    q = (torch.randn(B, Hq, S, 128, device=dev) * 0.3).to(torch.bfloat16)
    k = (torch.randn(B, Hkv, S, 128, device=dev) * 0.3).to(torch.bfloat16)
    v = (torch.randn(B, Hkv, S, 128, device=dev) * 0.3).to(torch.bfloat16)
    k[0, 0, 300] = (q[0, 0, 400].float() * 4).to(torch.bfloat16)  # spike vs q row 400
    scale = 1.0 / math.sqrt(128)
    o, lse = attn_fwd(q, k, v, scale)
    oref, lref = ref_attn(q, k, v, scale)
    err = (o.float() - oref).abs().max().item()
    print(f"spike: max|dO|={err:.4g} ok={err < 3e-2}", flush=True)

    # ---- backward parity vs torch autograd (fp32 reference)
    for (B, Hq, Hkv, S) in [(1, 1, 1, 256), (1, 2, 1, 512), (2, 4, 2, 256), (1, 8, 2, 1024)]:
        q = (torch.randn(B, Hq, S, 128, device=dev) * 0.5).to(torch.bfloat16)
        k = (torch.randn(B, Hkv, S, 128, device=dev) * 0.5).to(torch.bfloat16)
        v = (torch.randn(B, Hkv, S, 128, device=dev) * 0.5).to(torch.bfloat16)
        do = (torch.randn(B, Hq, S, 128, device=dev) * 0.5).to(torch.bfloat16)
        scale = 1.0 / math.sqrt(128)
        o, lse = attn_fwd(q, k, v, scale)
        dq, dk, dv_ = L.attn_bwd(q, k, v, o, lse, do, scale)
        qf = q.float().requires_grad_(True)
        kf = k.float().requires_grad_(True)
        vf = v.float().requires_grad_(True)
        oref, _ = ref_attn(qf, kf, vf, scale)
        oref.backward(do.float())
        edq = (dq.float() - qf.grad).abs().max().item()
        edk = (dk.float() - kf.grad).abs().max().item()
        edv = (dv_.float() - vf.grad).abs().max().item()
        sc = qf.grad.abs().max().item()
        ok = edq < 0.06 * max(sc, 1.0) and edk < 0.06 * max(kf.grad.abs().max().item(), 1.0) \
            and edv < 0.06 * max(vf.grad.abs().max().item(), 1.0)
        print(f"bwd B{B} Hq{Hq} Hkv{Hkv} S{S}: max|ddq|={edq:.4g} max|ddk|={edk:.4g} "
              f"max|ddv|={edv:.4g} (|dq|max={sc:.3g}) ok={ok}", flush=True)
        if not ok:
            print("dq[0,0,:2,:6]    ", dq.float()[0, 0, :2, :6].tolist())
            print("dqref[0,0,:2,:6] ", qf.grad[0, 0, :2, :6].tolist())
            print("dk[0,0,:2,:6]    ", dk.float()[0, 0, :2, :6].tolist())
            print("dkref[0,0,:2,:6] ", kf.grad[0, 0, :2, :6].tolist())
            print("dv[0,0,:2,:6]    ", dv_.float()[0, 0, :2, :6].tolist())
            print("dvref[0,0,:2,:6] ", vf.grad[0, 0, :2, :6].tolist())
            return

    # backward timing at llama shape
    B, Hq, Hkv, S = 1, 32, 8, 4096
    scale = 1.0 / math.sqrt(128)
    q = (torch.randn(B, Hq, S, 128, device=dev) * 0.5).to(torch.bfloat16)
    k = (torch.randn(B, Hkv, S, 128, device=dev) * 0.5).to(torch.bfloat16)
    v = (torch.randn(B, Hkv, S, 128, device=dev) * 0.5).to(torch.bfloat16)
    do = torch.randn_like(q)
    o, lse = attn_fwd(q, k, v, scale)
    for _ in range(3):
        L.attn_bwd(q, k, v, o, lse, do, scale)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(10):
        L.attn_bwd(q, k, v, o, lse, do, scale)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / 10
    flb = 3 * 2 * 2 * S * S * Hq * 128 * 0.5
    print(f"bwd llama shape: {dt*1e3:.3f} ms  {flb/dt/1e12:.0f} TF/s", flush=True)

    # A/B: round-1 per-Q-head dkv (v2 probe) vs the GQA-folded v5 in the path
    lib = L.get_lib()
    fn2 = lib.vh_attn_bwd2_dkv2probe_bf16
    fn2.restype = ctypes.c_int
    fn2.argtypes = [ctypes.c_void_p] * 8 + [ctypes.c_int] * 3 + [
        ctypes.c_int64, ctypes.c_float, ctypes.c_void_p]
    rows = B * Hq * S
    delta = torch.zeros(rows, dtype=torch.float32, device=dev)
    lse2 = (lse.flatten() * 1.4426950408889634).contiguous()
    dkh = torch.empty(B, Hq, S, 128, dtype=torch.bfloat16, device=dev)
    dvh = torch.empty(B, Hq, S, 128, dtype=torch.bfloat16, device=dev)
    for _ in range(3):
        fn2(q.data_ptr(), k.data_ptr(), v.data_ptr(), do.data_ptr(),
            delta.data_ptr(), lse2.data_ptr(), dkh.data_ptr(), dvh.data_ptr(),
            B, Hq, Hkv, S, scale, L.cur_stream())
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(10):
        fn2(q.data_ptr(), k.data_ptr(), v.data_ptr(), do.data_ptr(),
            delta.data_ptr(), lse2.data_ptr(), dkh.data_ptr(), dvh.data_ptr(),
            B, Hq, Hkv, S, scale, L.cur_stream())
    torch.cuda.synchronize()
    dt2p = (time.perf_counter() - t0) / 10
    print(f"dkv v2 probe (per-Q-head): {dt2p*1e3:.3f} ms", flush=True)

    # dkv v6 prefetch-depth A/B (PREF = 0 / 4 / 8)
    fn6 = lib.vh_attn_bwd2_dkv6probe_bf16
    fn6.restype = ctypes.c_int
    fn6.argtypes = [ctypes.c_void_p] * 8 + [ctypes.c_int] * 3 + [
        ctypes.c_int64, ctypes.c_float, ctypes.c_int, ctypes.c_void_p]
    dkv = torch.empty(B, Hkv, S, 128, dtype=torch.bfloat16, device=dev)
    dvv = torch.empty(B, Hkv, S, 128, dtype=torch.bfloat16, device=dev)
    for pref in (0, 4, 8):
        for _ in range(3):
            fn6(q.data_ptr(), k.data_ptr(), v.data_ptr(), do.data_ptr(),
                delta.data_ptr(), lse2.data_ptr(), dkv.data_ptr(), dvv.data_ptr(),
                B, Hq, Hkv, S, scale, pref, L.cur_stream())
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(10):
            fn6(q.data_ptr(), k.data_ptr(), v.data_ptr(), do.data_ptr(),
                delta.data_ptr(), lse2.data_ptr(), dkv.data_ptr(), dvv.data_ptr(),
                B, Hq, Hkv, S, scale, pref, L.cur_stream())
        torch.cuda.synchronize()
        dt6 = (time.perf_counter() - t0) / 10
        print(f"dkv v6 pref={pref}: {dt6*1e3:.3f} ms", flush=True)

    # AOTriton reference: time the torch SDPA backward at the same shape
    qf = q.clone().requires_grad_(True)
    kf = k.clone().requires_grad_(True)
    vf = v.clone().requires_grad_(True)
    import torch.nn.functional as F
    kk4 = kf.repeat_interleave(Hq // Hkv, dim=1)
    vv4 = vf.repeat_interleave(Hq // Hkv, dim=1)
    oo = F.scaled_dot_product_attention(qf, kk4, vv4, is_causal=True, scale=scale)
    for _ in range(3):
        oo.backward(do, retain_graph=True)
        qf.grad = kf.grad = vf.grad = None
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(10):
        oo.backward(do, retain_graph=True)
        qf.grad = kf.grad = vf.grad = None
    torch.cuda.synchronize()
    dts = (time.perf_counter() - t0) / 10
    print(f"torch sdpa bwd (incl. autograd + GQA-repeat grads): {dts*1e3:.3f} ms", flush=True)

    # timing at llama shape
    B, Hq, Hkv, S = 1, 32, 8, 4096
    q = (torch.randn(B, Hq, S, 128, device=dev) * 0.5).to(torch.bfloat16)
    k = (torch.randn(B, Hkv, S, 128, device=dev) * 0.5).to(torch.bfloat16)
    v = (torch.randn(B, Hkv, S, 128, device=dev) * 0.5).to(torch.bfloat16)
    for _ in range(3):
        o, lse = attn_fwd(q, k, v, scale)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(10):
        o, lse = attn_fwd(q, k, v, scale)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / 10
    fl = 2 * 2 * S * S * Hq * 128 * 0.5
    print(f"fwd llama shape: {dt*1e3:.3f} ms  {fl/dt/1e12:.0f} TF/s", flush=True)
    # sdpa comparison
    import torch.nn.functional as F
    kk = k.repeat_interleave(4, dim=1)
    vv = v.repeat_interleave(4, dim=1)
    for _ in range(3):
        F.scaled_dot_product_attention(q, kk, vv, is_causal=True, scale=scale)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(10):
        F.scaled_dot_product_attention(q, kk, vv, is_causal=True, scale=scale)
    torch.cuda.synchronize()
    dt2 = (time.perf_counter() - t0) / 10
    print(f"torch sdpa:      {dt2*1e3:.3f} ms  {fl/dt2/1e12:.0f} TF/s", flush=True)


if __name__ == "__main__":
    main()


def probe():
    import ctypes
    lib = L.get_lib()
    fn = lib.vh_attn_fwd_probe_bf16
    fn.restype = ctypes.c_int
    fn.argtypes = [ctypes.c_void_p] * 5 + [ctypes.c_int] * 3 + [ctypes.c_int64, ctypes.c_float, ctypes.c_int, ctypes.c_void_p]
    dev = "cuda"
    B, Hq, Hkv, S = 1, 32, 8, 4096
    import math
    scale = 1.0 / math.sqrt(128)
    q = (torch.randn(B, Hq, S, 128, device=dev) * 0.5).to(torch.bfloat16)
    k = (torch.randn(B, Hkv, S, 128, device=dev) * 0.5).to(torch.bfloat16)
    v = (torch.randn(B, Hkv, S, 128, device=dev) * 0.5).to(torch.bfloat16)
    o = torch.empty_like(q)
    lse = torch.empty(B, Hq, S, dtype=torch.float32, device=dev)
    fl = 2 * 2 * S * S * Hq * 128 * 0.5
    for mode, name in [(0, "full"), (1, "no-softmax"), (2, "no-PV"), (3, "no-QK")]:
        for _ in range(3):
            fn(q.data_ptr(), k.data_ptr(), v.data_ptr(), o.data_ptr(), lse.data_ptr(),
               B, Hq, Hkv, S, scale, mode, L.cur_stream())
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(10):
            fn(q.data_ptr(), k.data_ptr(), v.data_ptr(), o.data_ptr(), lse.data_ptr(),
               B, Hq, Hkv, S, scale, mode, L.cur_stream())
        torch.cuda.synchronize()
        dt = (time.perf_counter() - t0) / 10
        print(f"probe {name}: {dt*1e3:.3f} ms ({fl/dt/1e12:.0f} TF/s-equiv)", flush=True)


if os.environ.get("VH_ATTN_PROBE"):
    probe()


def probe_bwd():
    import ctypes
    lib = L.get_lib()
    fn = lib.vh_attn_bwd_probe_bf16
    fn.restype = ctypes.c_int
    fn.argtypes = [ctypes.c_void_p] * 9 + [ctypes.c_int] * 3 + [ctypes.c_int64, ctypes.c_float, ctypes.c_int, ctypes.c_void_p]
    dev = "cuda"
    B, Hq, Hkv, S = 1, 32, 8, 4096
    scale = 1.0 / math.sqrt(128)
    q = (torch.randn(B, Hq, S, 128, device=dev) * 0.5).to(torch.bfloat16)
    k = (torch.randn(B, Hkv, S, 128, device=dev) * 0.5).to(torch.bfloat16)
    v = (torch.randn(B, Hkv, S, 128, device=dev) * 0.5).to(torch.bfloat16)
    do = torch.randn_like(q)
    o, lse = attn_fwd(q, k, v, scale)
    rows = B * Hq * S
    delta = torch.zeros(rows, dtype=torch.float32, device=dev)
    lse2 = lse.flatten() * 1.4426950408889634
    dqacc = torch.zeros(B, Hq, S, 128, dtype=torch.float32, device=dev)
    dk = torch.empty(B, Hq, S, 128, dtype=torch.bfloat16, device=dev)
    dv = torch.empty(B, Hq, S, 128, dtype=torch.bfloat16, device=dev)
    flb = 3 * 2 * 2 * S * S * Hq * 128 * 0.5
    names = {0: "full", 1: "no-dQ", 2: "no-phase2(dV/dK)", 3: "no-softmax", 4: "no-qt-staging", 5: "no-qrow-loads", 6: "no-global-flush", 7: "no-lds-dsadd", 8: "no-dq-mfma"}
    for mode in [0, 1, 2, 3, 4, 5, 6, 7, 8]:
        for _ in range(2):
            rc = fn(q.data_ptr(), k.data_ptr(), v.data_ptr(), do.data_ptr(),
                    delta.data_ptr(), lse2.data_ptr(), dqacc.data_ptr(),
                    dk.data_ptr(), dv.data_ptr(), B, Hq, Hkv, S, scale, mode, L.cur_stream())
            assert rc == 0, lib.vh_last_error()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(5):
            fn(q.data_ptr(), k.data_ptr(), v.data_ptr(), do.data_ptr(),
               delta.data_ptr(), lse2.data_ptr(), dqacc.data_ptr(),
               dk.data_ptr(), dv.data_ptr(), B, Hq, Hkv, S, scale, mode, L.cur_stream())
        torch.cuda.synchronize()
        dt = (time.perf_counter() - t0) / 5
        print(f"bwd probe {names[mode]}: {dt*1e3:.3f} ms ({flb/dt/1e12:.0f} TF/s-equiv)", flush=True)


if os.environ.get("VH_ATTN_PROBE_BWD"):
    probe_bwd()
