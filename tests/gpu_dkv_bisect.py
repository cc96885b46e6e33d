#!/usr/bin/env python3
"""On-device bisect of the padded dkv_g kernel: variants read A/B frags from
LDS or global. Err per variant tells which read path mis-indexes."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import ctypes
import math

import torch

from veomni_amd.ops import hip_lib as L


def main():
    torch.manual_seed(0)
    dev = "cuda"
    B, Hq, Hkv, S = 1, 2, 1, 256
    scale = 1.0 / math.sqrt(128)
    q = (torch.randn(B, Hq, S, 128, device=dev) * 0.5).to(torch.bfloat16)
    k = (torch.randn(B, Hkv, S, 128, device=dev) * 0.5).to(torch.bfloat16)
    v = (torch.randn(B, Hkv, S, 128, device=dev) * 0.5).to(torch.bfloat16)
    do = (torch.randn(B, Hq, S, 128, device=dev) * 0.5).to(torch.bfloat16)
    o, lse = L.attn_fwd(q, k, v, scale)

    # fp32 reference
    rep = Hq // Hkv
    qf = q.float().requires_grad_(True)
    kf = k.float().requires_grad_(True)
    vf = v.float().requires_grad_(True)
    kk = kf.repeat_interleave(rep, dim=1)
    vv = vf.repeat_interleave(rep, dim=1)
    sc = torch.matmul(qf, kk.transpose(-1, -2)) * scale
    mask = torch.triu(torch.ones(S, S, dtype=torch.bool, device=dev), 1)
    p = torch.softmax(sc.masked_fill(mask, float("-inf")), dim=-1)
    torch.matmul(p, vv).backward(do.float())

    rows = B * Hq * S
    delta = torch.empty(rows, dtype=torch.float32, device=dev)
    lse2 = torch.empty(rows, dtype=torch.float32, device=dev)
    lib = L.get_lib()
    L.check(lib.vh_attn_bwd_pre_bf16(L.dptr(do), L.dptr(o), L.dptr(lse.contiguous()),
                                     L.dptr(delta), L.dptr(lse2), rows,
                                     L.cur_stream()), "pre")
    fn6 = lib.vh_attn_bwd2_dkv6probe_bf16
    fn6.restype = ctypes.c_int
    fn6.argtypes = [ctypes.c_void_p] * 8 + [ctypes.c_int] * 3 + [
        ctypes.c_int64, ctypes.c_float, ctypes.c_int, ctypes.c_void_p]
    for var in (0, 3, 7, 15):
        dk = torch.zeros(B, Hkv, S, 128, dtype=torch.bfloat16, device=dev)
        dv = torch.zeros(B, Hkv, S, 128, dtype=torch.bfloat16, device=dev)
        rc = fn6(q.data_ptr(), k.data_ptr(), v.data_ptr(), do.data_ptr(),
                 delta.data_ptr(), lse2.data_ptr(), dk.data_ptr(), dv.data_ptr(),
                 B, Hq, Hkv, S, scale, var, L.cur_stream())
        assert rc == 0
        torch.cuda.synchronize()
        ekr = (dk.float() - kf.grad).abs().amax(dim=(0, 1, 3))  # per kv row
        evr = (dv.float() - vf.grad).abs().amax(dim=(0, 1, 3))
        ek = ekr.max().item()
        ev = evr.max().item()
        print(f"variant {var} (A={'glb' if var & 1 else 'lds'} "
              f"B={'glb' if var & 2 else 'lds'}): |dK err|={ek:.4g} |dV err|={ev:.4g}",
              flush=True)
        if var == 0:
            # per-32-row error profile: which kv slices are wrong?
            prof = [round(evr[i:i + 32].max().item(), 3) for i in range(0, S, 32)]
            print("  dV err per 32-kv-row block:", prof, flush=True)
            print("  dv[0,0,0,:4]   ", dv.float()[0, 0, 0, :4].tolist(), flush=True)
            print("  dvref[0,0,0,:4]", vf.grad[0, 0, 0, :4].tolist(), flush=True)
            print("  dv[0,0,40,:4]  ", dv.float()[0, 0, 40, :4].tolist(), flush=True)
            print("  dvref[0,0,40,:4]", vf.grad[0, 0, 40, :4].tolist(), flush=True)


if __name__ == "__main__":
    main()
