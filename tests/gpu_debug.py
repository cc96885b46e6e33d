#!/usr/bin/env python3
"""Standalone GPU kernel diagnostics (run on the GPU box; writes stdout).

Purpose: if pytest -m gpu fails, this pinpoints WHERE (layout transposed?
swizzle broken? staging wrong?) with small printable cases."""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from veomni_amd.ops import hip_lib as L


def hdr(s):
    print(f"\n===== {s} =====", flush=True)


def report(name, out, ref, tol=2e-2):
    ok = torch.allclose(out.float(), ref.float(), rtol=tol, atol=tol)
    err = (out.float() - ref.float()).abs().max().item()
    errT = None
    if out.dim() == 2 and out.shape[0] == out.shape[1]:
        errT = (out.float().t() - ref.float()).abs().max().item()
    print(f"{name}: ok={ok} max_err={err:.4g} max_err_vs_refT={errT}", flush=True)
    if not ok:
        print("out[0,:8] ", out.float()[0, :8].tolist())
        print("ref[0,:8] ", ref.float()[0, :8].tolist())
        print("out[1,:8] ", out.float()[1, :8].tolist())
        print("ref[1,:8] ", ref.float()[1, :8].tolist())
        print("out[:8,0] ", out.float()[:8, 0].tolist())
        print("ref[:8,0] ", ref.float()[:8, 0].tolist())
    return ok


def main():
    torch.manual_seed(0)
    dev = "cuda"
    print("build:", L.get_lib().vh_build_info().decode())
    print("device:", torch.cuda.get_device_name())

    hdr("histogram")
    idx = torch.randint(0, 16, (1000,), device=dev)
    h = L.expert_histogram(idx, 16)
    ref = torch.bincount(idx.cpu(), minlength=16).to(torch.int32)
    print("hist ok:", torch.equal(h.cpu(), ref))

    hdr("silu_mul")
    g = (torch.randn(64, 64, device=dev) * 1.0).to(torch.bfloat16)
    u = (torch.randn(64, 64, device=dev)).to(torch.bfloat16)
    out = L.silu_mul(g, u)
    ref = torch.nn.functional.silu(g.float()) * u.float()
    report("silu_mul", out, ref)

    hdr("rmsnorm")
    x = torch.randn(16, 128, device=dev).to(torch.bfloat16)
    w = (torch.randn(128, device=dev) * 0.1 + 1).to(torch.bfloat16)
    y, rstd = L.rmsnorm_fwd(x, w, 1e-6)
    xf = x.float()
    rs = torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + 1e-6)
    ref = (w.float() * (xf * rs))
    report("rmsnorm_fwd", y, ref)
    print("rstd[:4]", rstd[:4].tolist(), "expect", rs[:4, 0].tolist())

    hdr("group_gemm_nk trans_b single-group identity-ish")
    # B = identity -> C == A (M=64 rows, N=K=64)
    a = (torch.randn(64, 64, device=dev) * 0.5).to(torch.bfloat16)
    b = torch.eye(64, device=dev).to(torch.bfloat16)[None]  # [1,64,64] (sym!)
    cumsum = torch.tensor([64], device=dev)
    c = L.group_gemm_nk(a, b, cumsum, trans_b=True)
    report("nk_identity", c, a.float())

    hdr("group_gemm_nk trans_b asymmetric small")
    M, N, K = 128, 128, 64
    a = (torch.randn(M, K, device=dev) * 0.5).to(torch.bfloat16)
    b = (torch.randn(1, N, K, device=dev) * 0.5).to(torch.bfloat16)
    cumsum = torch.tensor([M], device=dev)
    c = L.group_gemm_nk(a, b, cumsum, trans_b=True)
    ref = a.float() @ b[0].float().t()
    report("nk_asym_128", c, ref)

    hdr("group_gemm_nk trans_b ragged 2 groups")
    counts = [37, 91]
    rows = sum(counts)
    a = (torch.randn(rows, 64, device=dev) * 0.5).to(torch.bfloat16)
    b = (torch.randn(2, 32, 64, device=dev) * 0.5).to(torch.bfloat16)
    cumsum = torch.tensor(counts, device=dev).cumsum(0)
    c = L.group_gemm_nk(a, b, cumsum, trans_b=True)
    ref = torch.cat([a[:37].float() @ b[0].float().t(), a[37:].float() @ b[1].float().t()])
    report("nk_ragged", c, ref)

    hdr("group_gemm_nk NO-trans_b (transposed staging)")
    a = (torch.randn(100, 128, device=dev) * 0.5).to(torch.bfloat16)
    b = (torch.randn(1, 128, 96, device=dev) * 0.5).to(torch.bfloat16)  # [G,K,N]
    cumsum = torch.tensor([100], device=dev)
    c = L.group_gemm_nk(a, b, cumsum, trans_b=False)
    ref = a.float() @ b[0].float()
    report("nk_notrans", c, ref)

    hdr("group_gemm_mn (wgrad)")
    rows = 150
    a = (torch.randn(rows, 64, device=dev) * 0.5).to(torch.bfloat16)
    bb = (torch.randn(rows, 96, device=dev) * 0.5).to(torch.bfloat16)
    cumsum = torch.tensor([100, 150], device=dev)
    c = L.group_gemm_mn(a, bb, cumsum, 2)
    ref0 = a[:100].float().t() @ bb[:100].float()
    ref1 = a[100:].float().t() @ bb[100:].float()
    report("mn_g0", c[0], ref0)
    report("mn_g1", c[1], ref1)

    hdr("bigger gemm K=2048")
    a = (torch.randn(512, 2048, device=dev) * 0.3).to(torch.bfloat16)
    b = (torch.randn(1, 1536, 2048, device=dev) * 0.3).to(torch.bfloat16)
    cumsum = torch.tensor([512], device=dev)
    c = L.group_gemm_nk(a, b, cumsum, trans_b=True)
    ref = a.float() @ b[0].float().t()
    report("nk_big", c, ref, tol=3e-2)

    hdr("ce_fwd")
    logits = (torch.randn(16, 512, device=dev) * 2).to(torch.bfloat16)
    labels = torch.randint(0, 512, (16,), device=dev)
    lr, dl = L.ce_fwd(logits, labels, 1.0 / 16)
    lf = logits.float().cpu().requires_grad_(True)
    ref = torch.nn.functional.cross_entropy(lf, labels.cpu(), reduction="sum") / 16
    ref.backward()
    print("ce loss:", float(lr.sum() / 16), "ref:", float(ref))
    report("ce_grad", dl.cpu(), lf.grad)

    print("\nALL DONE", flush=True)


if __name__ == "__main__":
    main()
