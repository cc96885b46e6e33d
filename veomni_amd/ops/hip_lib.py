"""ctypes binding of libveomni_hip.so (the C ABI in include/veomni_hip.h).

The product path calls HIP kernels ONLY through here. There is no fallback:
if the library is missing on a machine with a GPU, `get_lib()` raises — the
"hip" kernel registrations must never silently degrade to eager/torch.
"""

from __future__ import annotations

import ctypes
import os
from typing import Optional

import torch

_LIB: Optional[ctypes.CDLL] = None

_PKG_DIR = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
SO_PATH = os.path.join(_PKG_DIR, "libveomni_hip.so")

c_i64 = ctypes.c_int64
c_int = ctypes.c_int
c_f32 = ctypes.c_float
c_p = ctypes.c_void_p


def _sig(lib: ctypes.CDLL, name: str, *argtypes) -> None:
    fn = getattr(lib, name)
    fn.restype = ctypes.c_int
    fn.argtypes = list(argtypes)


def get_lib() -> ctypes.CDLL:
    global _LIB
    if _LIB is not None:
        return _LIB
    if not os.path.exists(SO_PATH):
        raise RuntimeError(
            f"libveomni_hip.so not found at {SO_PATH}. Build it with "
            "`python veomni_amd/csrc/build.py` (or __graft_entry__.build()). "
            "The HIP op path has NO fallback by design."
        )
    lib = ctypes.CDLL(SO_PATH)
    lib.vh_last_error.restype = ctypes.c_char_p
    lib.vh_build_info.restype = ctypes.c_char_p
    _sig(lib, "vh_expert_histogram", c_p, c_i64, c_int, c_p, c_p)
    _sig(lib, "vh_moe_scatter_bf16", c_p, c_p, c_p, c_i64, c_i64, c_int, c_p)
    _sig(lib, "vh_moe_gather_bf16", c_p, c_p, c_p, c_i64, c_i64, c_int, c_p)
    _sig(lib, "vh_group_gemm_nk_bf16", c_p, c_p, c_p, c_p, c_int, c_i64, c_i64,
         c_i64, c_int, c_int, c_int, c_p)
    _sig(lib, "vh_group_gemm_nk8_bf16", c_p, c_p, c_p, c_p, c_int, c_i64, c_i64,
         c_i64, c_int, c_p)
    _sig(lib, "vh_group_gemm_nk256_bf16", c_p, c_p, c_p, c_p, c_int, c_i64,
         c_i64, c_i64, c_p)
    _sig(lib, "vh_group_gemm_nk256s_bf16", c_p, c_p, c_p, c_p, c_int, c_i64,
         c_i64, c_i64, c_p)
    _sig(lib, "vh_group_gemm_nk8s_bf16", c_p, c_p, c_p, c_p, c_int, c_i64,
         c_i64, c_i64, c_p)
    _sig(lib, "vh_group_gemm_nkp_bf16", c_p, c_p, c_p, c_p, c_int, c_i64,
         c_i64, c_i64, c_p)
    _sig(lib, "vh_group_gemm_mn_bf16", c_p, c_p, c_p, c_p, c_int, c_i64, c_i64, c_p)
    _sig(lib, "vh_transpose_pad_bf16", c_p, c_p, c_p, c_p, c_int, c_i64, c_i64, c_p)
    _sig(lib, "vh_wtranspose_bf16", c_p, c_p, c_int, c_i64, c_i64, c_p)
    _sig(lib, "vh_group_gemm_wg256_bf16", c_p, c_p, c_p, c_p, c_int, c_i64,
         c_i64, c_i64, c_p)
    _sig(lib, "vh_moe_silu_mul_weighted_bf16", c_p, c_p, c_p, c_i64, c_i64, c_int, c_p)
    _sig(lib, "vh_moe_silu_mul_weighted_bwd_bf16", c_p, c_p, c_p, c_p, c_p,
         c_i64, c_i64, c_int, c_p)
    _sig(lib, "vh_rmsnorm_fwd_bf16", c_p, c_p, c_p, c_p, c_i64, c_i64, c_f32, c_p)
    _sig(lib, "vh_rmsnorm_bwd_bf16", c_p, c_p, c_p, c_p, c_p, c_p, c_i64, c_i64, c_p)
    _sig(lib, "vh_rope_bf16", c_p, c_p, c_p, c_p, c_p, c_p, c_i64, c_i64, c_i64,
         c_i64, c_i64, c_int, c_p)
    _sig(lib, "vh_silu_mul_bf16", c_p, c_p, c_p, c_i64, c_p)
    _sig(lib, "vh_silu_mul_bwd_bf16", c_p, c_p, c_p, c_p, c_p, c_i64, c_p)
    _sig(lib, "vh_ce_fwd_bf16", c_p, c_p, c_p, c_p, c_i64, c_i64, c_f32, c_i64, c_p)
    _sig(lib, "vh_adamw_bf16", c_p, c_p, c_p, c_p, c_p, c_int, c_i64, c_f32,
         c_f32, c_f32, c_f32, c_f32, c_int, c_p, c_p)
    _sig(lib, "vh_attn_fwd_bf16", c_p, c_p, c_p, c_p, c_p, c_int, c_int, c_int,
         c_i64, c_f32, c_p, c_p)
    _sig(lib, "vh_attn_bwd_pre_bf16", c_p, c_p, c_p, c_p, c_p, c_i64, c_p)
    _sig(lib, "vh_attn_bwd_bf16", c_p, c_p, c_p, c_p, c_p, c_p, c_p, c_p, c_p,
         c_int, c_int, c_int, c_i64, c_f32, c_p)
    _sig(lib, "vh_attn_bwd2_bf16", c_p, c_p, c_p, c_p, c_p, c_p, c_p, c_p, c_p,
         c_int, c_int, c_int, c_i64, c_f32, c_p, c_p, c_p)
    _LIB = lib
    return lib


def cur_stream() -> int:
    return torch.cuda.current_stream().cuda_stream


# --------------------------------------------------------- event profiling
# bench.py's roofline leg: when enabled, each wrapped kernel call records a
# (start, end) HIP event pair on ITS launch stream plus the call's
# algorithmic work, so average per-launch durations can be read back.
_PROFILE = False
_PROF_EVENTS: dict = {}


def profile_enable(on: bool = True) -> None:
    global _PROFILE
    _PROFILE = on
    if on:
        _PROF_EVENTS.clear()


def profile_events() -> dict:
    return _PROF_EVENTS


def profile_summary() -> dict:
    """{name: {"count": n, "ms_avg": per-launch ms, "work": [payloads]}}."""
    torch.cuda.synchronize()
    out = {}
    for name, recs in _PROF_EVENTS.items():
        times = [s.elapsed_time(e) for s, e, _ in recs]
        out[name] = {
            "count": len(recs),
            "ms_avg": sum(times) / max(len(times), 1),
            "ms_total": sum(times),
            "work": [w for _, _, w in recs],
        }
    return out


class _prof:
    def __init__(self, name: str, work=None):
        self.name = name
        self.work = work

    def __enter__(self):
        if _PROFILE:
            self.s = torch.cuda.Event(enable_timing=True)
            self.e = torch.cuda.Event(enable_timing=True)
            self.s.record()
        return self

    def __exit__(self, *a):
        if _PROFILE:
            self.e.record()
            _PROF_EVENTS.setdefault(self.name, []).append((self.s, self.e, self.work))


def check(rc: int, what: str) -> None:
    if rc != 0:
        raise RuntimeError(f"{what} failed (rc={rc}): {get_lib().vh_last_error().decode()}")


def dptr(t: torch.Tensor) -> int:
    return t.data_ptr()


# ------------------------------------------------------------- typed wrappers
def expert_histogram(expert_index: torch.Tensor, num_experts: int) -> torch.Tensor:
    assert expert_index.dtype == torch.int64
    flat = expert_index.flatten().contiguous()
    out = torch.empty(num_experts, dtype=torch.int32, device=flat.device)
    check(get_lib().vh_expert_histogram(dptr(flat), flat.numel(), num_experts,
                                        dptr(out), cur_stream()), "vh_expert_histogram")
    return out


def moe_scatter(x: torch.Tensor, index: torch.Tensor) -> torch.Tensor:
    M, N = x.shape
    topk = index.shape[1] if index.dim() == 2 else 1
    idx = index.reshape(-1).to(torch.int32).contiguous()
    out = torch.empty(M * topk, N, dtype=x.dtype, device=x.device)
    check(get_lib().vh_moe_scatter_bf16(dptr(x.contiguous()), dptr(idx), dptr(out),
                                        M, N, topk, cur_stream()), "vh_moe_scatter")
    return out


def moe_gather(x: torch.Tensor, index: torch.Tensor) -> torch.Tensor:
    M, topk = index.shape
    N = x.shape[1]
    idx = index.reshape(-1).to(torch.int32).contiguous()
    out = torch.empty(M, N, dtype=x.dtype, device=x.device)
    check(get_lib().vh_moe_gather_bf16(dptr(x.contiguous()), dptr(idx), dptr(out),
                                       M, N, topk, cur_stream()), "vh_moe_gather")
    return out


def group_gemm_nk(a: torch.Tensor, b: torch.Tensor, cumsum: torch.Tensor,
                  trans_b: bool, c: torch.Tensor | None = None,
                  activation: int = 0) -> torch.Tensor:
    """C_g = A_g @ (B_g^T if trans_b else B_g); rows partitioned by cumsum."""
    assert a.dtype == torch.bfloat16 and b.dtype == torch.bfloat16
    G = b.shape[0]
    N = b.shape[1] if trans_b else b.shape[2]
    K = b.shape[2] if trans_b else b.shape[1]
    assert a.shape[1] == K, (a.shape, b.shape, trans_b)
    if K % 64 != 0:
        # zero-pad the reduction dim to the kernel's BK multiple (only test
        # shapes hit this; every production K — H, I, 2I — is a 64-multiple)
        pad = 64 - K % 64
        a = torch.nn.functional.pad(a, (0, pad))
        b = torch.nn.functional.pad(b, (0, pad) if trans_b else (0, 0, 0, pad))
        K = K + pad
    accumulate = c is not None
    if c is None:
        c = torch.empty(a.shape[0], N, dtype=a.dtype, device=a.device)
    cs = cumsum.to(torch.int64).contiguous()
    # algorithmic flops = 2 * total_rows * N * K (every A row belongs to one group)
    with _prof("group_gemm_nk", 2.0 * a.shape[0] * N * K):
        check(get_lib().vh_group_gemm_nk_bf16(
            dptr(a.contiguous()), dptr(b.contiguous()), dptr(c), dptr(cs), G, N, K,
            a.shape[0], int(trans_b), int(accumulate), activation, cur_stream()),
            "vh_group_gemm_nk")
    return c


def group_gemm_mn(a: torch.Tensor, b: torch.Tensor, cumsum: torch.Tensor,
                  G: int) -> torch.Tensor:
    """C[g] = A_g^T @ B_g; A [rows, M], B [rows, N] -> C [G, M, N].

    Large shapes take the transpose-pad + wg256 path: one memory-bound
    transpose of each operand buys the wgrad a K-contiguous glds kernel
    (~2x the TF/s of transposed register staging). The padded total is the
    host-known bound rows + 64*G, so no device->host sync is needed."""
    assert a.dtype == torch.bfloat16 and b.dtype == torch.bfloat16
    rows = a.shape[0]
    M, N = a.shape[1], b.shape[1]
    c = torch.empty(G, M, N, dtype=a.dtype, device=a.device)
    cs = cumsum.to(torch.int64).contiguous()
    if M >= 512 and N >= 512 and rows >= 2048:
        counts = torch.diff(cs, prepend=cs.new_zeros(1))
        pad_cs = ((counts + 63) // 64 * 64).cumsum(0).contiguous()
        PR = rows + 64 * G
        at = torch.empty(M, PR, dtype=a.dtype, device=a.device)
        bt = torch.empty(N, PR, dtype=a.dtype, device=a.device)
        lib = get_lib()
        with _prof("wgrad_transpose", 2.0 * rows * (M + N) * 2):
            check(lib.vh_transpose_pad_bf16(dptr(a.contiguous()), dptr(at), dptr(cs),
                                            dptr(pad_cs), G, M, PR, cur_stream()),
                  "vh_transpose_pad(a)")
            check(lib.vh_transpose_pad_bf16(dptr(b.contiguous()), dptr(bt), dptr(cs),
                                            dptr(pad_cs), G, N, PR, cur_stream()),
                  "vh_transpose_pad(b)")
        with _prof("group_gemm_mn", 2.0 * rows * M * N):
            check(lib.vh_group_gemm_wg256_bf16(dptr(at), dptr(bt), dptr(c),
                                               dptr(pad_cs), G, M, N, PR,
                                               cur_stream()), "vh_group_gemm_wg256")
        return c
    with _prof("group_gemm_mn", 2.0 * rows * M * N):
        check(get_lib().vh_group_gemm_mn_bf16(
            dptr(a.contiguous()), dptr(b.contiguous()), dptr(c), dptr(cs), G, M, N,
            cur_stream()), "vh_group_gemm_mn")
    return c


def weight_transpose(b: torch.Tensor) -> torch.Tensor:
    """[E, M, N] -> [E, N, M] contiguous (dgrad W^T). HIP tile kernel for
    64-multiple shapes; torch copy otherwise."""
    E, M, N = b.shape
    if M % 64 or N % 64 or b.dtype != torch.bfloat16:
        return b.transpose(1, 2).contiguous()
    out = torch.empty(E, N, M, dtype=b.dtype, device=b.device)
    with _prof("wtranspose", 2.0 * E * M * N * 2):
        check(get_lib().vh_wtranspose_bf16(dptr(b.contiguous()), dptr(out), E,
                                           M, N, cur_stream()), "vh_wtranspose")
    return out


def silu_mul_weighted(fc1: torch.Tensor, w_row: torch.Tensor | None) -> torch.Tensor:
    rows, twoI = fc1.shape
    I = twoI // 2
    out = torch.empty(rows, I, dtype=fc1.dtype, device=fc1.device)
    has_w = w_row is not None
    check(get_lib().vh_moe_silu_mul_weighted_bf16(
        dptr(fc1.contiguous()), dptr(w_row.contiguous()) if has_w else None,
        dptr(out), rows, I, int(has_w), cur_stream()), "vh_moe_silu_mul_weighted")
    return out


def silu_mul_weighted_bwd(dy: torch.Tensor, fc1: torch.Tensor,
                          w_row: torch.Tensor | None):
    rows, twoI = fc1.shape
    I = twoI // 2
    dfc1 = torch.empty_like(fc1)
    has_w = w_row is not None
    dw_row = torch.zeros(rows, dtype=torch.float32, device=fc1.device) if has_w else None
    check(get_lib().vh_moe_silu_mul_weighted_bwd_bf16(
        dptr(dy.contiguous()), dptr(fc1.contiguous()),
        dptr(w_row.contiguous()) if has_w else None, dptr(dfc1),
        dptr(dw_row) if has_w else None, rows, I, int(has_w), cur_stream()),
        "vh_moe_silu_mul_weighted_bwd")
    return dfc1, dw_row


def rmsnorm_fwd(x: torch.Tensor, w: torch.Tensor, eps: float):
    T = x.numel() // x.shape[-1]
    H = x.shape[-1]
    y = torch.empty_like(x)
    rstd = torch.empty(T, dtype=torch.float32, device=x.device)
    with _prof("rmsnorm_fwd", 2.0 * T * H * 2 + T * H * 2):
        check(get_lib().vh_rmsnorm_fwd_bf16(dptr(x.contiguous()), dptr(w.contiguous()),
                                            dptr(y), dptr(rstd), T, H, eps,
                                            cur_stream()), "vh_rmsnorm_fwd")
    return y, rstd


def rmsnorm_bwd(dy: torch.Tensor, x: torch.Tensor, w: torch.Tensor,
                rstd: torch.Tensor):
    T = x.numel() // x.shape[-1]
    H = x.shape[-1]
    dx = torch.empty_like(x)
    dw = torch.zeros(H, dtype=torch.float32, device=x.device)
    check(get_lib().vh_rmsnorm_bwd_bf16(dptr(dy.contiguous()), dptr(x.contiguous()),
                                        dptr(w.contiguous()), dptr(rstd), dptr(dx),
                                        dptr(dw), T, H, cur_stream()), "vh_rmsnorm_bwd")
    return dx, dw


def rope(q: torch.Tensor, k: torch.Tensor, cos: torch.Tensor, sin: torch.Tensor,
         negate_sin: bool = False):
    B, hq, S, D = q.shape
    hk = k.shape[1]
    qo = torch.empty_like(q)
    ko = torch.empty_like(k)
    check(get_lib().vh_rope_bf16(dptr(q.contiguous()), dptr(k.contiguous()),
                                 dptr(cos.contiguous()), dptr(sin.contiguous()),
                                 dptr(qo), dptr(ko), B, hq, hk, S, D,
                                 int(negate_sin), cur_stream()), "vh_rope")
    return qo, ko


def silu_mul(gate: torch.Tensor, up: torch.Tensor) -> torch.Tensor:
    out = torch.empty_like(gate)
    with _prof("silu_mul", 3.0 * gate.numel() * 2):
        check(get_lib().vh_silu_mul_bf16(dptr(gate.contiguous()), dptr(up.contiguous()),
                                         dptr(out), gate.numel(), cur_stream()),
              "vh_silu_mul")
    return out


def silu_mul_bwd(dy: torch.Tensor, gate: torch.Tensor, up: torch.Tensor):
    dg = torch.empty_like(gate)
    du = torch.empty_like(up)
    check(get_lib().vh_silu_mul_bwd_bf16(dptr(dy.contiguous()), dptr(gate.contiguous()),
                                         dptr(up.contiguous()), dptr(dg), dptr(du),
                                         gate.numel(), cur_stream()), "vh_silu_mul_bwd")
    return dg, du


def ce_fwd(logits: torch.Tensor, labels: torch.Tensor, grad_scale: float,
           ignore_index: int = -100, dlogits_out: torch.Tensor | None = None):
    rows, V = logits.shape
    loss_rows = torch.zeros(rows, dtype=torch.float32, device=logits.device)
    dlogits = dlogits_out if dlogits_out is not None else torch.empty_like(logits)
    assert dlogits.is_contiguous() and dlogits.shape == logits.shape
    with _prof("ce_fwd", 3.0 * rows * V * 2):
        check(get_lib().vh_ce_fwd_bf16(dptr(logits.contiguous()),
                                       dptr(labels.to(torch.int64).contiguous()),
                                       dptr(loss_rows), dptr(dlogits), rows, V,
                                       grad_scale, ignore_index, cur_stream()),
              "vh_ce_fwd")
    return loss_rows, dlogits


def attn_fwd(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor, scale: float,
             doc_start: torch.Tensor | None = None):
    """Causal GQA flash attention forward: q [B,Hq,S,128] bf16 contiguous,
    k/v [B,Hkv,S,128]. Returns (o, lse). doc_start (int32 [S], B == 1)
    selects the packed-varlen block-diagonal mask (reference cu_seqlens
    path, attention/flash.py:61-91)."""
    B, Hq, S, D = q.shape
    Hkv = k.shape[1]
    assert D == 128 and S % 256 == 0, (S, D)
    if doc_start is not None:
        assert B == 1 and doc_start.dtype == torch.int32 and doc_start.numel() == S
    o = torch.empty_like(q)
    lse = torch.empty(B, Hq, S, dtype=torch.float32, device=q.device)
    with _prof("attn_fwd", 2.0 * 2 * S * S * Hq * D * 0.5 * B):
        check(get_lib().vh_attn_fwd_bf16(dptr(q), dptr(k), dptr(v), dptr(o),
                                         dptr(lse), B, Hq, Hkv, S, scale,
                                         dptr(doc_start) if doc_start is not None else None,
                                         cur_stream()), "vh_attn_fwd")
    return o, lse


def attn_bwd(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
             o: torch.Tensor, lse: torch.Tensor, do: torch.Tensor,
             scale: float, doc_start: torch.Tensor | None = None,
             doc_end: torch.Tensor | None = None):
    """Backward for attn_fwd. Returns (dq bf16 [B,Hq,S,D], dk/dv bf16
    [B,Hkv,S,D] — the GQA head group is folded inside the dkv kernel)."""
    B, Hq, S, D = q.shape
    Hkv = k.shape[1]
    rows = B * Hq * S
    do = do.contiguous()
    delta = torch.empty(rows, dtype=torch.float32, device=q.device)
    lse2 = torch.empty(rows, dtype=torch.float32, device=q.device)
    assert (doc_start is None) == (doc_end is None)
    lib = get_lib()
    with _prof("attn_bwd", 3.0 * 2 * 2 * S * S * Hq * D * 0.5 * B):
        check(lib.vh_attn_bwd_pre_bf16(dptr(do), dptr(o), dptr(lse.contiguous()),
                                       dptr(delta), dptr(lse2), rows,
                                       cur_stream()), "vh_attn_bwd_pre")
        dq = torch.empty(B, Hq, S, D, dtype=torch.bfloat16, device=q.device)
        dk = torch.empty(B, Hkv, S, D, dtype=torch.bfloat16, device=q.device)
        dv = torch.empty(B, Hkv, S, D, dtype=torch.bfloat16, device=q.device)
        check(lib.vh_attn_bwd2_bf16(dptr(q), dptr(k), dptr(v), dptr(do),
                                    dptr(delta), dptr(lse2), dptr(dq),
                                    dptr(dk), dptr(dv), B, Hq, Hkv, S, scale,
                                    dptr(doc_start) if doc_start is not None else None,
                                    dptr(doc_end) if doc_end is not None else None,
                                    cur_stream()), "vh_attn_bwd2")
    return dq, dk, dv
