"""HIP-backed RMSNorm / RoPE / SwiGLU OpSlot kernels (impl name "hip").

Slot signatures mirror the reference's Liger registrations
(ops/liger/__init__.py:28-153):
  rms_norm(hidden_states, weight, eps) -> Tensor
  rotary_pos_emb(q, k, cos, sin, position_ids=None, unsqueeze_dim=1) -> (q, k)
  swiglu_mlp(self, x) -> Tensor   (= down_proj(silu_mul(gate(x), up(x))))
"""

from __future__ import annotations

import torch

from .. import hip_lib
from ..kernel_registry import KERNEL_REGISTRY, HardwareRequirement, KernelSpec


class HipRMSNorm(torch.autograd.Function):
    @staticmethod
    def forward(ctx, hidden_states, weight, eps):
        assert hidden_states.dtype == torch.bfloat16, "hip rms_norm is bf16-only"
        x = hidden_states.contiguous()
        y, rstd = hip_lib.rmsnorm_fwd(x, weight, eps)
        ctx.save_for_backward(x, weight, rstd)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, weight, rstd = ctx.saved_tensors
        dx, dw = hip_lib.rmsnorm_bwd(dy, x, weight, rstd)
        return dx, dw.to(weight.dtype), None


def hip_rms_norm(hidden_states, weight, eps):
    return HipRMSNorm.apply(hidden_states, weight, eps)


class HipRoPE(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, cos, sin):
        assert q.dtype == torch.bfloat16
        qe, ke = hip_lib.rope(q, k, cos, sin, negate_sin=False)
        ctx.save_for_backward(cos, sin)
        return qe, ke

    @staticmethod
    def backward(ctx, dq, dk):
        cos, sin = ctx.saved_tensors
        # rope backward = rope with negated sin (duplicated-half table)
        dqi, dki = hip_lib.rope(dq, dk, cos, sin, negate_sin=True)
        return dqi, dki, None, None


def hip_rotary_pos_emb(q, k, cos, sin, position_ids=None, unsqueeze_dim=1):
    # cos/sin arrive [B, S, D]; kernel indexes them per (b, s) row directly.
    return HipRoPE.apply(q, k, cos.to(q.dtype), sin.to(q.dtype))


class HipSiluMul(torch.autograd.Function):
    @staticmethod
    def forward(ctx, gate, up):
        assert gate.dtype == torch.bfloat16
        out = hip_lib.silu_mul(gate.contiguous(), up.contiguous())
        ctx.save_for_backward(gate, up)
        return out

    @staticmethod
    def backward(ctx, dy):
        gate, up = ctx.saved_tensors
        dg, du = hip_lib.silu_mul_bwd(dy, gate, up)
        return dg, du


class HipFc1SiluMul(torch.autograd.Function):
    """silu(fc1[:, :I]) * fc1[:, I:] on the fused gate|up projection output
    (vh_moe_silu_mul_weighted with has_w=0) — one pass over [T, 2I]."""

    @staticmethod
    def forward(ctx, fc1):
        assert fc1.dtype == torch.bfloat16
        flat = fc1.reshape(-1, fc1.shape[-1]).contiguous()
        out = hip_lib.silu_mul_weighted(flat, None)
        ctx.save_for_backward(flat)
        return out.view(*fc1.shape[:-1], fc1.shape[-1] // 2)

    @staticmethod
    def backward(ctx, dy):
        (flat,) = ctx.saved_tensors
        dfc1, _ = hip_lib.silu_mul_weighted_bwd(
            dy.reshape(-1, dy.shape[-1]), flat, None)
        return dfc1.view(*dy.shape[:-1], dy.shape[-1] * 2)


def hip_swiglu_mlp(self, x):
    # single fused gate|up GEMM (one wide hipBLASLt call instead of two) and
    # a one-pass SiLU-mul epilogue over the fused fc1
    w12 = torch.cat((self.gate_proj.weight, self.up_proj.weight))
    fc1 = torch.nn.functional.linear(x, w12)
    return self.down_proj(HipFc1SiluMul.apply(fc1))


for op_name, variant, fn, desc in [
    ("rms_norm", "standard", lambda: hip_rms_norm, "gfx950 fused RMSNorm (vh_rmsnorm_*)"),
    ("rotary_pos_emb", "full", lambda: hip_rotary_pos_emb, "gfx950 fused RoPE (vh_rope)"),
    ("swiglu_mlp", "standard", lambda: hip_swiglu_mlp, "gfx950 SiLU-mul (vh_silu_mul)"),
]:
    KERNEL_REGISTRY.register(
        KernelSpec(name="hip", op_name=op_name, variant=variant, factory=fn,
                   hardware=HardwareRequirement(device_type="gpu"), description=desc)
    )
