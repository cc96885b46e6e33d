"""CPU oracle for the memory-bound hot-path ops: RMSNorm, RoPE, SwiGLU.

Restates the eager semantics the reference's Liger-backed OpSlots replace:
  /root/reference/veomni/models/transformers/qwen3_moe/generated/
      patched_modeling_qwen3_moe_gpu.py  (Qwen3MoeRMSNorm ~:345-360,
      rotate_half/apply_rotary_pos_emb :86-111, Qwen3MoeMLP.forward :236-243)
  /root/reference/veomni/ops/liger/__init__.py:28-60,115-153 (slot signatures:
      rms_norm(hidden_states, weight, eps); rotary(q,k,cos,sin,position_ids,
      unsqueeze_dim); swiglu_mlp(self, x) = down(silu(gate(x)) * up(x))).

Test infrastructure only — see oracle/__init__.py.
"""

from __future__ import annotations

import torch


def rms_norm(hidden_states: torch.Tensor, weight: torch.Tensor, eps: float) -> torch.Tensor:
    """x * rsqrt(mean(x^2) + eps) * w, variance in fp32, output in input dtype.

    Ref: Qwen3MoeRMSNorm.forward (patched_modeling_qwen3_moe_gpu.py):
    hidden cast to fp32, variance = mean(x^2, -1), normalized, cast back,
    then multiplied by weight (weight multiply AFTER the downcast).
    """
    input_dtype = hidden_states.dtype
    x = hidden_states.to(torch.float32)
    variance = x.pow(2).mean(-1, keepdim=True)
    x = x * torch.rsqrt(variance + eps)
    return weight * x.to(input_dtype)


def rms_norm_bwd(dy: torch.Tensor, x: torch.Tensor, weight: torch.Tensor, eps: float):
    """Autograd-derived backward of `rms_norm` (fp32 math), returns (dx, dw)."""
    x = x.detach().clone().requires_grad_(True)
    w = weight.detach().clone().requires_grad_(True)
    out = rms_norm(x, w, eps)
    out.backward(dy)
    return x.grad, w.grad


def rotate_half(x: torch.Tensor) -> torch.Tensor:
    x1 = x[..., : x.shape[-1] // 2]
    x2 = x[..., x.shape[-1] // 2 :]
    return torch.cat((-x2, x1), dim=-1)


def apply_rotary_pos_emb(q, k, cos, sin, unsqueeze_dim: int = 1):
    """q,k [B,h,S,D]; cos,sin [B,S,D]. Ref: patched modeling :94-111."""
    cos = cos.unsqueeze(unsqueeze_dim)
    sin = sin.unsqueeze(unsqueeze_dim)
    q_embed = (q * cos) + (rotate_half(q) * sin)
    k_embed = (k * cos) + (rotate_half(k) * sin)
    return q_embed, k_embed


def silu_mul(gate: torch.Tensor, up: torch.Tensor) -> torch.Tensor:
    """silu(gate) * up — the LigerSiLUMulFunction slot's math."""
    return torch.nn.functional.silu(gate) * up


def rope_cos_sin(head_dim: int, seq_len: int, theta: float, dtype=torch.float32):
    """Standard RoPE table: inv_freq over even dims, angles duplicated to D.

    Ref: HF Qwen3MoeRotaryEmbedding default rope init (used by the patched
    modeling via ROPE_INIT_FUNCTIONS["default"]).
    """
    inv_freq = 1.0 / (theta ** (torch.arange(0, head_dim, 2, dtype=torch.float32) / head_dim))
    t = torch.arange(seq_len, dtype=torch.float32)
    freqs = torch.outer(t, inv_freq)
    emb = torch.cat((freqs, freqs), dim=-1)
    return emb.cos().to(dtype), emb.sin().to(dtype)
