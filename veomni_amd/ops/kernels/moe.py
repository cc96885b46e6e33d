"""HIP fused-MoE kernels: grouped-GEMM expert MLP, non-EP and EP variants.

Parity targets:
  - math order of the reference's merged-fc1 fused path (routing weights
    applied BEFORE fc2): ops/kernels/moe/group_gemm.py:320-520
    (MergedFc1TritonFusedMoeExpertFunction) and the EP merged variant
    distributed/moe/moe_layer.py:414-567 (EPMergedFc1GroupGemm);
  - module pointer + patch entry: ops/kernels/moe/__init__.py:27-118
    (`_fused_moe_forward`, `fused_moe_forward`, `apply_veomni_fused_moe_patch`);
  - OpSlot adapter contract: ops/kernels/moe/__init__.py:126-161;
  - scatter-index helper kept plain torch ON PURPOSE, like the reference
    (_scatter.py:30-33: "plain-torch helper on purpose").

Device work: vh_expert_histogram / vh_moe_scatter / vh_group_gemm_nk /
vh_moe_silu_mul_weighted / vh_group_gemm_mn / vh_moe_gather (C ABI).
"""

from __future__ import annotations

import torch

from ...distributed.moe import dispatch_to_ep_class
from ...distributed.parallel_state import get_parallel_state
from .. import hip_lib
from ..kernel_registry import KERNEL_REGISTRY, HardwareRequirement, KernelSpec

_fused_moe_forward = None


def compute_expert_scatter_index(expert_index: torch.Tensor):
    """One stable sort + O(N) inverse permutation (ref _scatter.py:40-79)."""
    flat = expert_index.flatten()
    sorted_order = flat.argsort(stable=True)
    inv = torch.empty_like(sorted_order)
    inv[sorted_order] = torch.arange(sorted_order.numel(), dtype=sorted_order.dtype,
                                     device=sorted_order.device)
    return sorted_order, inv.to(torch.int32).view(expert_index.shape)


class HipFusedMoeFunction(torch.autograd.Function):
    """Non-EP merged-fc1 fused MoE (parity: MergedFc1TritonFusedMoeExpertFunction)."""

    @staticmethod
    def forward(ctx, num_experts, gate_weights, expert_index, hidden_states,
                fc1_1_2_weight, fc2_weight):
        T, H = hidden_states.shape
        splits = hip_lib.expert_histogram(expert_index, num_experts)
        _, scatter_index = compute_expert_scatter_index(expert_index)
        scatter_output = hip_lib.moe_scatter(hidden_states, scatter_index)
        cumsum_t = torch.cumsum(splits, dim=0)

        fc1 = hip_lib.group_gemm_nk(scatter_output, fc1_1_2_weight, cumsum_t, trans_b=True)

        w = gate_weights.reshape(-1)
        scattered_w = torch.empty_like(w)
        scattered_w[scatter_index.flatten().to(torch.int64)] = w
        weighted = hip_lib.silu_mul_weighted(fc1, scattered_w)

        fc2 = hip_lib.group_gemm_nk(weighted, fc2_weight, cumsum_t, trans_b=True)
        out = hip_lib.moe_gather(fc2, scatter_index)

        ctx.num_experts = num_experts
        ctx.save_for_backward(gate_weights, fc1_1_2_weight, fc2_weight,
                              scatter_index, scatter_output, cumsum_t, fc1,
                              scattered_w, weighted)
        return out

    @staticmethod
    def backward(ctx, grad_output):
        (gate_weights, fc1_1_2_weight, fc2_weight, scatter_index,
         scatter_output, cumsum_t, fc1, scattered_w, weighted) = ctx.saved_tensors
        G = fc2_weight.shape[0]
        grad_output = grad_output.view(-1, grad_output.shape[-1]).contiguous()

        grad_fc2_out = hip_lib.moe_scatter(grad_output, scatter_index)
        # dgrad through fc2 — transpose the (small, persistent) expert weights
        # once so dgrad runs the fast trans_b kernel (HBM copy ~50 us vs
        # ~2x slower K-strided dgrad staging; profiles/r01_groupgemm_microbench)
        d_weighted = hip_lib.group_gemm_nk(
            grad_fc2_out, hip_lib.weight_transpose(fc2_weight), cumsum_t, trans_b=True)
        # wgrad fc2: [G, H, I]
        d_fc2_w = hip_lib.group_gemm_mn(grad_fc2_out, weighted, cumsum_t, G)

        # fused epilogue backward (silu recomputed from saved pre-activation)
        d_fc1, dw_rows = hip_lib.silu_mul_weighted_bwd(d_weighted, fc1, scattered_w)
        grad_gate = dw_rows[scatter_index.flatten().to(torch.int64)]
        grad_gate = grad_gate.reshape(gate_weights.shape).to(gate_weights.dtype)

        # dgrad + wgrad through merged fc1 (same weight-transpose trick)
        d_scatter = hip_lib.group_gemm_nk(
            d_fc1, hip_lib.weight_transpose(fc1_1_2_weight), cumsum_t, trans_b=True)
        d_fc1_w = hip_lib.group_gemm_mn(d_fc1, scatter_output, cumsum_t, G)

        grad_hidden = hip_lib.moe_gather(d_scatter, scatter_index)
        return None, grad_gate, None, grad_hidden, d_fc1_w, d_fc2_w


class EPMergedFc1HipGroupGemm(torch.autograd.Function):
    """EP expert-MLP (parity: EPMergedFc1GroupGemm, moe_layer.py:414-567);
    routing weights applied later by `unpermute` in tokens_post_all2all.
    Kept for the plain dispatch path; the DISPATCHED EP path is
    EPMergedFc1HipGroupGemmA2A below (owns the dispatch a2a for the
    wgrad/a2a backward overlap)."""

    @staticmethod
    def forward(ctx, permute_tokens, cumsum, fc1_1_2_weight, fc2_weight):
        fc1 = hip_lib.group_gemm_nk(permute_tokens, fc1_1_2_weight, cumsum, trans_b=True)
        act = hip_lib.silu_mul_weighted(fc1, None)
        fc2 = hip_lib.group_gemm_nk(act, fc2_weight, cumsum, trans_b=True)
        ctx.save_for_backward(permute_tokens, cumsum, fc1_1_2_weight, fc2_weight, fc1, act)
        return fc2

    @staticmethod
    def backward(ctx, grad_output):
        permute_tokens, cumsum, fc1_1_2_weight, fc2_weight, fc1, act = ctx.saved_tensors
        G = fc2_weight.shape[0]
        grad_output = grad_output.contiguous()
        d_act = hip_lib.group_gemm_nk(
            grad_output, hip_lib.weight_transpose(fc2_weight), cumsum, trans_b=True)
        d_fc2_w = hip_lib.group_gemm_mn(grad_output, act, cumsum, G)
        d_fc1, _ = hip_lib.silu_mul_weighted_bwd(d_act, fc1, None)
        d_tokens = hip_lib.group_gemm_nk(
            d_fc1, hip_lib.weight_transpose(fc1_1_2_weight), cumsum, trans_b=True)
        d_fc1_w = hip_lib.group_gemm_mn(d_fc1, permute_tokens, cumsum, G)
        return d_tokens, None, d_fc1_w, d_fc2_w


# ---- overlapped EP class: dispatch a2a owned by the autograd node so its
# backward launches the return exchange before the wgrad GEMMs ------------

def _ep_mlp_fwd(tokens, cumsum, fc1_1_2_weight, fc2_weight):
    fc1 = hip_lib.group_gemm_nk(tokens, fc1_1_2_weight, cumsum, trans_b=True)
    act = hip_lib.silu_mul_weighted(fc1, None)
    fc2 = hip_lib.group_gemm_nk(act, fc2_weight, cumsum, trans_b=True)
    return fc2, (tokens, fc1_1_2_weight, fc2_weight, fc1, act)


def _ep_mlp_bwd_dgrad(dY, cumsum, saved):
    _, w1, w2, fc1, _ = saved
    d_act = hip_lib.group_gemm_nk(dY, hip_lib.weight_transpose(w2), cumsum,
                                  trans_b=True)
    d_fc1, _ = hip_lib.silu_mul_weighted_bwd(d_act, fc1, None)
    d_tokens = hip_lib.group_gemm_nk(d_fc1, hip_lib.weight_transpose(w1),
                                     cumsum, trans_b=True)
    return d_tokens, (dY, d_fc1)


def _ep_mlp_bwd_wgrad(cumsum, saved, stash):
    tokens, _, w2, _, act = saved
    dY, d_fc1 = stash
    G = w2.shape[0]
    d_fc2_w = hip_lib.group_gemm_mn(dY, act, cumsum, G)
    d_fc1_w = hip_lib.group_gemm_mn(d_fc1, tokens, cumsum, G)
    return (d_fc1_w, d_fc2_w)


def _make_ep_a2a():
    from ...distributed.moe import make_ep_a2a_class

    return make_ep_a2a_class(_ep_mlp_fwd, _ep_mlp_bwd_dgrad, _ep_mlp_bwd_wgrad)


EPMergedFc1HipGroupGemmA2A = _make_ep_a2a()


def hip_fused_moe_forward(num_experts, routing_weights, selected_experts,
                          hidden_states, fc1_1_weight, fc1_2_weight, fc2_weight,
                          fc1_1_2_weight=None, swiglu_limit=None):
    """Signature parity: group_gemm_fused_moe_forward (group_gemm.py:523-603).

    Round-1 scope: merged fc1 only (the layout our models and the reference's
    v5 experts use); swiglu_limit (gpt-oss clamp) is off-path."""
    if swiglu_limit is not None:
        raise NotImplementedError("swiglu_limit is off the §8 hot path")
    if fc1_1_2_weight is None:
        if fc1_1_weight is None or fc1_2_weight is None:
            raise ValueError("need merged fc1_1_2_weight or both split weights")
        fc1_1_2_weight = torch.cat([fc1_1_weight, fc1_2_weight], dim=1).contiguous()
    hs = hidden_states.reshape(-1, hidden_states.shape[-1])
    if get_parallel_state().ep_enabled:
        from ...distributed.moe import dispatch_to_ep_a2a_class

        out = dispatch_to_ep_a2a_class(
            EPMergedFc1HipGroupGemmA2A, num_experts, routing_weights,
            selected_experts, hs, fc1_1_2_weight, fc2_weight,
        )
    else:
        out = HipFusedMoeFunction.apply(
            num_experts, routing_weights, selected_experts, hs,
            fc1_1_2_weight, fc2_weight,
        )
    return out.reshape(hidden_states.shape)


def fused_moe_forward(num_experts, routing_weights, selected_experts,
                      hidden_states, fc1_1_weight, fc1_2_weight, fc2_weight,
                      fc1_1_2_weight=None, swiglu_limit=None):
    """Module-pointer shim (parity: ops/kernels/moe/__init__.py:30-61)."""
    if _fused_moe_forward is None:
        raise NotImplementedError("No fused MoE kernel bound. Call apply_veomni_fused_moe_patch('hip').")
    assert routing_weights.dtype in (torch.bfloat16, torch.float16)
    assert hidden_states.dtype in (torch.bfloat16, torch.float16)
    return _fused_moe_forward(num_experts, routing_weights, selected_experts,
                              hidden_states, fc1_1_weight, fc1_2_weight,
                              fc2_weight, fc1_1_2_weight, swiglu_limit=swiglu_limit)


def apply_veomni_fused_moe_patch(fused_moe_kernel: str = "hip") -> None:
    """Bind the global `_fused_moe_forward` pointer (parity:
    ops/kernels/moe/__init__.py:64-118). Only "hip" exists on MI355X."""
    global _fused_moe_forward
    if fused_moe_kernel != "hip":
        raise ValueError(f"Invalid fused_moe_kernel: {fused_moe_kernel!r}; MI355X build provides 'hip'.")
    _fused_moe_forward = hip_fused_moe_forward


def _make_moe_experts_adapter(raw_forward):
    """OpSlot adapter (parity: ops/kernels/moe/__init__.py:126-161): pull
    num_experts / gate_up_proj / down_proj off the experts module."""

    def adapter(self, hidden_states, top_k_index, top_k_weights):
        return raw_forward(
            num_experts=self.num_experts,
            routing_weights=top_k_weights.to(hidden_states.dtype),
            selected_experts=top_k_index,
            hidden_states=hidden_states,
            fc1_1_weight=None,
            fc1_2_weight=None,
            fc2_weight=self.down_proj,
            fc1_1_2_weight=self.gate_up_proj,
            swiglu_limit=getattr(self, "limit", None),
        )

    return adapter


def _hip_moe_experts_factory():
    apply_veomni_fused_moe_patch("hip")
    return _make_moe_experts_adapter(fused_moe_forward)


KERNEL_REGISTRY.register(
    KernelSpec(
        name="hip", op_name="moe_experts", variant="standard",
        factory=_hip_moe_experts_factory,
        hardware=HardwareRequirement(device_type="gpu"),
        description="gfx950 grouped-GEMM fused MoE (vh_group_gemm_*)",
    )
)
