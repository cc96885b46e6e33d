"""CPU oracle for the MoE dispatch / grouped-GEMM path.

Restates (does not copy) the algorithms of:
  /root/reference/veomni/ops/kernels/moe/_kernels/kernel/moe.py  (histogram,
      scatter, gather: lines 29-333)
  /root/reference/veomni/ops/kernels/moe/_scatter.py:40-79       (scatter index)
  /root/reference/veomni/ops/kernels/moe/_kernels/kernel/group_gemm.py:66-397
      (grouped GEMM same-NK / same-MN semantics)
  /root/reference/veomni/ops/kernels/moe/group_gemm.py:320-520   (fused MoE
      forward math order: routing weights applied BEFORE fc2)
  /root/reference/veomni/models/transformers/qwen3_moe/generated/
      patched_modeling_qwen3_moe_gpu.py:254-330 (eager experts loop — weights
      applied AFTER down_proj — and router)

Test infrastructure only — see oracle/__init__.py.
"""

from __future__ import annotations

import torch


# ---------------------------------------------------------------- int paths
def expert_histogram(expert_index: torch.Tensor, num_experts: int) -> torch.Tensor:
    """Bin counts of expert ids. Bit-exact contract.

    Ref: kernel/moe.py:29-82 (atomic histogram -> int32 [E]).
    """
    flat = expert_index.flatten().to(torch.int64)
    return torch.bincount(flat, minlength=num_experts).to(torch.int32)[:num_experts]


def compute_expert_scatter_index(expert_index: torch.Tensor):
    """(sorted_order int64 [N], scatter_index int32 same shape as input).

    Stable sort by expert id; scatter_index = inverse permutation.
    Ref: _scatter.py:40-79. Bit-exact contract (stability load-bearing:
    same-expert rows must keep original (token, slot) order).
    """
    flat = expert_index.flatten()
    sorted_order = flat.argsort(stable=True)
    inv = torch.empty_like(sorted_order)
    inv[sorted_order] = torch.arange(sorted_order.numel(), dtype=sorted_order.dtype)
    return sorted_order, inv.to(torch.int32).view(expert_index.shape)


def moe_scatter(x: torch.Tensor, index: torch.Tensor) -> torch.Tensor:
    """out[index[t,k]] = x[t]  (row broadcast to topk slots).

    Ref: kernel/moe.py:253-333. x [M,N] -> out [M*topk, N], same dtype.
    """
    M, N = x.shape
    topk = index.shape[1]
    out = torch.empty(M * topk, N, dtype=x.dtype, device=x.device)
    out[index.flatten().to(torch.int64)] = x.repeat_interleave(topk, dim=0)
    return out


def moe_gather(x: torch.Tensor, index: torch.Tensor) -> torch.Tensor:
    """out[t] = sum_k x[index[t,k]] with fp32 accumulation, cast to x.dtype.

    Ref: kernel/moe.py:87-159 (y accumulated as float32).
    """
    gathered = x[index.flatten().to(torch.int64)].float()
    M, topk = index.shape
    return gathered.view(M, topk, -1).sum(dim=1).to(x.dtype)


def moe_add_gather(x: torch.Tensor, y: torch.Tensor, index: torch.Tensor) -> torch.Tensor:
    """out[t] = sum_k (x+y)[index[t,k]], fp32 accumulation.

    Ref: kernel/moe.py:164-248.
    """
    idx = index.flatten().to(torch.int64)
    s = (x[idx].float() + y[idx].float())
    M, topk = index.shape
    return s.view(M, topk, -1).sum(dim=1).to(x.dtype)


# ------------------------------------------------------------- grouped GEMM
def group_gemm_same_nk(a, b, cumsum_M, transpose_b, activation=None, c=None):
    """Per-group GEMM with shared N,K; rows of `a`/`c` partitioned by cumsum_M.

    Ref: kernel/group_gemm.py:66-234. fp32 accumulation, output in a.dtype
    unless `c` is given (then accumulate into c). transpose_b=True:
    C_g = A_g @ B_g^T with B [G,N,K]; else C_g = A_g @ B_g with B [G,K,N].
    activation: None | "silu".
    """
    G = b.shape[0]
    N = b.shape[1] if transpose_b else b.shape[2]
    rows = a.shape[0]
    out = torch.empty(rows, N, dtype=a.dtype) if c is None else c
    start = 0
    for g in range(G):
        end = int(cumsum_M[g])
        ag = a[start:end].float()
        bg = b[g].float()
        r = ag @ (bg.t() if transpose_b else bg)
        if activation == "silu":
            r = torch.nn.functional.silu(r)
        if c is None:
            out[start:end] = r.to(a.dtype)
        else:
            out[start:end] = (out[start:end].float() + r).to(out.dtype)
        start = end
    return out


def group_gemm_same_mn(a, b, cumsum_K):
    """Per-group wgrad: C[g] = A_g^T @ B_g, per-group row-count K.

    Ref: kernel/group_gemm.py:252-397 (transpose_a=True path; zero-fills
    groups with k==0). a [rows,M], b [rows,N] -> c [G,M,N] in a.dtype.
    """
    G = cumsum_K.shape[0]
    M, N = a.shape[1], b.shape[1]
    out = torch.zeros(G, M, N, dtype=a.dtype)
    start = 0
    for g in range(G):
        end = int(cumsum_K[g])
        if end > start:
            out[g] = (a[start:end].float().t() @ b[start:end].float()).to(a.dtype)
        start = end
    return out


# ------------------------------------------------------- fused-MoE semantics
def fused_moe_forward(num_experts, routing_weights, selected_experts,
                      hidden_states, fc1_1_2_weight, fc2_weight):
    """Merged-fc1 fused MoE forward math order (weights BEFORE fc2).

    Ref: ops/kernels/moe/group_gemm.py:327-405
    (MergedFc1TritonFusedMoeExpertFunction.forward). fp32 compute here;
    the HIP path runs bf16 with fp32 accumulation.
    """
    T, H = hidden_states.shape
    splits = expert_histogram(selected_experts, num_experts)
    _, scatter_index = compute_expert_scatter_index(selected_experts)
    scatter_output = moe_scatter(hidden_states, scatter_index)
    cumsum_t = torch.cumsum(splits, dim=0)

    fc1 = group_gemm_same_nk(scatter_output, fc1_1_2_weight, cumsum_t, transpose_b=True)
    gate, up = fc1.chunk(2, dim=-1)
    act = torch.nn.functional.silu(gate.float()) * up.float()

    w = routing_weights.reshape(-1, 1)
    scattered_w = torch.empty_like(w)
    scattered_w[scatter_index.flatten().to(torch.int64)] = w
    weighted = (act * scattered_w.float()).to(hidden_states.dtype)

    fc2 = group_gemm_same_nk(weighted, fc2_weight, cumsum_t, transpose_b=True)
    return moe_gather(fc2, scatter_index)


def eager_moe_forward(hidden_states, top_k_index, top_k_weights,
                      gate_up_proj, down_proj):
    """Reference EAGER experts loop (weights AFTER down_proj).

    Ref: patched_modeling_qwen3_moe_gpu.py:254-294 (Qwen3MoeExperts.forward,
    eager branch). This is the parity anchor the reference's own fused-vs-
    eager tests compare against.
    """
    num_experts = gate_up_proj.shape[0]
    final = torch.zeros_like(hidden_states)
    expert_mask = torch.nn.functional.one_hot(top_k_index, num_classes=num_experts).permute(2, 1, 0)
    for e in range(num_experts):
        top_k_pos, token_idx = torch.where(expert_mask[e])
        if token_idx.numel() == 0:
            continue
        cur = hidden_states[token_idx]
        gate_up = torch.nn.functional.linear(cur, gate_up_proj[e])
        gate, up = gate_up.chunk(2, dim=-1)
        h = torch.nn.functional.silu(gate) * up
        h = torch.nn.functional.linear(h, down_proj[e])
        h = h * top_k_weights[token_idx, top_k_pos, None]
        final.index_add_(0, token_idx, h.to(final.dtype))
    return final


def router(hidden_states, router_weight, top_k, norm_topk_prob):
    """Qwen3-MoE top-k router.

    Ref: patched_modeling_qwen3_moe_gpu.py:303-330 (Qwen3MoeTopKRouter):
    raw logits kept; softmax in fp32; optional top-k renorm; top values cast
    back to logits dtype.
    """
    hidden_states = hidden_states.reshape(-1, router_weight.shape[1])
    router_logits = torch.nn.functional.linear(hidden_states, router_weight)
    routing_weights = torch.nn.functional.softmax(router_logits, dtype=torch.float, dim=-1)
    top_value, top_index = torch.topk(routing_weights, top_k, dim=-1)
    if norm_topk_prob:
        top_value = top_value / top_value.sum(dim=-1, keepdim=True)
    return router_logits, top_value.to(router_logits.dtype), top_index
