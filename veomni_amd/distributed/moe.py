"""Expert-parallel MoE dispatch/combine over RCCL all-to-all.

API parity target: /root/reference/veomni/distributed/moe/
  moe_layer.py:48-224 (`preprocess`, `token_pre_all2all`,
  `tokens_post_all2all`, `dispatch_to_ep_class`)
  comm.py:20-54 (`_AllToAll` autograd over dist.all_to_all_single)
  moe_utils.py:19-99 (`permute`, `unpermute`, `generate_weights_idx`,
  `sort_chunks_by_idxs`).

The expert-MLP autograd classes (grouped GEMMs on local experts) live in
veomni_amd.ops.kernels.moe — this module is pure dispatch plumbing, so the
gloo ws=2 CPU tests cover it with a torch ep_class stub while the GPU path
passes the HIP grouped-GEMM classes.

MI355X note: the a2a payload (~T·topk·H·2B per layer per rank) rides the
xGMI full mesh where all-to-all is link-parallel — the cheap collective.
"""

from __future__ import annotations

from typing import Any, Callable, Optional

import torch
import torch.distributed as dist

from .parallel_state import get_parallel_state


class _AllToAll(torch.autograd.Function):
    """Variable-split a2a; backward = a2a with the splits swapped
    (ref comm.py:20-54)."""

    @staticmethod
    def forward(ctx, group, x, output_split_sizes, input_split_sizes):
        ctx.group = group
        ctx.output_split_sizes = output_split_sizes
        ctx.input_split_sizes = input_split_sizes
        if dist.get_world_size(group=group) == 1:
            return x
        x = x.contiguous()
        if output_split_sizes is None:
            out = torch.empty_like(x)
        else:
            out = torch.empty((sum(output_split_sizes), x.size(1)), dtype=x.dtype, device=x.device)
        dist.all_to_all_single(
            out, x, output_split_sizes=output_split_sizes,
            input_split_sizes=input_split_sizes, group=group,
        )
        return out

    @staticmethod
    def backward(ctx, *grad_output):
        return (None, _AllToAll.apply(ctx.group, *grad_output, ctx.input_split_sizes, ctx.output_split_sizes), None, None)


def all_to_all(group, x, output_split_sizes=None, input_split_sizes=None):
    return _AllToAll.apply(group, x, output_split_sizes, input_split_sizes)


# NOTE on the four helpers below: their OUTPUT is a bit-exact
# index-bookkeeping contract with the reference (moe_utils.py:19-99) — the
# expert-major traversal order and the fp32 accumulation order ARE the spec
# (checked by tests/test_dist_cpu.py parity against the reference's own
# moe_utils on random routings). The implementations here use different
# torch idioms (nonzero over masked_select, index_add_ over expanded
# scatter_add_) chosen for the ROCm kernels they lower to.


def permute(tokens: torch.Tensor, routing_map: torch.Tensor):
    """Expert-major stable permutation.

    routing_map: [num_experts, num_tokens] 0/1. Returns (permuted, mapping)
    where mapping[i] = source token row of permuted row i. Row order =
    expert-major, token-index-minor — exactly the reference's order
    (moe_utils.py:19-41), produced here by row-major nonzero().
    """
    # nonzero on [E, T] enumerates (e, t) in expert-major order; column 1 is
    # the source token row
    sorted_indices = routing_map.bool().nonzero(as_tuple=True)[1]
    return tokens.index_select(0, sorted_indices), sorted_indices


def unpermute(tokens, routing_weights, hidden_states_shape, permutation_mapping, routing_map):
    """Weighted fp32 unpermute (ref contract moe_utils.py:44-72): each
    permuted row is scaled by its routing weight and accumulated in fp32
    into its source token row."""
    tokens_weight = routing_weights.T.masked_select(routing_map.bool())
    out = torch.zeros(hidden_states_shape, device=tokens.device, dtype=torch.float32)
    # index_add_ lowers to one atomics pass on ROCm (the reference's
    # dim-expanded scatter_add_ builds an [N, H] index tensor first)
    out.index_add_(0, permutation_mapping, (tokens * tokens_weight.unsqueeze(-1)).float())
    return out.to(tokens.dtype)


def generate_weights_idx(routing_weights, selected_experts, num_experts):
    """[T, topk] weights -> dense [T, E] (ref contract moe_utils.py:75-92;
    duplicate expert picks accumulate)."""
    T = routing_weights.shape[0]
    out = torch.zeros((T, num_experts), dtype=routing_weights.dtype,
                      device=routing_weights.device)
    out.scatter_add_(1, selected_experts, routing_weights)
    return out


def sort_chunks_by_idxs(x, split_sizes, sorted_idxs):
    """Reorder row-chunks (ref contract moe_utils.py:95-99) through one
    gather: build the row permutation from chunk offsets instead of
    split+cat (one kernel instead of ~E small cats)."""
    sizes = torch.as_tensor(split_sizes, device="cpu", dtype=torch.long)
    offs = torch.zeros(sizes.numel() + 1, dtype=torch.long)
    torch.cumsum(sizes, 0, out=offs[1:])
    order = torch.cat([torch.arange(int(offs[i]), int(offs[i + 1]))
                       for i in (sorted_idxs.tolist() if torch.is_tensor(sorted_idxs) else sorted_idxs)])
    return x.index_select(0, order.to(x.device))


def preprocess(expert_mask: torch.Tensor, num_experts: int, ep_group) -> tuple:
    """Exchange per-expert counts; derive a2a splits (ref moe_layer.py:48-87)."""
    ep_size = ep_group.size()
    num_local_experts = num_experts // ep_size
    rank = dist.get_rank(ep_group)
    num_local_tokens_per_expert = expert_mask.sum(dim=(1, 2))

    input_splits = (
        num_local_tokens_per_expert.reshape(ep_size, num_local_experts).sum(dim=1).tolist()
    )
    num_global = torch.zeros(
        ep_size * num_local_tokens_per_expert.size(0),
        dtype=num_local_tokens_per_expert.dtype, device=num_local_tokens_per_expert.device,
    )
    dist.all_gather_into_tensor(num_global, num_local_tokens_per_expert.contiguous(), group=ep_group)
    num_global = num_global.view(ep_size, -1)

    start, end = rank * num_local_experts, (rank + 1) * num_local_experts
    num_global_per_local = num_global[:, start:end].contiguous()
    output_splits = num_global_per_local.sum(dim=1).tolist()
    num_global_sum_per_local = num_global_per_local.sum(dim=0).to("cpu", non_blocking=True)
    num_global_per_local = num_global_per_local.view(-1, num_local_experts).to("cpu", non_blocking=True)
    return input_splits, output_splits, num_global_per_local, num_global_sum_per_local


def token_pre_all2all(hidden_states, expert_mask, num_experts, input_splits,
                      output_splits, num_global_tokens_per_local_expert, ep_group):
    """Permute + dispatch a2a + expert-major chunk resort (ref moe_layer.py:90-117)."""
    hidden_dim = hidden_states.size(-1)
    hidden_states = hidden_states.reshape(-1, hidden_dim)
    org_shape = hidden_states.shape
    routing_map = expert_mask.sum(dim=1)

    local_permuted, local_mapping = permute(hidden_states, routing_map)
    global_permuted = all_to_all(ep_group, local_permuted, output_splits, input_splits)

    num_local_experts = num_experts // ep_group.size()
    permute_order = torch.arange(num_experts).reshape(-1, num_local_experts).T.ravel().tolist()
    global_permuted = sort_chunks_by_idxs(
        global_permuted, num_global_tokens_per_local_expert.ravel(), permute_order
    )
    return global_permuted, routing_map, local_mapping, org_shape


def tokens_post_all2all(expert_outputs, routing_weights, selected_experts, num_experts,
                        input_splits, output_splits, num_global_tokens_per_local_expert,
                        routing_map, local_input_permutation_mapping,
                        org_hidden_states_shape, ep_group):
    """Inverse resort + combine a2a + weighted unpermute (ref moe_layer.py:189-224)."""
    num_local_experts = num_experts // ep_group.size()
    unpermute_order = torch.arange(num_experts).reshape(num_local_experts, -1).T.ravel().tolist()
    expert_outputs = sort_chunks_by_idxs(
        expert_outputs, num_global_tokens_per_local_expert.T.ravel(), unpermute_order
    )
    out = all_to_all(ep_group, expert_outputs, input_splits, output_splits)
    weights_idx = generate_weights_idx(routing_weights, selected_experts, num_experts)
    return unpermute(out, weights_idx, org_hidden_states_shape,
                     local_input_permutation_mapping, routing_map)


def make_ep_a2a_class(mlp_fwd: Callable, mlp_bwd_dgrad: Callable,
                      mlp_bwd_wgrad: Callable):
    """Build an EP expert autograd class that owns the DISPATCH a2a + chunk
    resort (instead of leaving them as separate autograd nodes, the
    reference's structure — moe_layer.py:120-186 + comm.py:20-54).

    Why: in backward, the return all-to-all of d_tokens does not depend on
    the wgrad GEMMs, but as a separate autograd node it can only start
    after the expert backward RETURNS — i.e. after the wgrads. Owning the
    a2a lets backward launch it (async, on the c10d comm stream) right
    after the dgrad chain and run the wgrad GEMMs underneath it — on 8
    GPUs the per-layer exchange (~134 MB/dir/rank) rides xGMI while ~2/3
    of the backward FLOPs execute.

    mlp_fwd(tokens, cumsum, *weights) -> (out, saved_tuple)
    mlp_bwd_dgrad(dY, cumsum, saved) -> (d_tokens, stash)
    mlp_bwd_wgrad(cumsum, saved, stash) -> tuple of weight grads
    """

    class _EPA2AClass(torch.autograd.Function):
        @staticmethod
        def forward(ctx, local_permuted, cumsum, group, input_splits,
                    output_splits, sort_sizes, sort_order, *weights):
            ws = dist.get_world_size(group=group) if group is not None else 1
            if ws == 1:
                global_permuted = local_permuted
            else:
                lp = local_permuted.contiguous()
                out = torch.empty((sum(output_splits), lp.size(1)),
                                  dtype=lp.dtype, device=lp.device)
                dist.all_to_all_single(out, lp,
                                       output_split_sizes=list(output_splits),
                                       input_split_sizes=list(input_splits),
                                       group=group)
                global_permuted = out
            sorted_tokens = sort_chunks_by_idxs(global_permuted, sort_sizes,
                                                sort_order)
            y, saved = mlp_fwd(sorted_tokens, cumsum, *weights)
            ctx.save_for_backward(cumsum, *saved)
            ctx.group = group
            ctx.splits = (input_splits, output_splits)
            ctx.sort = (sort_sizes, sort_order)
            ctx.n_weights = len(weights)
            return y

        @staticmethod
        def backward(ctx, dY):
            cumsum, *saved = ctx.saved_tensors
            group = ctx.group
            input_splits, output_splits = ctx.splits
            sort_sizes, sort_order = ctx.sort
            ws = dist.get_world_size(group=group) if group is not None else 1

            d_sorted, stash = mlp_bwd_dgrad(dY.contiguous(), cumsum, saved)

            # inverse chunk resort: chunk i of d_sorted has size
            # sort_sizes[sort_order[i]]; send it back to slot sort_order[i]
            order = sort_order.tolist() if torch.is_tensor(sort_order) else list(sort_order)
            sizes = sort_sizes.tolist() if torch.is_tensor(sort_sizes) else list(sort_sizes)
            inv = [0] * len(order)
            for i, o in enumerate(order):
                inv[o] = i
            d_global = sort_chunks_by_idxs(d_sorted, [sizes[o] for o in order], inv)

            # launch the return a2a NOW (comm stream) ...
            work = None
            if ws == 1:
                d_local = d_global
            else:
                d_global = d_global.contiguous()
                d_local = torch.empty((sum(input_splits), d_global.size(1)),
                                      dtype=d_global.dtype, device=d_global.device)
                work = dist.all_to_all_single(
                    d_local, d_global, output_split_sizes=list(input_splits),
                    input_split_sizes=list(output_splits), group=group,
                    async_op=True)
            # ... and run the wgrad GEMMs underneath it
            dws = mlp_bwd_wgrad(cumsum, saved, stash)
            if work is not None:
                work.wait()
            return (d_local, None, None, None, None, None, None, *dws)

    return _EPA2AClass


def dispatch_to_ep_a2a_class(ep_a2a_class, num_experts: int, routing_weights,
                             selected_experts, hidden_states, *weights):
    """dispatch_to_ep_class variant for make_ep_a2a_class classes: the
    dispatch a2a + resort live inside the autograd class (overlapped
    backward); the combine side is unchanged (tokens_post_all2all)."""
    ep_state = get_parallel_state()
    ep_group = ep_state.ep_group
    expert_mask = torch.nn.functional.one_hot(
        selected_experts, num_classes=num_experts
    ).permute(2, 1, 0)
    input_splits, output_splits, num_global_per_local, num_global_sum_per_local = preprocess(
        expert_mask, num_experts, ep_group
    )
    hidden_dim = hidden_states.size(-1)
    hs = hidden_states.reshape(-1, hidden_dim)
    routing_map = expert_mask.sum(dim=1)
    local_permuted, local_mapping = permute(hs, routing_map)
    cumsum = torch.cumsum(num_global_sum_per_local, dim=0).to(hs.device)
    num_local_experts = num_experts // ep_group.size()
    sort_order = torch.arange(num_experts).reshape(-1, num_local_experts).T.ravel().tolist()
    final = ep_a2a_class.apply(
        local_permuted, cumsum, ep_group, tuple(input_splits),
        tuple(output_splits), num_global_per_local.ravel(), sort_order, *weights)
    return tokens_post_all2all(
        final, routing_weights, selected_experts, num_experts,
        input_splits, output_splits, num_global_per_local, routing_map,
        local_mapping, hs.shape, ep_group,
    )


def dispatch_to_ep_class(ep_class: Callable, num_experts: int, routing_weights,
                         selected_experts, hidden_states, *ep_class_args: Any):
    """Shared EP plumbing (ref moe_layer.py:120-186): preprocess ->
    token_pre_all2all -> ep_class.apply(permute_tokens, cumsum, *args) ->
    tokens_post_all2all."""
    ep_state = get_parallel_state()
    expert_mask = torch.nn.functional.one_hot(
        selected_experts, num_classes=num_experts
    ).permute(2, 1, 0)
    input_splits, output_splits, num_global_per_local, num_global_sum_per_local = preprocess(
        expert_mask, num_experts, ep_state.ep_group
    )
    permute_tokens, routing_map, local_mapping, org_shape = token_pre_all2all(
        hidden_states, expert_mask, num_experts, input_splits, output_splits,
        num_global_per_local, ep_state.ep_group,
    )
    cumsum = torch.cumsum(num_global_sum_per_local, dim=0).to(permute_tokens.device)
    final_permute_tokens = ep_class.apply(permute_tokens, cumsum, *ep_class_args)
    return tokens_post_all2all(
        final_permute_tokens, routing_weights, selected_experts, num_experts,
        input_splits, output_splits, num_global_per_local, routing_map,
        local_mapping, org_shape, ep_state.ep_group,
    )
