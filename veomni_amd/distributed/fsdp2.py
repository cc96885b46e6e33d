"""FSDP2 sharded-training wrap + EP-aware grad-norm clip.

API parity target: /root/reference/veomni/distributed/torch_parallelize.py
(`build_parallelize_model` :590, `parallelize_model_fsdp2` :133-587) and
fsdp2/clip_grad_norm.py:17-317.

Semantics mirrored:
  - per-decoder-layer `fully_shard` bottom-up over the `dp_shard_sp` mesh
    with MixedPrecisionPolicy(param bf16, reduce fp32 by default)
    (ref :296-309, :412-457);
  - expert modules (EP-sliced params) wrapped separately over the `ep_fsdp`
    mesh with Shard(1) placement and gradient divide factor = world_size
    (ref :345-384, :431-438);
  - root wrapped without explicit reshard_after_forward so FSDP2's root
    auto-no-reshard keeps lm_head/embeddings unsharded between fwd and bwd
    (ref :459-469);
  - manual forward/backward prefetch lists across layers (ref :471-490);
  - `model.clip_grad_norm_` bound to the EP-aware clip (ref :583-585).

The AG/RS transport is torch FSDP2's c10d path = RCCL over xGMI; bf16
all-gathers, fp32 reduce-scatter (reference default MixedPrecisionConfig).
"""

from __future__ import annotations

import functools
from typing import List, Optional

import torch
import torch.distributed as dist
import torch.nn as nn
from torch.distributed.fsdp import FSDPModule, MixedPrecisionPolicy, fully_shard
from torch.distributed.tensor import DTensor, Shard

from .parallel_state import get_parallel_state


def _module_has_ep_params(mod: nn.Module) -> bool:
    return any(getattr(p, "_ep_param", False) for p in mod.parameters(recurse=True))


def build_parallelize_model(
    model: nn.Module,
    enable_reshard_after_forward: bool = True,
    param_dtype: torch.dtype = torch.bfloat16,
    reduce_dtype: torch.dtype = torch.float32,
    enable_forward_prefetch: bool = True,
    force_wrap: bool = False,
) -> nn.Module:
    """Apply EP slicing + per-layer FSDP2 wrap. Entry point parity:
    ref torch_parallelize.py:590 (`build_parallelize_model`)."""
    ps = get_parallel_state()

    # EP param slicing via the model's plan (ref :193-231)
    plan = model.get_parallel_plan() if hasattr(model, "get_parallel_plan") else None
    if plan is not None and ps.ep_enabled:
        plan.apply(model)

    if ps.fsdp_size == 1 and not ps.ep_enabled and not force_wrap:
        # degenerate world: fully_shard would only add allgather/copy churn
        # (no actual sharding); keep the plain module + the same clip entry.
        # force_wrap=True runs the full FSDP2 mechanics anyway (1-GPU RCCL
        # smoke of the wrap/prefetch/reduce machinery).
        model.clip_grad_norm_ = functools.partial(clip_grad_norm, model)
        return model

    mp_policy = MixedPrecisionPolicy(param_dtype=param_dtype, reduce_dtype=reduce_dtype)
    fsdp_kwargs = dict(
        mesh=ps.fsdp_mesh,
        reshard_after_forward=enable_reshard_after_forward,
        mp_policy=mp_policy,
    )
    ep_fsdp_kwargs = None
    if ps.ep_enabled:
        ep_fsdp_kwargs = dict(
            mesh=ps.ep_fsdp_mesh,
            reshard_after_forward=enable_reshard_after_forward,
            mp_policy=mp_policy,
            # EP expert weights FSDP-shard on dim 1 (hidden), EP already
            # owns dim 0 (ref :357,376).
            shard_placement_fn=lambda p: Shard(1),
        )

    layers = model.get_decoder_layers() if hasattr(model, "get_decoder_layers") else list(model.children())
    blocks: List[nn.Module] = []
    for layer in layers:
        layer._fsdp_modules = []
        if ps.ep_enabled:
            for sub in layer.modules():
                if getattr(sub, "_has_ep_params", False) and not isinstance(sub, FSDPModule):
                    fully_shard(sub, **ep_fsdp_kwargs)
                    # average EP grads so they match dense-grad scaling
                    # (ref :431-438): divide factor = world size.
                    sub.set_gradient_divide_factor(float(ps.ep_gradient_divide_factor))
                    layer._fsdp_modules.append(sub)
        if not isinstance(layer, FSDPModule):
            fully_shard(layer, **fsdp_kwargs)
            layer._fsdp_modules.append(layer)
        blocks.append(layer)

    # root: no explicit reshard_after_forward (ref :459-469)
    root_kwargs = {k: v for k, v in fsdp_kwargs.items() if k != "reshard_after_forward"}
    fully_shard(model, **root_kwargs)

    # manual prefetch (ref :471-490)
    if enable_forward_prefetch and ps.ep_enabled:
        nxt = blocks[1:] + [None]
        for cur, nx in zip(blocks, nxt):
            if nx is not None:
                cur.set_modules_to_forward_prefetch(list(reversed(nx._fsdp_modules)))
        rev = list(reversed(blocks))
        prev = rev[1:] + [None]
        for cur, pv in zip(rev, prev):
            if pv is not None:
                cur.set_modules_to_backward_prefetch(list(reversed(pv._fsdp_modules)))

    model.clip_grad_norm_ = functools.partial(clip_grad_norm, model)
    return model


# ---------------------------------------------------------------- grad norm
def _local_pth_sum(params, norm_type: float) -> torch.Tensor:
    """Sum of |g|^p over the LOCAL shards, fp32 accumulation without
    materializing fp32 grads (ref fsdp2/clip_grad_norm.py:246-270).
    Uses foreach-norm (one fused kernel sweep) instead of a per-param
    abs/pow/sum kernel triple."""
    grads = []
    for p in params:
        g = p.grad
        if g is None:
            continue
        if isinstance(g, DTensor):
            g = g.to_local()
        if g.numel() > 0:
            grads.append(g)
    if not grads:
        return torch.zeros(())
    norms = torch._foreach_norm(grads, norm_type)
    return torch.stack([n.float() for n in norms]).pow(norm_type).sum()


@torch.no_grad()
def clip_grad_norm(model: nn.Module, max_norm: float, norm_type: float = 2.0,
                   error_if_nonfinite: bool = False, foreach: Optional[bool] = None,
                   fused_optimizer: Optional[torch.optim.Optimizer] = None) -> torch.Tensor:
    """EP-aware global grad-norm clip (ref fsdp2/clip_grad_norm.py:86-226):
    dense p-th-power sums all-reduced over the fsdp shard group; EP sums over
    ep_fsdp then ep groups; one global clip coefficient."""
    ps = get_parallel_state()
    params = [p for p in model.parameters() if p.grad is not None]
    dense = [p for p in params if not getattr(p, "_ep_param", False)]
    ep = [p for p in params if getattr(p, "_ep_param", False)]

    total = _local_pth_sum(dense, norm_type)
    # sum each DISTINCT shard once: HSDP replicas hold identical grads, so
    # the reduce runs over the shard_sp group only
    if ps.fsdp_shard_size > 1:
        dist.all_reduce(total, group=ps.fsdp_shard_group)
    if ep:
        ep_sum = _local_pth_sum(ep, norm_type)
        if ps.ep_fsdp_size > 1:
            # distinct shards only: under HSDP the replicas hold identical
            # grads, so the reduce runs over the 1-D ep_fsdp group
            dist.all_reduce(ep_sum, group=ps.ep_shard_group)
        dist.all_reduce(ep_sum, group=ps.ep_group)
        total = total + ep_sum

    total_norm = total.pow(1.0 / norm_type)
    if error_if_nonfinite and (torch.isnan(total_norm) or torch.isinf(total_norm)):
        raise RuntimeError("non-finite grad norm")

    if fused_optimizer is not None and norm_type == 2.0:
        # fold the clip into the fused optimizer: torch's fused AdamW divides
        # grads by `optimizer.grad_scale` in-register (AMP-unscale plumbing,
        # optim/adam.py:267), so the explicit _foreach_mul_ over every grad
        # (~120 GB of traffic on a 30B model) disappears. Exact same math:
        # grad * coef == grad / max(1, (norm + eps) / max_norm). No host sync.
        fused_optimizer.grad_scale = torch.clamp(
            (total_norm + 1e-6) / max_norm, min=1.0).to(torch.float32)
        return total_norm
    clip_coef = max_norm / (total_norm + 1e-6)
    clip_coef = torch.clamp(clip_coef, max=1.0)
    grads = []
    for p in params:
        g = p.grad
        if isinstance(g, DTensor):
            g = g.to_local()
        if g.numel() > 0:
            grads.append(g)
    if grads:
        torch._foreach_mul_(grads, clip_coef.to(grads[0].device))
    return total_norm
