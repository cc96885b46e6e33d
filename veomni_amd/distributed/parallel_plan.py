"""ParallelPlan: EP (extra-parallel) parameter slicing.

API parity target: /root/reference/veomni/distributed/parallel_plan.py:50-218
(`ParallelPlan(extra_parallel_plan={"ep": {fqn_pattern: Shard(0)}})`,
`.apply(model)`), and per-model plans, e.g.
models/transformers/qwen3_moe/parallel_plan.py:6-16 (gate_up_proj/down_proj
-> Shard(0) over the ep mesh).

Round-1 mechanics: plain dim-0 narrow of the full parameter to the local
expert slice (every rank builds the full seeded init, then keeps its slice);
sliced params are tagged `_ep_param = True` so the EP-aware grad-norm and
the FSDP2 wrap can route them to the (ep_fsdp, ep) mesh.
"""

from __future__ import annotations

import fnmatch
from dataclasses import dataclass, field
from typing import Dict

import torch
import torch.distributed as dist
import torch.nn as nn

from .parallel_state import get_parallel_state


@dataclass
class ParallelPlan:
    # {"ep": {fqn_glob_pattern: shard_dim}} — dim is always 0 on the §8 path.
    extra_parallel_plan: Dict[str, Dict[str, int]] = field(default_factory=dict)

    def apply(self, model: nn.Module) -> None:
        ps = get_parallel_state()
        if not ps.ep_enabled:
            return
        plan = self.extra_parallel_plan.get("ep", {})
        if not plan:
            return
        ep_rank = ps.ep_rank
        ep_size = ps.ep_size
        model._ep_fqns = set()  # read by the checkpointer (EP-rank-keyed save)
        named = dict(model.named_parameters())
        for fqn, param in named.items():
            for pattern, dim in plan.items():
                if not fnmatch.fnmatch(fqn, pattern):
                    continue
                assert dim == 0, "only Shard(0) EP plans on the hot path"
                full = param.data
                assert full.shape[0] % ep_size == 0, (fqn, full.shape, ep_size)
                local_e = full.shape[0] // ep_size
                local = full.narrow(0, ep_rank * local_e, local_e).clone()
                new_param = nn.Parameter(local, requires_grad=param.requires_grad)
                new_param._ep_param = True
                _set_param_by_fqn(model, fqn, new_param)
                model._ep_fqns.add(fqn)
                break


def _set_param_by_fqn(model: nn.Module, fqn: str, param: nn.Parameter) -> None:
    parts = fqn.split(".")
    mod = model
    for p in parts[:-1]:
        mod = getattr(mod, p)
    setattr(mod, parts[-1], param)
    mod._has_ep_params = True
