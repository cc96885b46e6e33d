"""Global per-token loss normalization for token-imbalanced DP training.

Parity target: /root/reference/veomni/utils/loss_utils.py:54-96
(`mean_global_loss`): the per-rank mean loss is re-weighted so that, after
FSDP's gradient averaging (divide by world size), the effective objective is
the GLOBAL per-token mean instead of a mean of per-rank means:

    loss' = loss * local_tokens / all_reduce_sum(step_tokens) * fsdp_size
    (under SP, local_tokens is first summed over the sp group and the result
     is divided by sp_size — ref :75-76, :89-90)

Round-1 scope keeps the single-loss (text) form; the reference's multi-loss
dict (VLM image/decoder token classes) is a keyed loop over the same formula.
"""

from __future__ import annotations

from typing import Optional, Union

import torch
import torch.distributed as dist

from .parallel_state import get_parallel_state


def mean_global_loss(loss: torch.Tensor,
                     local_valid_tokens: Union[int, torch.Tensor],
                     global_step_tokens: Optional[float] = None) -> torch.Tensor:
    """Re-weight a per-rank mean `loss` to the global per-token mean.

    local_valid_tokens: valid (non-ignored) tokens in this rank's micro batch.
    global_step_tokens: pre-reduced denominator (ref reduce_global_loss_token,
    :98-100) — pass it when accumulating several micro batches; all-reduced
    here when None.
    """
    ps = get_parallel_state()
    nv = torch.as_tensor(local_valid_tokens, dtype=torch.float32,
                         device=loss.device)
    if global_step_tokens is None:
        # Denominator = world-sum of the PER-RANK counts (ref :77-78 reduces
        # micro_batches_token_len, never the sp-reduced numerator): under SP
        # each rank holds a seq slice, so the world sum already counts every
        # token exactly once.
        total = nv.clone()
        if dist.is_initialized() and dist.get_world_size() > 1:
            dist.all_reduce(total)
    else:
        total = torch.as_tensor(global_step_tokens, dtype=torch.float32,
                                device=loss.device)
    if ps.sp_enabled:
        # Numerator only (ref :74-75): full-sequence tokens of this sp group.
        dist.all_reduce(nv, group=ps.sp_group)
    # fsdp divides gradients by its size; multiply back (ref :85)
    loss = loss * nv / total * float(ps.fsdp_size)
    if ps.sp_enabled:
        loss = loss / ps.sp_size
    return loss
