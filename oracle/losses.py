"""CPU oracle for the loss-side hot-path ops.

Restates:
  /root/reference/veomni/ops/kernels/cross_entropy/eager.py:22-36 and
      transformers.loss.loss_utils.fixed_cross_entropy semantics
      (sum-reduction / num_items, ignore_index=-100)
  /root/reference/veomni/ops/kernels/cross_entropy/chunk_loss.py:89-144
      (chunked fused-linear CE: causal shift when SP disabled, per-chunk
      F.linear -> fp32 CE, num_items = valid labels)
  /root/reference/veomni/ops/kernels/load_balancing_loss/eager.py:28-114
  /root/reference/veomni/distributed/sequence_parallel/loss.py:24-68
      (token-weighted SP loss reduce with zero-valid-token guard)
  /root/reference/veomni/data/data_collator.py:317-427 (SP collator: shift
      labels BEFORE slicing, pad to sp multiple, slice per rank)

Test infrastructure only — see oracle/__init__.py.
"""

from __future__ import annotations

import torch


IGNORE_INDEX = -100


def fixed_cross_entropy(logits_f32: torch.Tensor, labels: torch.Tensor,
                        num_items_in_batch=None, ignore_index: int = IGNORE_INDEX):
    """sum-NLL / num_items (or mean over valid when num_items is None).

    Ref: transformers fixed_cross_entropy as called by eager.py:36.
    """
    reduction = "sum" if num_items_in_batch is not None else "mean"
    loss = torch.nn.functional.cross_entropy(
        logits_f32, labels, ignore_index=ignore_index, reduction=reduction
    )
    if reduction == "sum":
        num = num_items_in_batch
        if torch.is_tensor(num):
            num = num.to(loss.device)
        loss = loss / num
    return loss


def causal_lm_loss(hidden_states: torch.Tensor, lm_head_weight: torch.Tensor,
                   labels: torch.Tensor, shift: bool = True,
                   ignore_index: int = IGNORE_INDEX):
    """Full fused-linear causal CE: logits = h @ W^T in input dtype, upcast
    fp32, causal shift, sum/num_valid.

    Ref: chunk_loss.py:100-137 (shift when SP disabled; num_items = count of
    valid shifted labels; per-chunk math identical to unchunked in exact
    arithmetic — the oracle computes it unchunked in fp32).
    """
    if shift:
        labels = labels[..., 1:].contiguous()
        hidden_states = hidden_states[..., :-1, :].contiguous()
    h = hidden_states.reshape(-1, hidden_states.shape[-1])
    logits = torch.nn.functional.linear(h, lm_head_weight).float()
    flat_labels = labels.reshape(-1)
    num_items = (flat_labels != ignore_index).sum()
    return fixed_cross_entropy(logits, flat_labels, num_items, ignore_index)


def load_balancing_loss(gate_logits_tuple, num_experts: int, top_k: int,
                        attention_mask=None):
    """Switch-transformer aux loss E * sum_e f_e * P_e.

    Ref: load_balancing_loss/eager.py:28-114 (scatter_add formulation;
    final form dot(count, prob_sum) * E / total^2).
    """
    if gate_logits_tuple is None or not isinstance(gate_logits_tuple, tuple):
        return 0
    expert_count = torch.zeros(num_experts, dtype=torch.float32)
    router_prob_sum = torch.zeros(num_experts, dtype=torch.float32)
    total = torch.tensor(0.0)
    for layer_logits in gate_logits_tuple:
        probs = torch.softmax(layer_logits.float(), dim=-1)
        _, selected = torch.topk(probs, top_k, dim=-1)
        if attention_mask is not None:
            mask = attention_mask.to(torch.float32).reshape(-1)
            router_prob_sum += (probs * mask.unsqueeze(-1)).sum(dim=0)
            w = mask.unsqueeze(-1).expand_as(selected).reshape(-1)
            expert_count.scatter_add_(0, selected.reshape(-1), w)
            total = total + mask.sum()
        else:
            router_prob_sum += probs.sum(dim=0)
            expert_count.scatter_add_(0, selected.reshape(-1),
                                      torch.ones(selected.numel(), dtype=torch.float32))
            total = total + layer_logits.shape[0]
    if total == 0:
        return torch.tensor(0.0)
    return torch.dot(expert_count, router_prob_sum) * (num_experts / (total * total))


def sp_loss_reduce(losses, num_valid_tokens):
    """Token-weighted mean over SP ranks with zero-valid guard.

    Ref: sequence_parallel/loss.py:24-55: loss_i <- where(n_i>0, loss_i, 0);
    out = sum_i loss_i*n_i / max(sum_i n_i, 1). `losses`/`num_valid_tokens`
    are per-rank lists (the oracle models the all-reduce).
    """
    num = torch.tensor(0.0)
    den = torch.tensor(0.0)
    for loss, n in zip(losses, num_valid_tokens):
        n = torch.as_tensor(n, dtype=torch.float32)
        loss = torch.where(n > 0, torch.as_tensor(loss, dtype=torch.float32),
                           torch.tensor(0.0))
        num = num + loss * n
        den = den + n
    return num / den.clamp_min(1)


def sp_shift_pad_slice(tensor: torch.Tensor, sp_size: int, rank: int,
                       is_labels: bool, pad_value: int = 0):
    """SP collator per-tensor transform, bit-exact contract.

    Ref: data_collator.py:317-427 (SequenceParallelCollator): labels are
    causally shifted FIRST (labels[1:] + [IGNORE]), every tensor is padded
    along seq to a multiple of sp_size, then sliced into sp_size equal
    chunks, rank r keeping chunk r.
    """
    if is_labels:
        pad_value = IGNORE_INDEX
        tensor = torch.cat(
            [tensor[..., 1:], torch.full_like(tensor[..., :1], IGNORE_INDEX)], dim=-1
        )
    seq = tensor.shape[-1]
    if seq % sp_size != 0:
        pad = sp_size - seq % sp_size
        tensor = torch.cat(
            [tensor, torch.full_like(tensor[..., :1], pad_value).expand(*tensor.shape[:-1], pad)],
            dim=-1,
        )
    chunk = tensor.shape[-1] // sp_size
    return tensor[..., rank * chunk : (rank + 1) * chunk].contiguous()
