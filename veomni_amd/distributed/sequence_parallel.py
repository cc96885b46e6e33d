"""Ulysses sequence parallelism: seq<->head all-to-all re-sharding + SP loss.

API parity target: /root/reference/veomni/distributed/sequence_parallel/
  ulysses.py:74-307 (`all_to_all_tensor`, `_SeqAllToAll`,
  `gather_seq_scatter_heads`, `gather_heads_scatter_seq`, `_Slice`, `_Gather`)
  loss.py:24-68 (`reduce_sequence_parallel_loss`)
  comm.py:29-94 (test seam `set_ulysses_sequence_parallel_group`).

MI355X note: a2a over the xGMI full mesh is link-parallel (each peer pair has
a dedicated ~153 GB/s link), so the Ulysses exchange is the cheap collective
on this topology; the transport is torch c10d (= RCCL on ROCm).
"""

from __future__ import annotations

from typing import Any, Optional, Tuple

import torch
import torch.distributed as dist
from torch import Tensor

from .parallel_state import get_parallel_state

# Test-only injection seam (ref comm.py:29-77): unit tests drive SP without a
# ParallelState by setting the group directly.
_ULYSSES_SP_GROUP_OVERRIDE: Optional[dist.ProcessGroup] = None


def set_ulysses_sequence_parallel_group(group: Optional[dist.ProcessGroup]) -> None:
    global _ULYSSES_SP_GROUP_OVERRIDE
    _ULYSSES_SP_GROUP_OVERRIDE = group


def get_ulysses_sequence_parallel_group() -> Optional[dist.ProcessGroup]:
    if _ULYSSES_SP_GROUP_OVERRIDE is not None:
        return _ULYSSES_SP_GROUP_OVERRIDE
    return get_parallel_state().ulysses_group


def get_ulysses_sequence_parallel_world_size(group=None) -> int:
    group = get_ulysses_sequence_parallel_group() if group is None else group
    return dist.get_world_size(group) if group is not None else 1


def _all_to_all_single_2d(x: Tensor, scatter_dim: int, gather_dim: int, group) -> Tensor:
    """all_to_all over dims {0,1} via one all_to_all_single + reshapes.

    Semantics of ref ulysses.py:96-135: scatter_dim!=0 pre-shuffles so the
    exchanged chunks land contiguously; scatter_dim==0 post-concatenates
    along gather_dim.
    """
    ws = dist.get_world_size(group)
    assert scatter_dim <= 1 and gather_dim <= 1
    if scatter_dim != 0:
        g, s = x.shape[gather_dim], x.shape[scatter_dim]
        x = (
            x.reshape([g, ws, s // ws] + list(x.shape[2:]))
            .transpose(0, 1)
            .reshape([g * ws, s // ws] + list(x.shape[2:]))
            .contiguous()
        )
    x = x.contiguous()
    out = torch.empty_like(x)
    dist.all_to_all_single(out, x, group=group)
    if scatter_dim == 0:
        out = torch.cat(out.split(x.size(0) // ws), dim=gather_dim)
    return out


def all_to_all_tensor(x: Tensor, scatter_dim: int, gather_dim: int, group) -> Tensor:
    if scatter_dim <= 1 and gather_dim <= 1:
        return _all_to_all_single_2d(x, scatter_dim, gather_dim, group)
    ws = dist.get_world_size(group)
    inputs = [t.contiguous() for t in torch.tensor_split(x, ws, scatter_dim)]
    outputs = [torch.empty_like(inputs[0]) for _ in range(ws)]
    dist.all_to_all(outputs, inputs, group=group)
    return torch.cat(outputs, dim=gather_dim).contiguous()


class _SeqAllToAll(torch.autograd.Function):
    """a2a whose backward is the transposed a2a (ref ulysses.py:151-173)."""

    @staticmethod
    def forward(ctx: Any, group, x: Tensor, scatter_dim: int, gather_dim: int) -> Tensor:
        ctx.group = group
        ctx.scatter_dim = scatter_dim
        ctx.gather_dim = gather_dim
        return all_to_all_tensor(x, scatter_dim, gather_dim, group)

    @staticmethod
    def backward(ctx: Any, *grad_output: Tensor):
        return (None, all_to_all_tensor(grad_output[0], ctx.gather_dim, ctx.scatter_dim, ctx.group), None, None)


def _pad(x: Tensor, dim: int, pad: int) -> Tensor:
    shape = list(x.shape)
    shape[dim] = pad
    return torch.cat([x, x.new_zeros(shape)], dim=dim)


def gather_seq_scatter_heads(x: Tensor, seq_dim: int, head_dim: int,
                             unpadded_dim_size: int = 0, group=None) -> Tensor:
    """[.., S/sp, h, ..] -> [.., S, h/sp, ..] (ref ulysses.py:247-265)."""
    group = get_ulysses_sequence_parallel_group() if group is None else group
    if not group:
        return x
    sp = get_ulysses_sequence_parallel_world_size(group)
    x = _SeqAllToAll.apply(group, x, head_dim, seq_dim)
    if unpadded_dim_size and unpadded_dim_size % sp != 0:
        pad = x.size(seq_dim) - unpadded_dim_size
        x = x.narrow(seq_dim, 0, x.size(seq_dim) - pad).contiguous()
    return x


def gather_heads_scatter_seq(x: Tensor, head_dim: int, seq_dim: int, group=None) -> Tensor:
    """[.., S, h/sp, ..] -> [.., S/sp, h, ..] (ref ulysses.py:232-244)."""
    group = get_ulysses_sequence_parallel_group() if group is None else group
    if not group:
        return x
    sp = get_ulysses_sequence_parallel_world_size(group)
    size = x.size(seq_dim)
    if size % sp != 0:
        x = _pad(x, seq_dim, sp - size % sp)
    return _SeqAllToAll.apply(group, x, seq_dim, head_dim)


class _Gather(torch.autograd.Function):
    """all-gather along dim; backward all-reduces then takes own slice
    (ref ulysses.py:198-229)."""

    @staticmethod
    def forward(ctx, group, x: Tensor, dim: int, grad_scale: bool = False) -> Tensor:
        ctx.group = group
        ctx.dim = dim
        ctx.grad_scale = grad_scale
        ws = dist.get_world_size(group)
        ctx.ws = ws
        ctx.rank = dist.get_rank(group)
        x = x.contiguous()
        ctx.dim_size = x.size(dim)
        out = [torch.empty_like(x) for _ in range(ws)]
        dist.all_gather(out, x, group=group)
        return torch.cat(out, dim=dim)

    @staticmethod
    def backward(ctx, grad_output: Tensor):
        if ctx.grad_scale:
            grad_output = grad_output * ctx.ws
        dist.all_reduce(grad_output, op=dist.ReduceOp.SUM, group=ctx.group)
        return (None, grad_output.narrow(ctx.dim, ctx.rank * ctx.dim_size, ctx.dim_size).contiguous(), None, None)


class _Slice(torch.autograd.Function):
    """slice own chunk along dim; backward all-gathers (ref ulysses.py:176-195)."""

    @staticmethod
    def forward(ctx, group, x: Tensor, dim: int, scale_grad: bool = True) -> Tensor:
        ctx.group = group
        ctx.dim = dim
        ctx.scale_grad = scale_grad
        ws = dist.get_world_size(group)
        ctx.ws = ws
        rank = dist.get_rank(group)
        chunk = x.shape[dim] // ws
        return x.narrow(dim, rank * chunk, chunk).contiguous()

    @staticmethod
    def backward(ctx, grad_output: Tensor):
        grad_output = grad_output.contiguous()
        ws = ctx.ws
        shape = list(grad_output.shape)
        shape[0] = shape[0] * ws
        out = torch.empty(shape, dtype=grad_output.dtype, device=grad_output.device)
        dist.all_gather_into_tensor(out, grad_output, group=ctx.group)
        if ctx.scale_grad:
            out = out / ws
        split = grad_output.shape[0]
        return (None, torch.cat(out.split(split), dim=ctx.dim), None, None)


def gather_outputs(x: Tensor, gather_dim: int, padding_dim: Optional[int] = None,
                   unpad_dim_size: Optional[int] = None, scale_grad: bool = True,
                   group=None) -> Tensor:
    group = get_ulysses_sequence_parallel_group() if group is None else group
    if not group:
        return x
    x = _Gather.apply(group, x, gather_dim, scale_grad)
    if unpad_dim_size and padding_dim is not None and x.size(padding_dim) > unpad_dim_size:
        x = x.narrow(padding_dim, 0, unpad_dim_size).contiguous()
    return x


def slice_input_tensor(x: Tensor, dim: int, scale_grad: bool = True,
                       group=None) -> Tensor:
    """Slice this rank's seq chunk (ref sequence_parallel/data.py
    slice_input_tensor); backward all-gathers."""
    group = get_ulysses_sequence_parallel_group() if group is None else group
    if not group:
        return x
    return _Slice.apply(group, x, dim, scale_grad)


class ReduceLoss(torch.autograd.Function):
    """Token-weighted SP loss mean with zero-valid guard (ref loss.py:24-65)."""

    @staticmethod
    def forward(ctx, loss: Tensor, num_valid_tokens: Tensor, group=None) -> Tensor:
        if group is None:
            group = get_ulysses_sequence_parallel_group()
        loss = torch.where(num_valid_tokens > 0, loss, torch.zeros_like(loss))
        local_num = num_valid_tokens.detach().clone()
        loss = loss * num_valid_tokens
        dist.all_reduce(loss, group=group)
        dist.all_reduce(num_valid_tokens, group=group)
        ctx.save_for_backward(local_num, num_valid_tokens)
        ctx.ws = dist.get_world_size(group) if group else 1
        return loss / num_valid_tokens.clamp_min(1)

    @staticmethod
    def backward(ctx, grad_output: Tensor) -> Tuple[Tensor, None, None]:
        local_num, global_num = ctx.saved_tensors
        grad = ctx.ws * local_num * grad_output / global_num.clamp(min=1)
        return grad, None, None


def reduce_sequence_parallel_loss(loss: Tensor, num_valid_tokens: Tensor, group=None) -> Tensor:
    return ReduceLoss.apply(loss, num_valid_tokens, group)


# ---------------------------------------------------------------- async path
# Async Ulysses (parity target: async_ulysses.py:48-419): the three QKV
# all-to-alls are launched per-tensor with async_op=True so the exchange of
# q overlaps the k/v projections (and all three overlap on the comm
# stream). BACKWARD MIRRORS the overlap: _A2AWait.backward STARTS the
# reverse exchange async and _A2AStartSeqHeads.backward finishes it — in
# autograd's reverse order the three reverse a2a all start back-to-back and
# fly while the v/k projection weight-grad GEMMs execute.
_ASYNC_WORK: dict = {}


class _A2AStartSeqHeads(torch.autograd.Function):
    """Start [.., S/sp, h, ..] -> [.., S, h/sp, ..] async (scatter over the
    head dim, gather over the LEADING seq dim — the buffer is directly the
    gathered layout, no post-concat needed)."""

    @staticmethod
    def forward(ctx, group, x: Tensor, seq_dim: int, head_dim: int) -> Tensor:
        ctx.group = group
        ctx.seq_dim = seq_dim
        ctx.head_dim = head_dim
        ws = dist.get_world_size(group)
        assert seq_dim == 0 and head_dim == 1, "async path expects [S, h, D] tensors"
        g, s = x.shape[0], x.shape[1]
        xs = (
            x.reshape([g, ws, s // ws] + list(x.shape[2:]))
            .transpose(0, 1)
            .reshape([g * ws, s // ws] + list(x.shape[2:]))
            .contiguous()
        )
        out = torch.empty_like(xs)
        work = dist.all_to_all_single(out, xs, group=group, async_op=True)
        # (work, input, output): both buffers stay referenced until the wait
        _ASYNC_WORK[id(out)] = (work, xs, out)
        return out

    @staticmethod
    def backward(ctx, grad_output: Tensor):
        ent = _ASYNC_WORK.pop(id(grad_output), None)
        ws = dist.get_world_size(ctx.group)
        if ent is not None:
            # mirrored async: _A2AWait.backward started the reverse exchange;
            # finish it and apply the post-concat (scatter seq, gather heads)
            ent[0].wait()
            out = torch.cat(grad_output.split(grad_output.size(0) // ws),
                            dim=ctx.head_dim)
            return (None, out, None, None)
        return (None, all_to_all_tensor(grad_output, ctx.seq_dim, ctx.head_dim, ctx.group), None, None)


class _A2AWait(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x: Tensor, group) -> Tensor:
        ctx.group = group
        ent = _ASYNC_WORK.pop(id(x), None)
        if ent is not None:
            ent[0].wait()
        return x.view_as(x)

    @staticmethod
    def backward(ctx, grad_output: Tensor):
        if ctx.group is not None:
            # start the reverse exchange NOW; the paired _A2AStartSeqHeads
            # backward (several autograd nodes later) finishes it. The stash
            # holds (work, input, output) so both buffers stay alive until
            # the pop even if the engine were ever to hand the pair node a
            # different (accumulated) tensor — in that unreachable-by-
            # construction case the pair falls back to the sync exchange
            # and the orphan entry keeps the in-flight buffers valid.
            x = grad_output.contiguous()
            out = torch.empty_like(x)
            work = dist.all_to_all_single(out, x, group=ctx.group, async_op=True)
            _ASYNC_WORK[id(out)] = (work, x, out)
            return out, None
        return grad_output, None


def gather_seq_scatter_heads_async(x: Tensor, group=None) -> Tensor:
    """Start the seq->head exchange; returns the (not yet valid) buffer."""
    group = get_ulysses_sequence_parallel_group() if group is None else group
    if not group:
        return x
    return _A2AStartSeqHeads.apply(group, x, 0, 1)


def wait_gathered(x: Tensor, group=None) -> Tensor:
    group = get_ulysses_sequence_parallel_group() if group is None else group
    if not group:
        return x
    return _A2AWait.apply(x, group)
