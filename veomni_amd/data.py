"""Synthetic data + sequence-parallel collation for the training step.

Parity targets: the reference's DummyDataset / data_generators synthetic
batches (veomni/data/dummy_dataset.py, tests/tools/data_generators.py:14 —
`randint(0, vocab, [1, seq_len], seed)`), and SequenceParallelCollator
(data_collator.py:317-427): labels shifted BEFORE padding/slicing so the CE
needs no shift under SP; every tensor padded to an sp multiple then sliced
per rank.
"""

from __future__ import annotations

import torch

from .distributed.parallel_state import get_parallel_state

IGNORE_INDEX = -100


def synthetic_batch(vocab_size: int, seq_len: int, batch: int = 1,
                    seed: int = 42, device="cpu"):
    g = torch.Generator().manual_seed(seed)
    input_ids = torch.randint(0, vocab_size, (batch, seq_len), generator=g)
    labels = input_ids.clone()
    return {
        "input_ids": input_ids.to(device),
        "labels": labels.to(device),
        "position_ids": torch.arange(seq_len)[None].expand(batch, -1).to(device),
    }


def sp_shift(labels: torch.Tensor) -> torch.Tensor:
    """Causal shift applied in the collator under SP (ref :374-378)."""
    shifted = labels[..., 1:].contiguous()
    return torch.nn.functional.pad(shifted, (0, 1), "constant", IGNORE_INDEX)


def sp_pad(t: torch.Tensor, sp_size: int, pad_value, dim: int = -1) -> torch.Tensor:
    seq = t.size(dim)
    chunk = (seq + sp_size - 1) // sp_size
    pad = chunk * sp_size - seq
    if pad == 0:
        return t
    shape = list(t.shape)
    shape[dim] = pad
    return torch.cat([t, torch.full(shape, pad_value, dtype=t.dtype, device=t.device)], dim=dim)


def sp_slice(t: torch.Tensor, sp_size: int, sp_rank: int, dim: int = -1) -> torch.Tensor:
    chunk = t.size(dim) // sp_size
    return t.narrow(dim, sp_rank * chunk, chunk).contiguous()


def sp_collate(batch: dict, sp_size: int | None = None, sp_rank: int | None = None) -> dict:
    """SequenceParallelCollator semantics for the text path (ref :317-427)."""
    ps = get_parallel_state()
    sp_size = ps.sp_size if sp_size is None else sp_size
    sp_rank = ps.sp_rank if sp_rank is None else sp_rank
    if sp_size <= 1:
        return batch
    out = dict(batch)
    out["labels"] = sp_slice(sp_pad(sp_shift(batch["labels"]), sp_size, IGNORE_INDEX), sp_size, sp_rank)
    out["input_ids"] = sp_slice(sp_pad(batch["input_ids"], sp_size, 0), sp_size, sp_rank)
    if "position_ids" in batch:
        out["position_ids"] = sp_slice(sp_pad(batch["position_ids"], sp_size, 0), sp_size, sp_rank)
    return out
