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


def synthetic_vlm_batch(vl_config, seq_len: int, batch: int = 1,
                        image_frac: float = 0.4, seed: int = 42, device="cpu"):
    """Synthetic image-text batch for the VLM bench (BASELINE config 4):
    per sequence, ~image_frac of the tokens are image placeholders backed by
    one synthetic image whose (1, h, w) patch grid merges down to that token
    count; the rest are random text. No network/datasets — same policy as
    synthetic_batch (reference DummyDataset pattern)."""
    g = torch.Generator().manual_seed(seed)
    v = vl_config.vision
    merge = v.spatial_merge_size
    n_img_tok = int(seq_len * image_frac)
    # grid (1, h, w) with h*w/merge^2 == n_img_tok; keep w fixed-ish
    gw = 16 * merge
    gh = max(merge, (n_img_tok * merge * merge // gw // merge) * merge)
    n_img_tok = (gh * gw) // (merge * merge)
    patch_feat = v.in_channels * v.temporal_patch_size * v.patch_size ** 2
    ids = torch.randint(0, vl_config.text.vocab_size - 2, (batch, seq_len), generator=g)
    grids = []
    pixel_rows = []
    for b in range(batch):
        start = 8
        ids[b, start:start + n_img_tok] = vl_config.image_token_id
        grids.append([1, gh, gw])
        pixel_rows.append(torch.randn(gh * gw, patch_feat, generator=g))
    labels = ids.clone()
    labels[ids == vl_config.image_token_id] = IGNORE_INDEX
    return {
        "input_ids": ids.to(device),
        "labels": labels.to(device),
        "pixel_values": torch.cat(pixel_rows).to(device),
        "image_grid_thw": torch.tensor(grids),
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
    """SequenceParallelCollator semantics for the text path (ref :317-427).

    Packed batches: the flash-attention cu_seq_lens are computed from the
    FULL (sp-padded) position_ids BEFORE slicing (ref :415 — the kwargs
    describe the gathered sequence the Ulysses exchange reconstructs; the
    sp-pad tail becomes its own 1-token segment via the position_ids == 0
    convention)."""
    ps = get_parallel_state()
    sp_size = ps.sp_size if sp_size is None else sp_size
    sp_rank = ps.sp_rank if sp_rank is None else sp_rank
    if sp_size <= 1:
        return batch
    out = dict(batch)
    out["labels"] = sp_slice(sp_pad(sp_shift(batch["labels"]), sp_size, IGNORE_INDEX), sp_size, sp_rank)
    out["input_ids"] = sp_slice(sp_pad(batch["input_ids"], sp_size, 0), sp_size, sp_rank)
    if "position_ids" in batch:
        padded_pos = sp_pad(batch["position_ids"], sp_size, 0)
        cu, max_len = fa_kwargs_from_position_ids(padded_pos)
        out["cu_seq_lens_q"] = cu
        out["cu_seq_lens_k"] = cu
        out["max_length_q"] = max_len
        out["max_length_k"] = max_len
        out["position_ids"] = sp_slice(padded_pos, sp_size, sp_rank)
    return out


# --------------------------------------------------------------------------
# Dynamic batching (§8f-2). Behavior parity targets:
#   SampleBuffer           ~ DynBszBuffer (ref data/dynamic_batching.py:29-143)
#   TextBatchingStrategy   ~ ref :166-240
#   DynamicBatchDataLoader ~ DynamicBatchSizeDataLoader (ref :249-404)
# Token-count selection is integer work — bit-exact vs the reference on the
# same sample stream (tests/golden/dynbatch.json, generated by RUNNING the
# reference). Restated, not copied: same observable behavior (selection
# order, warmup schedule, tail padding, resume), own structure.
import copy as _copy
import sys as _sys
from collections import deque as _deque


class SampleBuffer:
    """Holds pending samples; selects a micro batch greedily in arrival
    order: a sample is taken when it fits the remaining effective budget AND
    the physical cap; the first sample of a batch is always taken (may alone
    exceed the caps). Selected samples are lazily deleted until flush()."""

    def __init__(self, get_length_fn=None, get_physical_length_fn=None):
        self._items = []
        self._lens = []
        self._plens = []
        self._taken = []          # indices selected since the last flush
        self._scan = 0            # resume point of the selection scan
        self.total_tokens = 0
        self.total_physical_tokens = 0
        self._len_fn = get_length_fn
        self._plen_fn = get_physical_length_fn

    def append(self, item):
        n = (int(self._len_fn(item)) if self._len_fn is not None
             else int(item["attention_mask"].sum()))
        pn = int(self._plen_fn(item)) if self._plen_fn is not None else n
        self._items.append(item)
        self._lens.append(n)
        self._plens.append(pn)
        self.total_tokens += n
        self.total_physical_tokens += pn

    def __len__(self):
        return len(self._items)

    def get_samples(self, n_token_per_iter, force=True, physical_token_cap=None):
        got = []
        eff = 0
        phys = 0
        while self._scan < len(self._items) and eff < n_token_per_iter:
            i = self._scan
            n, pn = self._lens[i], self._plens[i]
            fits = n <= n_token_per_iter - eff and (
                physical_token_cap is None or pn <= physical_token_cap - phys)
            if i not in self._taken and ((force and eff == 0) or fits):
                eff += n
                phys += pn
                got.append(self._items[i])
                self._taken.append(i)
            self._scan += 1
        assert got, "no sample selected"
        return got

    def flush(self):
        self._scan = 0
        taken = set(self._taken)
        self.total_tokens -= sum(self._lens[i] for i in self._taken)
        self.total_physical_tokens -= sum(self._plens[i] for i in self._taken)
        keep = [i for i in range(len(self._items)) if i not in taken]
        self._items = [self._items[i] for i in keep]
        self._lens = [self._lens[i] for i in keep]
        self._plens = [self._plens[i] for i in keep]
        self._taken = []


class TextBatchingStrategy:
    """Emit a micro batch once >= buffer_size samples are pending AND the
    pending tokens cover one micro batch (effective or physical). Linear
    batch-size warmup over bsz_warmup_steps."""

    def __init__(self, token_micro_bsz, buffer_size=500, bsz_warmup_steps=0,
                 bsz_warmup_init_mbtoken=200, get_length_fn=None,
                 physical_token_cap=None, get_physical_length_fn=None):
        self.token_micro_bsz = token_micro_bsz
        self.buffer_size = buffer_size
        self.bsz_warmup_steps = bsz_warmup_steps
        self.bsz_warmup_init_mbtoken = bsz_warmup_init_mbtoken
        if bsz_warmup_steps > 0:
            assert bsz_warmup_init_mbtoken > 0
        self.physical_token_cap = physical_token_cap
        self.buffer = SampleBuffer(get_length_fn, get_physical_length_fn)
        self._step = 0

    def is_ready_for_micro_batch(self):
        enough = len(self.buffer) >= self.buffer_size
        eff = enough and self.buffer.total_tokens >= self.token_micro_bsz
        phys = (enough and self.physical_token_cap is not None
                and self.buffer.total_physical_tokens >= self.physical_token_cap)
        return eff or phys

    def put_item(self, item):
        if item["input_ids"].shape[-1] <= 1:
            return  # empty string guard (ref :216-218)
        self.buffer.append(item)

    def current_budget(self):
        if self.bsz_warmup_steps > 0 and self._step <= self.bsz_warmup_steps:
            span = self.token_micro_bsz - self.bsz_warmup_init_mbtoken
            return (span * self._step // self.bsz_warmup_steps
                    + self.bsz_warmup_init_mbtoken)
        return self.token_micro_bsz

    def get_micro_batch(self, step):
        self._step = step
        got = self.buffer.get_samples(self.current_budget(),
                                      physical_token_cap=self.physical_token_cap)
        self.buffer.flush()
        return got

    def empty(self):
        return len(self.buffer) == 0


class DynamicBatchDataLoader:
    """Wraps a DataLoader: drains it into the strategy, yields lists of
    num_micro_batch collated micro batches; wraps epochs until `length`
    steps; with drop_last=False flushes the tail and pads the final group
    with deep-copied batches tagged padding_flag=True. state_dict/
    load_state_dict resume mid-stream (non-underscore fields)."""

    def __init__(self, dataloader, batching_strategy, collate_fn=None,
                 num_micro_batch=1, length=0, drop_last=True):
        self.batching_strategy = batching_strategy
        self.num_micro_batch = num_micro_batch
        self.step = 0
        self._collate = collate_fn
        self._source = dataloader
        self._drop_last = drop_last
        self._resume = False
        if length > 0:
            self._length = length
        elif length == -1:
            self._length = _sys.maxsize
        else:
            self._length = len(dataloader)

    def __len__(self):
        return self._length

    def __iter__(self):
        if not self._resume:
            self.step = 0
            self._it = iter(self._source)
            self._gen = self._generate()
        self._resume = False
        return self

    def __next__(self):
        return next(self._gen)

    def _emit(self, group):
        mb = self.batching_strategy.get_micro_batch(self.step)
        if self._collate:
            mb = self._collate(mb)
        group.append(mb)
        return mb

    def _generate(self):
        group = []
        while True:
            if self._length and self.step >= self._length:
                return
            if self.batching_strategy.is_ready_for_micro_batch():
                self._emit(group)
                if len(group) == self.num_micro_batch:
                    yield group
                    self.step += 1
                    group = []
            try:
                item = next(self._it)
            except StopIteration:
                if self.step < self._length:
                    self._it = iter(self._source)   # epoch wrap (ref :325-328)
                    item = next(self._it)
                elif not self._drop_last and not self.batching_strategy.empty():
                    mb = None
                    while not self.batching_strategy.empty():
                        mb = self._emit(group)
                        if len(group) == self.num_micro_batch:
                            yield group
                            self.step += 1
                            group = []
                    # ref :339-345 pads unconditionally from the last micro
                    # batch (a tail that drained to an exact multiple yields
                    # one extra all-padding group — mirrored for parity)
                    while len(group) < self.num_micro_batch:
                        pad = _copy.deepcopy(mb)
                        pad["padding_flag"] = True
                        group.append(pad)
                    yield group
                    self.step += 1
                    return
                else:
                    return
            items = [item] if isinstance(item, dict) else item
            for it in items:
                self.batching_strategy.put_item(it)

    def state_dict(self):
        state = {k: v for k, v in self.__dict__.items()
                 if not k.startswith("_") and k != "batching_strategy"}
        if hasattr(self._source, "state_dict"):
            state["dataloader_state"] = self._source.state_dict()
        return _copy.deepcopy(state)

    def load_state_dict(self, state):
        state = dict(state)
        if state.get("num_micro_batch", self.num_micro_batch) != self.num_micro_batch:
            state.pop("num_micro_batch")
        dl_state = state.pop("dataloader_state", None)
        self.__dict__.update(state)
        self._resume = True
        if dl_state is not None and hasattr(self._source, "load_state_dict"):
            self._source.load_state_dict(dl_state)
        self._it = iter(self._source)
        self._gen = self._generate()

    def set_epoch(self, epoch):
        if hasattr(self._source, "set_epoch"):
            self._source.set_epoch(epoch)


# --------------------------------------------------------------------------
# Sample packing (§8f-2 text path). Parity targets:
#   PackingCollator         ~ ref data/data_collator.py:219-317 (text keys)
#   fa_kwargs_from_position_ids ~ ref utils/seqlen_pos_transform_utils.py:59-102
#   tail coalescing         ~ ref :104-143
IGNORE_INDEX = -100

# per-key (pad_value) for the packed/padded last-dim keys
_PACK_PAD_VALUE = {"input_ids": 0, "labels": IGNORE_INDEX,
                   "attention_mask": 1, "position_ids": 0,
                   "image_mask": 0, "video_mask": 0}
# modality keys concatenated along dim 0 (ref DEFAULT_DATA_COLLATE_INFO
# :138-150: pixel tensors and grid tables)
_PACK_DIM0 = ("pixel_values", "pixel_values_videos", "image_grid_hw",
              "image_grid_thw", "video_grid_thw")


def fa_kwargs_from_position_ids(position_ids: torch.Tensor, tail_padding: int = 0):
    """cu_seqlens/max_length for varlen attention from packed position_ids
    (sequence starts where position == 0); optional coalescing of a known
    tail-padding run (each pad token is its own 1-token segment otherwise)."""
    if position_ids.dim() == 3:
        position_ids = position_ids[:, 0, :]
    flat = position_ids.reshape(-1)
    starts = (flat == 0).nonzero().view(-1).to(torch.int32)
    total = torch.tensor([flat.numel()], dtype=torch.int32, device=flat.device)
    cu = torch.cat((starts, total))
    if tail_padding > 0:
        diff = cu[1:] - cu[:-1]
        valid_end = torch.clamp(cu[-1:] - tail_padding, min=0)
        valid = diff[cu[1:] <= valid_end]
        cu = torch.unique_consecutive(
            torch.cat((torch.cumsum(torch.cat(
                (torch.zeros(1, dtype=torch.int64, device=cu.device),
                 valid.to(torch.int64))), 0).to(torch.int32),
                cu[-1:])))
    seqlens = cu[1:] - cu[:-1]
    max_len = int(seqlens.max().item()) if seqlens.numel() else 0
    return cu, max_len


class PackingCollator:
    """Packs a micro batch of 1-D samples into one [1, sum(L)] sequence:
    labels of every non-first sample get IGNORE_INDEX at their boundary,
    optional right-pad to pad_to_length, varlen-attention kwargs derived from
    position_ids (deferred under SP, where the SP collator slices first).

    collate_infos: optional {key: (pack_dim, pad_value)} overriding/extending
    the defaults (reference DataCollateInfo subset: pack_dim -1 = last-dim
    concat + unsqueeze(0) + padding, pack_dim 0 = dim-0 concat).
    metadata_collate_func: reference hook, called on the packed batch when SP
    is NOT enabled (under SP the SP collator calls it after slicing)."""

    def __init__(self, pad_to_length=None, seq_classification=False,
                 sp_enabled=False, collate_infos=None,
                 metadata_collate_func=None):
        self.pad_to_length = pad_to_length
        self.seq_classification = seq_classification
        self.sp_enabled = sp_enabled
        self.pad_values = dict(_PACK_PAD_VALUE)
        self.dim0_keys = set(_PACK_DIM0)
        for key, (dim, pv) in (collate_infos or {}).items():
            if dim == -1:
                self.pad_values[key] = pv
                self.dim0_keys.discard(key)
            elif dim == 0:
                self.dim0_keys.add(key)
                self.pad_values.pop(key, None)
            else:
                raise ValueError(f"pack_dim {dim} unsupported (ref uses -1/0)")
        self.metadata_collate_func = metadata_collate_func

    def __call__(self, features):
        batch = {}
        keys = features[0].keys()
        for key in keys:
            parts = [f[key] for f in features if key in f]
            if key in self.pad_values:
                if key == "labels" and not self.seq_classification:
                    parts = [parts[0]] + [p.clone() for p in parts[1:]]
                    for p in parts[1:]:
                        p[..., 0] = IGNORE_INDEX
                batch[key] = torch.cat(parts, dim=-1).unsqueeze(0)
            elif key in self.dim0_keys:
                batch[key] = torch.cat(parts, dim=0)
            else:
                batch[key] = torch.stack(parts)
        tail = 0
        if self.pad_to_length:
            cur = batch["input_ids"].shape[-1]
            assert cur <= self.pad_to_length, "pad_to_length too small"
            tail = self.pad_to_length - cur
            if tail:
                for key, pv in self.pad_values.items():
                    if key in batch:
                        t = batch[key]
                        pad = t.new_full(t.shape[:-1] + (tail,), pv)
                        batch[key] = torch.cat((t, pad), dim=-1)
        if not self.sp_enabled:
            cu, max_len = fa_kwargs_from_position_ids(batch["position_ids"], tail)
            batch["cu_seq_lens_q"] = cu
            batch["cu_seq_lens_k"] = cu
            batch["max_length_q"] = max_len
            batch["max_length_k"] = max_len
            if tail:
                batch["tail_padding_length"] = torch.tensor(tail, dtype=torch.int32)
            if self.metadata_collate_func is not None:
                self.metadata_collate_func(
                    batch, {"pixel_values": 0, "pixel_values_videos": 0})
        elif tail:
            batch["_linear_attn_tail_padding_length"] = tail
        return batch
