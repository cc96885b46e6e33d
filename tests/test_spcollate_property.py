"""Property-based parity: our sp_collate (shift/pad/slice) vs the reference's
SequenceParallelCollator methods, across random lengths and sp sizes. Runs
where /root/reference is mounted (goldens cover fixed cases elsewhere)."""

import os

import pytest
import torch

REF = "/root/reference"

pytestmark = pytest.mark.skipif(
    not os.path.isdir(os.path.join(REF, "veomni")),
    reason="reference tree not mounted",
)

try:
    from hypothesis import given, settings
    from hypothesis import strategies as st

    HAVE_HYP = True
except ImportError:  # pragma: no cover
    HAVE_HYP = False

_REF_COLLATOR = None


def _ref_collator_cls():
    global _REF_COLLATOR
    if _REF_COLLATOR is None:
        import importlib
        import sys
        import types

        sys.path.insert(0, REF)
        torch.cuda.get_device_capability = lambda *a, **k: (8, 0)
        torch.cuda.get_device_name = lambda *a, **k: "cpu-x"
        torch.cpu.get_device_name = lambda *a, **k: "cpu-x"
        if "torchdata" not in sys.modules:
            from torch.utils.data import DataLoader as _DL
            from torch.utils.data.distributed import DistributedSampler as _DS

            td = types.ModuleType("torchdata")
            sd = types.ModuleType("torchdata.stateful_dataloader")
            sd.__path__ = []
            samp = types.ModuleType("torchdata.stateful_dataloader.sampler")
            sd.StatefulDataLoader = _DL
            samp.StatefulDistributedSampler = _DS
            td.stateful_dataloader = sd
            sys.modules["torchdata"] = td
            sys.modules["torchdata.stateful_dataloader"] = sd
            sys.modules["torchdata.stateful_dataloader.sampler"] = samp
        if "veomni.data" not in sys.modules:
            pkg = types.ModuleType("veomni.data")
            pkg.__path__ = [os.path.join(REF, "veomni", "data")]
            sys.modules["veomni.data"] = pkg
        _REF_COLLATOR = importlib.import_module(
            "veomni.data.data_collator").SequenceParallelCollator
    return _REF_COLLATOR


if HAVE_HYP:

    @settings(max_examples=30, deadline=None)
    @given(
        L=st.integers(min_value=2, max_value=97),
        sp_size=st.sampled_from([2, 4, 8]),
        seed=st.integers(min_value=0, max_value=10_000),
    )
    def test_shift_pad_slice_matches_reference(L, sp_size, seed):
        from veomni_amd.data import sp_collate

        cls = _ref_collator_cls()
        g = torch.Generator().manual_seed(seed)
        ids = torch.randint(0, 1000, (1, L), generator=g)
        labels = ids.clone()
        for rank in range(sp_size):
            col = object.__new__(cls)
            col.sp_size = sp_size
            col.sp_rank = rank
            shifted = torch.nn.functional.pad(labels[..., 1:], (0, 1),
                                              "constant", -100)
            ref_lab = col.sp_slice(
                "labels", col.sp_padding("labels", shifted, dim=-1,
                                         pad_value=-100), dim=-1)
            ref_ids = col.sp_slice(
                "input_ids", col.sp_padding("input_ids", ids, dim=-1,
                                            pad_value=0), dim=-1)
            ours = sp_collate({"input_ids": ids, "labels": labels.clone()},
                              sp_size=sp_size, sp_rank=rank)
            assert torch.equal(ours["input_ids"], ref_ids), (L, sp_size, rank)
            assert torch.equal(ours["labels"], ref_lab), (L, sp_size, rank)
