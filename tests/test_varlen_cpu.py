"""Packed-varlen (cu_seqlens) attention semantics on CPU.

The reference routes packed multi-document batches through flash-attn's
varlen path (ops/kernels/attention/flash.py:61-91) with cu_seq_lens kwargs
produced by the collator (data/data_collator.py:50). Here the eager oracle is
a block-diagonal causal mask derived from per-token document starts; these
tests pin that oracle against literal per-document attention, and the model's
kwargs plumbing end-to-end. The HIP kernels are pinned against the same
oracle on the GPU (tests/test_gpu_kernels.py::test_flash_attention_varlen_*).
"""

import math

import pytest
import torch

from veomni_amd.ops.kernels.attention import docs_from_cu_seqlens


def test_docs_from_cu_seqlens():
    cu = torch.tensor([0, 3, 7, 10], dtype=torch.int32)
    ds, de = docs_from_cu_seqlens(cu, 10)
    assert ds.tolist() == [0, 0, 0, 3, 3, 3, 3, 7, 7, 7]
    assert de.tolist() == [3, 3, 3, 7, 7, 7, 7, 10, 10, 10]
    # single document -> plain causal, no varlen tensors
    ds1, de1 = docs_from_cu_seqlens(torch.tensor([0, 10], dtype=torch.int32), 10)
    assert ds1 is None and de1 is None
    # cu must cover the padded sequence (tail coalesced by the collator)
    with pytest.raises(AssertionError):
        docs_from_cu_seqlens(cu, 12)


def test_sdpa_doc_mask_vs_per_document():
    """Block-diagonal eager oracle == literal per-document causal attention."""
    from veomni_amd.models.modeling import sdpa_attention

    torch.manual_seed(0)
    S, Hq, Hkv, D = 48, 4, 2, 16
    cu = torch.tensor([0, 13, 30, 48], dtype=torch.int32)
    ds, _ = docs_from_cu_seqlens(cu, S)
    q = torch.randn(1, Hq, S, D)
    k = torch.randn(1, Hkv, S, D)
    v = torch.randn(1, Hkv, S, D)
    scale = 1.0 / math.sqrt(D)
    out, _ = sdpa_attention(None, q, k, v, None, scaling=scale, doc_start=ds)
    # per-document reference
    for i in range(cu.numel() - 1):
        a, b = int(cu[i]), int(cu[i + 1])
        ref, _ = sdpa_attention(None, q[:, :, a:b], k[:, :, a:b], v[:, :, a:b],
                                None, scaling=scale)
        torch.testing.assert_close(out[:, a:b], ref, rtol=1e-5, atol=1e-5)


def test_model_packed_varlen_vs_per_document():
    """Eager model on a packed 3-document batch (cu_seq_lens kwargs +
    per-document position ids, the collator contract) must reproduce each
    document's standalone logits — no cross-document attention."""
    from veomni_amd.models import build_model
    from veomni_amd.models.modeling import bind_ops

    bind_ops("eager")
    torch.manual_seed(1)
    model = build_model("tiny-dense", dtype=torch.float32, device="cpu")
    model.eval()
    S = 32
    cu = torch.tensor([0, 9, 21, 32], dtype=torch.int32)
    ids = torch.randint(0, model.config.vocab_size, (1, S))
    pos = torch.cat([torch.arange(int(cu[i + 1]) - int(cu[i]))
                     for i in range(cu.numel() - 1)])[None]
    with torch.no_grad():
        packed_logits, _ = model(ids, position_ids=pos, cu_seq_lens_q=cu,
                                 cu_seq_lens_k=cu)
        for i in range(cu.numel() - 1):
            a, b = int(cu[i]), int(cu[i + 1])
            solo_logits, _ = model(ids[:, a:b])
            torch.testing.assert_close(packed_logits[:, a:b], solo_logits,
                                       rtol=2e-4, atol=2e-4)


def test_model_packed_varlen_single_doc_noop():
    """cu_seqlens spanning the whole batch (one document) must be a no-op."""
    from veomni_amd.models import build_model
    from veomni_amd.models.modeling import bind_ops

    bind_ops("eager")
    torch.manual_seed(2)
    model = build_model("tiny-dense", dtype=torch.float32, device="cpu")
    model.eval()
    ids = torch.randint(0, model.config.vocab_size, (1, 24))
    cu = torch.tensor([0, 24], dtype=torch.int32)
    with torch.no_grad():
        a, _ = model(ids, cu_seq_lens_q=cu, cu_seq_lens_k=cu)
        b, _ = model(ids)
    torch.testing.assert_close(a, b, rtol=0, atol=0)
