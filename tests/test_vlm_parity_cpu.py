"""Qwen2.5-VL parity vs the committed reference golden (vl_golden.pt:
weights + inputs + logits + vision features + 3D rope index, produced by
RUNNING the reference patched modeling — tests/golden/make_vl_golden.py)."""

import os

import pytest
import torch

GOLD = os.path.join(os.path.dirname(__file__), "golden", "vl_golden.pt")


@pytest.fixture(scope="module")
def gold():
    return torch.load(GOLD, weights_only=False)


def _load_reference_weights(model, sd):
    """Map the reference state dict into our VLM (names are kept aligned
    modulo the documented prefixes; patch_embed conv -> linear reshape)."""
    out = {}
    for k, v in sd.items():
        nk = None
        if k.startswith("model.visual."):
            nk = "visual." + k[len("model.visual."):]
            if nk == "visual.patch_embed.proj.weight":
                v = v.reshape(v.shape[0], -1)
        elif k.startswith("model.language_model."):
            nk = "model." + k[len("model.language_model."):]
        elif k == "lm_head.weight":
            nk = k
        if nk is not None:
            out[nk] = v
    missing, unexpected = model.load_state_dict(out, strict=False)
    assert not missing, missing
    assert not unexpected, unexpected


@pytest.fixture(scope="module")
def model(gold):
    from veomni_amd.distributed.parallel_state import (init_parallel_state,
                                                       set_parallel_state)
    from veomni_amd.models import build_vl_model
    from veomni_amd.models.modeling import bind_ops

    set_parallel_state(None)
    init_parallel_state(device_type="cpu")
    bind_ops("eager")
    m = build_vl_model("tiny-vl", dtype=torch.float32)
    _load_reference_weights(m, gold["state_dict"])
    m.eval()
    return m


def test_vision_tower_parity(gold, model):
    """Windowed ViT features vs the reference's pooler_output (fp32)."""
    with torch.no_grad():
        feats = model.visual(gold["inputs/pixel_values"],
                             gold["inputs/grid_thw"].tolist())
    torch.testing.assert_close(feats, gold["vision/pooler_output"],
                               rtol=2e-4, atol=2e-4)


def test_position_ids_3d_parity(gold):
    """3D rope index bit-exact vs the reference get_rope_index."""
    from veomni_amd.models.vlm import vl_position_ids

    pos = vl_position_ids(gold["inputs/input_ids"], 511,
                          gold["inputs/grid_thw"].tolist(), 2)
    assert torch.equal(pos.to(gold["position_ids_3d"].dtype),
                       gold["position_ids_3d"])


def test_vlm_logits_parity(gold, model):
    """End-to-end logits vs the reference forward (fp32)."""
    with torch.no_grad():
        logits, _ = model(gold["inputs/input_ids"],
                          pixel_values=gold["inputs/pixel_values"],
                          image_grid_thw=gold["inputs/grid_thw"])
    ref = gold["logits"]
    torch.testing.assert_close(logits, ref, rtol=2e-3, atol=2e-3)


def test_vlm_train_step_runs(gold, model):
    """fwd+bwd with labels through the shared CE path; grads finite."""
    ids = gold["inputs/input_ids"]
    labels = ids.clone()
    labels[ids == 511] = -100
    loss, _ = model(ids, labels=labels,
                    pixel_values=gold["inputs/pixel_values"],
                    image_grid_thw=gold["inputs/grid_thw"])
    loss.backward()
    for n, p in model.named_parameters():
        if p.grad is not None:
            assert torch.isfinite(p.grad).all(), n
    model.zero_grad(set_to_none=True)


def test_vl_moe_train_step_runs():
    """BASELINE config 5 model shape (VL + MoE text): fwd+bwd, aux loss on,
    grads finite (tiny-vl-moe)."""
    import torch as t

    from veomni_amd.data import synthetic_vlm_batch
    from veomni_amd.distributed.parallel_state import (init_parallel_state,
                                                       set_parallel_state)
    from veomni_amd.models import VL_PRESETS, _init_vl_presets, build_vl_model
    from veomni_amd.models.modeling import bind_ops

    set_parallel_state(None)
    init_parallel_state(device_type="cpu")
    bind_ops("eager")
    _init_vl_presets()
    m = build_vl_model("tiny-vl-moe", dtype=t.float32)
    b = synthetic_vlm_batch(VL_PRESETS["tiny-vl-moe"], 128, batch=2, seed=1)
    loss, aux = m(**b)
    assert aux is not None and float(aux) > 0
    (loss + 0.001 * aux).backward()
    for n, p_ in m.named_parameters():
        if p_.grad is not None:
            assert t.isfinite(p_.grad).all(), n
