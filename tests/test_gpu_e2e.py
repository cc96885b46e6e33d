"""End-to-end GPU parity (pytest -m gpu): full tiny-model training step
through the HIP op path vs (a) the same model run eagerly on GPU and
(b) the committed reference golden (fp32 CPU run of the reference itself)."""

import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.fixture(autouse=True)
def _eager_after():
    yield
    from veomni_amd.models.modeling import bind_ops

    bind_ops("eager")


def _loss_and_gradnorm(model, batch):
    loss, _ = model(**batch)
    loss.backward()
    gn = torch.nn.utils.get_total_norm(
        [p.grad for p in model.parameters() if p.grad is not None]
    )
    model.zero_grad(set_to_none=True)
    return float(loss), float(gn)


@pytest.mark.parametrize("preset", ["tiny-moe", "tiny-dense"])
def test_model_hip_vs_eager_gpu(preset):
    from veomni_amd.data import synthetic_batch
    from veomni_amd.models import build_model
    from veomni_amd.models.modeling import bind_ops
    from veomni_amd.ops import HIP_OPS_CONFIG

    torch.manual_seed(0)
    batch = synthetic_batch(512, 256, seed=3, device="cuda")
    bind_ops("eager")
    model = build_model(preset, dtype=torch.bfloat16, device="cuda")
    l_eager, g_eager = _loss_and_gradnorm(model, batch)
    bind_ops(HIP_OPS_CONFIG)
    l_hip, g_hip = _loss_and_gradnorm(model, batch)
    assert abs(l_hip - l_eager) / abs(l_eager) < 2e-2, (l_hip, l_eager)
    assert abs(g_hip - g_eager) / max(abs(g_eager), 1e-6) < 5e-2, (g_hip, g_eager)


def test_model_hip_vs_reference_golden(golden):
    """bf16 HIP run vs the fp32 CPU run of the REFERENCE itself (loss within
    bf16 envelope; north-star contract: loss within 1e-3 rel is measured on
    matched precision — here precision differs, so the envelope is wider)."""
    from veomni_amd.models.modeling import bind_ops
    from veomni_amd.ops import HIP_OPS_CONFIG
    from tests.test_model_parity_cpu import _build_from_golden

    model = _build_from_golden(golden).cuda().to(torch.bfloat16)
    bind_ops(HIP_OPS_CONFIG)
    ids = golden["e2e/input_ids"].cuda()
    loss, _ = model(ids, labels=ids.clone())
    ref = float(golden["e2e/loss"])
    assert abs(float(loss) - ref) / ref < 2e-2, (float(loss), ref)


def test_hip_chunk_ce_vs_eager():
    from veomni_amd.ops.kernels.cross_entropy import hip_causal_lm_loss

    torch.manual_seed(9)
    B, T, H, V = 1, 128, 64, 512
    hs = (torch.randn(B, T, H) * 0.3).to(torch.bfloat16).cuda().requires_grad_(True)
    w = (torch.randn(V, H) * 0.05).to(torch.bfloat16).cuda().requires_grad_(True)
    labels = torch.randint(0, V, (B, T)).cuda()
    loss, _, _ = hip_causal_lm_loss(hidden_states=hs, weights=w, labels=labels,
                                    chunk_size=48)
    loss.backward()

    hs2 = hs.detach().clone().requires_grad_(True)
    w2 = w.detach().clone().requires_grad_(True)
    lab = labels[..., 1:].contiguous()
    h = hs2[..., :-1, :].reshape(-1, H)
    logits = torch.nn.functional.linear(h, w2).float()
    ref = torch.nn.functional.cross_entropy(logits, lab.reshape(-1), reduction="sum")
    ref = ref / (lab != -100).sum()
    ref.backward()
    torch.testing.assert_close(loss.float(), ref.detach().float(), rtol=1e-2, atol=1e-3)
    torch.testing.assert_close(hs.grad.float(), hs2.grad.float(), rtol=5e-2, atol=1e-3)
    torch.testing.assert_close(w.grad.float(), w2.grad.float(), rtol=5e-2, atol=1e-3)


def test_smoke_entry():
    import __graft_entry__ as ge

    ge.smoke()


def test_rccl_loopback_fsdp2_smoke():
    """1-GPU RCCL smoke: the FSDP2 wrap machinery and
    the collectives the N>=2 path issues run on hardware over a real NCCL
    (=RCCL) process group BEFORE the driver's first multi-GPU execution.
    world_size=1 -> every collective is a loopback, but the full RCCL init,
    the c10d stream handoffs, and torch's FSDP2 AG/RS mechanics execute."""
    import os

    import torch.distributed as dist

    from veomni_amd.data import synthetic_batch
    from veomni_amd.distributed.fsdp2 import build_parallelize_model
    from veomni_amd.distributed.parallel_state import (init_parallel_state,
                                                       set_parallel_state)
    from veomni_amd.models import build_model
    from veomni_amd.models.modeling import bind_ops
    from veomni_amd.ops import HIP_OPS_CONFIG

    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29531")
    os.environ.setdefault("RANK", "0")
    os.environ.setdefault("WORLD_SIZE", "1")
    created = not dist.is_initialized()
    if created:
        dist.init_process_group("nccl", rank=0, world_size=1)
    try:
        # (a) the raw collectives the EP/SP/FSDP paths issue, as loopbacks
        x = torch.randn(64, 32, device="cuda", dtype=torch.bfloat16)
        out = torch.empty_like(x)
        dist.all_to_all_single(out, x)
        torch.testing.assert_close(out, x)
        ag = torch.empty(64, 32, device="cuda", dtype=torch.bfloat16)
        dist.all_gather_into_tensor(ag, x)
        torch.testing.assert_close(ag, x)
        rs = torch.empty(64, 32, device="cuda", dtype=torch.bfloat16)
        dist.reduce_scatter_tensor(rs, x)
        torch.testing.assert_close(rs, x)
        s = x.sum()
        dist.all_reduce(s)

        # (b) FSDP2 wrap mechanics (fully_shard on a 1-rank mesh) through a
        # full HIP-op training step, parity vs the unwrapped module
        set_parallel_state(None)
        init_parallel_state(device_type="cuda")
        bind_ops(HIP_OPS_CONFIG)
        torch.manual_seed(0)
        batch = synthetic_batch(512, 256, seed=7, device="cuda")

        plain = build_model("tiny-d128", dtype=torch.bfloat16, device="cuda")
        l_plain, g_plain = _loss_and_gradnorm(plain, batch)

        torch.manual_seed(0)
        wrapped = build_model("tiny-d128", dtype=torch.bfloat16, device="cuda")
        wrapped = build_parallelize_model(wrapped, force_wrap=True)
        loss, _ = wrapped(**batch)
        loss.backward()
        gn = wrapped.clip_grad_norm_(1e9)
        wrapped.zero_grad(set_to_none=True)
        assert abs(float(loss) - l_plain) < 2e-2 * max(abs(l_plain), 1.0), \
            (float(loss), l_plain)
        assert abs(float(gn) - g_plain) < 5e-2 * max(g_plain, 1.0), \
            (float(gn), g_plain)
    finally:
        set_parallel_state(None)
        bind_ops("eager")
        if created:
            dist.destroy_process_group()


def test_vlm_hip_vs_eager_gpu():
    """tiny-vl image+text step: HIP op stack vs eager on GPU (the vision
    tower's own SDPA path + the text stack's HIP kernels)."""
    import torch

    from veomni_amd.data import synthetic_vlm_batch
    from veomni_amd.models import VL_PRESETS, _init_vl_presets, build_vl_model
    from veomni_amd.models.modeling import bind_ops
    from veomni_amd.ops import HIP_OPS_CONFIG

    _init_vl_presets()
    torch.manual_seed(0)
    batch = synthetic_vlm_batch(VL_PRESETS["tiny-vl"], 128, seed=5, device="cuda")
    batch["image_grid_thw"] = batch["image_grid_thw"]
    bind_ops("eager")
    model = build_vl_model("tiny-vl", dtype=torch.bfloat16, device="cuda")
    l_eager, g_eager = _loss_and_gradnorm(model, batch)
    bind_ops(HIP_OPS_CONFIG)
    l_hip, g_hip = _loss_and_gradnorm(model, batch)
    bind_ops("eager")
    assert abs(l_hip - l_eager) < 2e-2 * max(abs(l_eager), 1.0), (l_hip, l_eager)
    assert abs(g_hip - g_eager) < 5e-2 * max(g_eager, 1.0), (g_hip, g_eager)
