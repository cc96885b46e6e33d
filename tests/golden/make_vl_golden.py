#!/usr/bin/env python3
"""Qwen2.5-VL golden vectors by RUNNING THE REFERENCE (build container only).

Builds the reference's patched Qwen2.5-VL at the tiny-vl shape, runs one
forward on seeded image+text inputs, and commits weights + inputs + logits +
vision features so our VLM's parity tests never need the reference at run
time. Anchor: /root/reference/veomni/models/transformers/qwen2_5vl/generated/
patched_modeling_qwen2_5_vl_gpu.py.

Usage: python tests/golden/make_vl_golden.py  -> tests/golden/vl_golden.pt
"""

import os
import sys

import torch

REF = "/root/reference"
assert os.path.isdir(REF), "reference repo required (build container only)"
sys.path.insert(0, REF)

torch.manual_seed(0)
# the patched VL forward all-gathers input_ids over the sp group even in the
# single-process default state — give it a 1-rank gloo world
import torch.distributed as dist  # noqa: E402

os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
os.environ.setdefault("MASTER_PORT", "29541")
if not dist.is_initialized():
    dist.init_process_group("gloo", rank=0, world_size=1)
torch.cuda.get_device_capability = lambda *a, **k: (8, 0)
torch.cuda.get_device_name = lambda *a, **k: "cpu-golden"
torch.cpu.get_device_name = lambda *a, **k: "cpu-golden"

from transformers.models.qwen2_5_vl.configuration_qwen2_5_vl import (  # noqa: E402
    Qwen2_5_VLConfig, Qwen2_5_VLVisionConfig)

from veomni.models.transformers.qwen2_5vl.generated import (  # noqa: E402
    patched_modeling_qwen2_5_vl_gpu as m,
)

vc = Qwen2_5_VLVisionConfig(
    depth=2, hidden_size=64, num_heads=4, intermediate_size=128,
    out_hidden_size=128, patch_size=2, temporal_patch_size=1, in_channels=3,
    spatial_merge_size=2, window_size=8, fullatt_block_indexes=[1],
    hidden_act="silu")
cfg = Qwen2_5_VLConfig(
    vision_config=vc.to_dict(), vocab_size=512, hidden_size=128,
    intermediate_size=256, num_hidden_layers=2, num_attention_heads=4,
    num_key_value_heads=2, rope_theta=1000000.0, rms_norm_eps=1e-6,
    rope_scaling={"type": "mrope", "mrope_section": [4, 6, 6]},
    image_token_id=511, video_token_id=510, vision_start_token_id=509,
    bos_token_id=0, eos_token_id=1, attn_implementation="sdpa",
    tie_word_embeddings=False)

model = m.Qwen2_5_VLForConditionalGeneration(cfg)
model = model.float().eval()
gen = torch.Generator().manual_seed(7)
with torch.no_grad():
    for name, p in sorted(model.named_parameters(), key=lambda kv: kv[0]):
        if p.dim() <= 1 and ("norm" in name or "ln_q" in name):
            p.fill_(1.0)
        elif name.endswith(".bias"):
            # small nonzero biases so the bias paths are actually exercised
            p.copy_(torch.randn(p.shape, generator=gen) * 0.02)
        else:
            p.copy_(torch.randn(p.shape, generator=gen) * 0.05)

# inputs: two images (grids (1,4,4) and (1,2,4) -> 16+8 patches -> 4+2 merged
# tokens) inside a 32-token sequence
t1 = (1, 4, 4)
t2 = (1, 2, 4)
n_patch = t1[0] * t1[1] * t1[2] + t2[0] * t2[1] * t2[2]
pixel_values = torch.randn(n_patch, 3 * 1 * 2 * 2, generator=gen)
grid = torch.tensor([t1, t2])
S = 32
input_ids = torch.randint(2, 500, (1, S), generator=gen)
input_ids[0, 4:8] = 511    # image 1: 16 patches / merge_unit 4 = 4 tokens
input_ids[0, 20:22] = 511  # image 2: 8 / 4 = 2 tokens

with torch.no_grad():
    vis = model.model.get_image_features(pixel_values, grid, return_dict=True)
    out = model(input_ids=input_ids, pixel_values=pixel_values,
                image_grid_thw=grid, use_cache=False)
    # 3D rope index for the same sequence (mrope parity)
    mm_tt = m.mm_token_type_ids_from_input_ids(input_ids, model.config)
    pos3, _ = model.model.get_rope_index(input_ids, mm_tt, image_grid_thw=grid)

golden = {
    "cfg/vision": vc.to_dict(),
    "inputs/pixel_values": pixel_values,
    "inputs/grid_thw": grid,
    "inputs/input_ids": input_ids,
    "vision/pooler_output": vis.pooler_output,
    "logits": out.logits,
    "position_ids_3d": pos3,
    "state_dict": {k: v.clone() for k, v in model.state_dict().items()},
}

out_path = os.path.join(os.path.dirname(os.path.abspath(__file__)), "vl_golden.pt")
torch.save(golden, out_path)
print(f"wrote {out_path} ({os.path.getsize(out_path)/2**20:.1f} MiB)")
