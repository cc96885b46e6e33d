"""Model presets for the BASELINE.json configs."""

from .modeling import ForCausalLM, ModelConfig, bind_ops  # noqa: F401

PRESETS = {
    # BASELINE config 1 plumbing model (Qwen2-0.5B-shaped)
    "qwen2-0.5b": ModelConfig(
        name="qwen2-0.5b", vocab_size=151936, hidden_size=896, intermediate_size=4864,
        num_hidden_layers=24, num_attention_heads=14, num_key_value_heads=2,
        head_dim=64, rope_theta=1000000.0, attention_bias=True, qk_norm=False,
        tie_word_embeddings=True,
    ),
    # BASELINE config 2: Llama-3-8B
    "llama3-8b": ModelConfig(
        name="llama3-8b", vocab_size=128256, hidden_size=4096, intermediate_size=14336,
        num_hidden_layers=32, num_attention_heads=32, num_key_value_heads=8,
        head_dim=128, rope_theta=500000.0, rms_norm_eps=1e-5, qk_norm=False,
    ),
    # BASELINE config 3 (north star): Qwen3-30B-A3B
    "qwen3-moe-30b": ModelConfig(
        name="qwen3-moe-30b", vocab_size=151936, hidden_size=2048, intermediate_size=6144,
        num_hidden_layers=48, num_attention_heads=32, num_key_value_heads=4,
        head_dim=128, rope_theta=1000000.0, qk_norm=True,
        num_experts=128, num_experts_per_tok=8, moe_intermediate_size=768,
        norm_topk_prob=True, router_aux_loss_coef=0.001, initializer_range=0.02,
    ),
    # toy fixture matching the reference's tests/toy_config/qwen3_moe_toy
    "qwen3-moe-toy": ModelConfig(
        name="qwen3-moe-toy", vocab_size=151936, hidden_size=2048, intermediate_size=6144,
        num_hidden_layers=4, num_attention_heads=32, num_key_value_heads=4,
        head_dim=128, rope_theta=1000000.0, qk_norm=True, initializer_range=0.05,
        num_experts=16, num_experts_per_tok=2, moe_intermediate_size=768,
        norm_topk_prob=True, router_aux_loss_coef=0.001,
    ),
    # small variants for unit tests
    "tiny-dense": ModelConfig(
        name="tiny-dense", vocab_size=512, hidden_size=128, intermediate_size=256,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        head_dim=32, qk_norm=False,
    ),
    # full-width head_dim at test scale: exercises the HIP flash pair
    # (D=128 requirement) without a BASELINE-sized model
    "tiny-d128": ModelConfig(
        name="tiny-d128", vocab_size=512, hidden_size=256, intermediate_size=512,
        num_hidden_layers=2, num_attention_heads=2, num_key_value_heads=1,
        head_dim=128, qk_norm=False,
    ),
    # qwen2-architecture knobs at test scale: attention bias on, no qk norm
    "tiny-qwen2": ModelConfig(
        name="tiny-qwen2", vocab_size=512, hidden_size=128, intermediate_size=256,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        head_dim=32, qk_norm=False, attention_bias=True,
    ),
    "tiny-moe": ModelConfig(
        name="tiny-moe", vocab_size=512, hidden_size=128, intermediate_size=256,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        head_dim=32, qk_norm=True, num_experts=8, num_experts_per_tok=2,
        moe_intermediate_size=64, router_aux_loss_coef=0.001,
    ),
}

# VLM presets (VLConfig: text + vision towers), built via build_vl_model
VL_PRESETS = {}


def _init_vl_presets():
    from .modeling import ModelConfig
    from .vlm import VisionConfig, VLConfig

    if VL_PRESETS:
        return
    # BASELINE config 4: Qwen2.5-VL-7B (text = Qwen2.5-7B w/ attention bias
    # + mrope sections [16,24,24]; vision: 32-block windowed ViT)
    VL_PRESETS["qwen25-vl-7b"] = VLConfig(
        text=ModelConfig(
            name="qwen25-vl-7b-text", vocab_size=152064, hidden_size=3584,
            intermediate_size=18944, num_hidden_layers=28,
            num_attention_heads=28, num_key_value_heads=4, head_dim=128,
            rope_theta=1000000.0, rms_norm_eps=1e-6, attention_bias=True,
            qk_norm=False, mrope_section=(16, 24, 24),
        ),
        vision=VisionConfig(),
        image_token_id=151655,
        name="qwen25-vl-7b",
    )
    # BASELINE config 5: Qwen3-VL-MoE-30B-A3B — the 30B MoE text stack
    # (ref qwen3_vl_moe family: same decoder as qwen3_moe + mrope) under the
    # windowed ViT. Vision tower kept at the Qwen2.5-VL shape (the ref's
    # Qwen3-VL ViT differs in deepstack taps; the compute shape is the same
    # class) — documented deviation, DESIGN.md VLM section.
    VL_PRESETS["qwen3-vl-moe-30b"] = VLConfig(
        text=ModelConfig(
            name="qwen3-vl-moe-30b-text", vocab_size=151936, hidden_size=2048,
            intermediate_size=6144, num_hidden_layers=48,
            num_attention_heads=32, num_key_value_heads=4, head_dim=128,
            rope_theta=1000000.0, qk_norm=True, num_experts=128,
            num_experts_per_tok=8, moe_intermediate_size=768,
            norm_topk_prob=True, router_aux_loss_coef=0.001,
            initializer_range=0.02, mrope_section=(24, 20, 20),
        ),
        vision=VisionConfig(out_hidden_size=2048),
        image_token_id=151655,
        name="qwen3-vl-moe-30b",
    )
    VL_PRESETS["tiny-vl-moe"] = VLConfig(
        text=ModelConfig(
            name="tiny-vl-moe-text", vocab_size=512, hidden_size=128,
            intermediate_size=256, num_hidden_layers=2,
            num_attention_heads=4, num_key_value_heads=2, head_dim=32,
            qk_norm=True, mrope_section=(4, 6, 6), num_experts=8,
            num_experts_per_tok=2, moe_intermediate_size=64,
            norm_topk_prob=True, router_aux_loss_coef=0.001,
        ),
        vision=VisionConfig(
            depth=2, hidden_size=64, num_heads=4, intermediate_size=128,
            out_hidden_size=128, patch_size=2, temporal_patch_size=1,
            in_channels=3, spatial_merge_size=2, window_size=8,
            fullatt_block_indexes=(1,),
        ),
        image_token_id=511,
        name="tiny-vl-moe",
    )
    VL_PRESETS["tiny-vl"] = VLConfig(
        text=ModelConfig(
            name="tiny-vl-text", vocab_size=512, hidden_size=128,
            intermediate_size=256, num_hidden_layers=2,
            num_attention_heads=4, num_key_value_heads=2, head_dim=32,
            attention_bias=True, qk_norm=False, mrope_section=(4, 6, 6),
        ),
        vision=VisionConfig(
            depth=2, hidden_size=64, num_heads=4, intermediate_size=128,
            out_hidden_size=128, patch_size=2, temporal_patch_size=1,
            in_channels=3, spatial_merge_size=2, window_size=8,
            fullatt_block_indexes=(1,),
        ),
        image_token_id=511,
        name="tiny-vl",
    )


def build_vl_model(preset: str, dtype=None, device=None):
    import torch

    from .vlm import VLForCausalLM

    _init_vl_presets()
    cfg = VL_PRESETS[preset]
    with torch.device(device or "cpu"):
        model = VLForCausalLM(cfg)
    if dtype is not None:
        model = model.to(dtype)
    return model


def build_model(preset: str, dtype=None, device=None, empty_init=False) -> ForCausalLM:
    import torch

    cfg = PRESETS[preset]
    if empty_init:
        # timing-only instantiation (bench cpu_baseline): meta-build +
        # to_empty skips the (minutes-long on host, for 30B) random init;
        # weights are filled with a small constant so the step is
        # numerically tame but costs exactly the same FLOPs.
        with torch.device("meta"):
            model = ForCausalLM(cfg)
        model = model.to_empty(device=device or "cpu")
        with torch.no_grad():
            for p in model.parameters():
                p.fill_(0.01)
            for b in model.buffers():
                if b.is_floating_point():
                    b.fill_(0.01)
        if dtype is not None:
            model = model.to(dtype)
        return model
    with torch.device(device) if device is not None else torch.device("cpu"):
        model = ForCausalLM(cfg)
    if dtype is not None:
        model = model.to(dtype)
    return model
