"""CPU-checkable pieces of the bench contract: the analytic FLOP/param model
(the MFU denominator) against known model-card figures, and preset sanity."""

from veomni_amd.models import PRESETS

import bench


def test_active_params_known_values():
    # Qwen3-30B-A3B: ~3B activated parameters (the "A3B" in the name);
    # our count follows the reference count_flops.py convention (lm_head
    # included, input embedding lookup excluded).
    a = bench.active_params(PRESETS["qwen3-moe-30b"])
    assert 2.9e9 < a < 3.4e9, a
    # Llama-3-8B: 8.03B total, ~7.5B compute-active under the 6PT convention
    b = bench.active_params(PRESETS["llama3-8b"])
    assert 7.3e9 < b < 8.1e9, b


def test_step_flops_scales_linearly_in_tokens():
    cfg = PRESETS["qwen3-moe-30b"]
    f1 = bench.step_flops(cfg, 4096, 4096)
    f2 = bench.step_flops(cfg, 8192, 4096)
    assert abs(f2 / f1 - 2.0) < 1e-6


def test_moe_preset_matches_qwen3_30b_a3b_card():
    c = PRESETS["qwen3-moe-30b"]
    assert (c.num_hidden_layers, c.hidden_size, c.num_experts,
            c.num_experts_per_tok, c.moe_intermediate_size,
            c.num_attention_heads, c.num_key_value_heads,
            c.vocab_size) == (48, 2048, 128, 8, 768, 32, 4, 151936)
