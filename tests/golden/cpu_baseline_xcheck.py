"""Cross-check: our eager host-mirror ("port") vs the REFERENCE's own model
stepping the same config/batch on CPU — validates that bench.py's
cpu_baseline (kind="port", the host-mirror eager step — the reference tree
does not exist on the GPU box) is a fair speed proxy for the reference's
own CPU path. Run in the build container. Measured 2026-09: ratio
port/reference = 0.87 (the port is ~13% FASTER, i.e. the reported CPU
baseline slightly overstates the reference's CPU speed — conservative for
the GPU/CPU comparison)."""
import os, sys, time
sys.path.insert(0, "/root/repo")
import torch

REF = "/root/reference"
sys.path.insert(0, REF)
torch.cuda.get_device_capability = lambda *a, **k: (8, 0)
torch.cuda.get_device_name = lambda *a, **k: "cpu-x"
torch.cpu.get_device_name = lambda *a, **k: "cpu-x"
torch.manual_seed(0)

CFG = dict(vocab_size=512, hidden_size=256, intermediate_size=512,
           moe_intermediate_size=192, num_hidden_layers=4,
           num_attention_heads=8, num_key_value_heads=4, head_dim=32,
           num_experts=16, num_experts_per_tok=4)
SEQ = 512
STEPS = 5

def bench(step_fn):
    step_fn()  # warmup
    t0 = time.time()
    for _ in range(STEPS):
        step_fn()
    return (time.time() - t0) / STEPS

ids = torch.randint(0, 512, (1, SEQ))
pos = torch.arange(SEQ)[None]

# ---- ours (the bench cpu_baseline "port")
from veomni_amd.models.modeling import ModelConfig, ForCausalLM, bind_ops
bind_ops("eager")
ours = ForCausalLM(ModelConfig(**CFG, qk_norm=True)).to(torch.bfloat16)
opt1 = torch.optim.AdamW(ours.parameters(), lr=1e-5)
def step_ours():
    loss, _ = ours(input_ids=ids, labels=ids.clone(), position_ids=pos)
    loss.backward(); opt1.step(); opt1.zero_grad(set_to_none=True)
dt_ours = bench(step_ours)
print(f"port  (veomni_amd eager bf16): {dt_ours*1e3:7.1f} ms/step")

# ---- reference model, eager ops, same config
from veomni.models.transformers.qwen3_moe.generated import (
    patched_modeling_qwen3_moe_gpu as m,
)
from veomni.ops.kernels.cross_entropy import install_loss_mapping
install_loss_mapping("chunk_loss")
from transformers.models.qwen3_moe.configuration_qwen3_moe import Qwen3MoeConfig
rcfg = Qwen3MoeConfig(
    vocab_size=512, hidden_size=256, intermediate_size=512,
    moe_intermediate_size=192, num_hidden_layers=4, num_attention_heads=8,
    num_key_value_heads=4, head_dim=32, num_experts=16, num_experts_per_tok=4,
    attn_implementation="sdpa", output_router_logits=False,
)
refm = m.Qwen3MoeForCausalLM(rcfg).to(torch.bfloat16)
opt2 = torch.optim.AdamW(refm.parameters(), lr=1e-5)
def step_ref():
    out = refm(input_ids=ids, labels=ids.clone())
    out.loss.backward(); opt2.step(); opt2.zero_grad(set_to_none=True)
dt_ref = bench(step_ref)
print(f"reference (Qwen3MoeForCausalLM eager bf16): {dt_ref*1e3:7.1f} ms/step")
print(f"ratio port/reference: {dt_ours/dt_ref:.2f}")
