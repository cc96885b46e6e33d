"""One llama train step under torch.profiler: top CUDA kernels with op names."""
import sys, os
sys.path.insert(0, "/root/repo")
import torch
from veomni_amd.ops import HIP_OPS_CONFIG
from veomni_amd.distributed.fsdp2 import build_parallelize_model
from veomni_amd.distributed.parallel_state import init_parallel_state
from veomni_amd.models import build_model
from veomni_amd.models.modeling import bind_ops
from veomni_amd.data import synthetic_batch

init_parallel_state(ep_size=1, device_type="cuda")
bind_ops(HIP_OPS_CONFIG)
import sys as _s
preset = _s.argv[1] if len(_s.argv) > 1 else "llama3-8b"
model = build_model(preset, dtype=torch.bfloat16, device="cuda")
model = build_parallelize_model(model)
opt = torch.optim.AdamW(model.parameters(), lr=1e-5, fused=True)
batch = synthetic_batch(model.config.vocab_size, 4096, batch=(4 if preset=="llama3-8b" else 1), seed=42, device="cuda")
model.use_checkpoint = model.config.is_moe
def step():
    loss, _ = model(**batch)
    loss.backward()
    model.clip_grad_norm_(1.0)
    opt.step()
    opt.zero_grad(set_to_none=True)
for _ in range(3):
    step()
torch.cuda.synchronize()
from torch.profiler import profile, ProfilerActivity
with profile(activities=[ProfilerActivity.CPU, ProfilerActivity.CUDA], record_shapes=True) as prof:
    step()
    torch.cuda.synchronize()
print(prof.key_averages(group_by_input_shape=True).table(sort_by="self_cuda_time_total", row_limit=30, max_name_column_width=60))
