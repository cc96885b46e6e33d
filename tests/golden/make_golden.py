#!/usr/bin/env python3
"""Generate golden parity vectors by RUNNING THE REFERENCE (build container only).

Imports ByteDance-Seed/VeOmni from /root/reference (read-only mount, present
only in the build container — never on the GPU box) and executes its own
eager CPU code paths on seeded inputs. The outputs are committed as small
fixtures (golden.pt) so parity tests never need the reference at run time.

Usage:  python tests/golden/make_golden.py          # writes tests/golden/golden.pt

Every entry records the reference call site it exercises.
"""

import os
import sys

import torch

REF = "/root/reference"
assert os.path.isdir(REF), "reference repo required (build container only)"
sys.path.insert(0, REF)

torch.manual_seed(0)

# No GPU in the build container; the reference's pretuned-config path probes
# torch.cuda.get_device_capability() at import time (moe/_kernels/utils/
# device.py:24). Patch it to a fixed value — only a config-file path prefix
# depends on it, and we never launch the Triton kernels here.
torch.cuda.get_device_capability = lambda *a, **k: (8, 0)
torch.cuda.get_device_name = lambda *a, **k: "cpu-golden"
torch.cpu.get_device_name = lambda *a, **k: "cpu-golden"

golden = {}

# ── scatter index (bit-exact) ─ ref: veomni/ops/kernels/moe/_scatter.py:40 ──
from veomni.ops.kernels.moe._scatter import compute_expert_scatter_index  # noqa: E402

for name, (T, topk, E) in {
    "small": (64, 8, 16),
    "tiny": (3, 2, 4),
    "one_expert": (32, 2, 5),
    "full": (256, 8, 128),
}.items():
    g = torch.Generator().manual_seed(hash(name) % (2**31))
    if name == "one_expert":
        idx = torch.full((T, topk), 3, dtype=torch.int64)
    else:
        idx = torch.randint(0, E, (T, topk), generator=g)
    sorted_order, scatter_index = compute_expert_scatter_index(idx)
    golden[f"scatter/{name}/expert_index"] = idx
    golden[f"scatter/{name}/sorted_order"] = sorted_order
    golden[f"scatter/{name}/scatter_index"] = scatter_index
    golden[f"scatter/{name}/histogram"] = torch.bincount(idx.flatten(), minlength=E).to(torch.int32)
    golden[f"scatter/{name}/num_experts"] = torch.tensor(E)

# ── eager MoE experts loop ─ ref: patched_modeling_qwen3_moe_gpu.py:254-294 ──
from veomni.models.transformers.qwen3_moe.generated import (  # noqa: E402
    patched_modeling_qwen3_moe_gpu as m,
)


class _Cfg:
    num_experts = 8
    hidden_size = 64
    moe_intermediate_size = 48
    hidden_act = "silu"
    num_experts_per_tok = 2
    norm_topk_prob = True
    rms_norm_eps = 1e-6


torch.manual_seed(1)
experts = m.Qwen3MoeExperts(_Cfg())
with torch.no_grad():
    experts.gate_up_proj.normal_(0, 0.05)
    experts.down_proj.normal_(0, 0.05)
T = 96
hidden = torch.randn(T, _Cfg.hidden_size) * 0.5
router = m.Qwen3MoeTopKRouter(_Cfg())
with torch.no_grad():
    router.weight.normal_(0, 0.05)

router_logits, top_w, top_i = router(hidden)
golden["moe/hidden"] = hidden
golden["moe/gate_up_proj"] = experts.gate_up_proj.detach().clone()
golden["moe/down_proj"] = experts.down_proj.detach().clone()
golden["moe/router_weight"] = router.weight.detach().clone()
golden["moe/router_logits"] = router_logits.detach()
golden["moe/top_w"] = top_w.detach()
golden["moe/top_i"] = top_i.detach()

hidden_g = hidden.clone().requires_grad_(True)
experts.gate_up_proj.requires_grad_(True)
experts.down_proj.requires_grad_(True)
out = experts(hidden_g, top_i, top_w.detach())
dy = torch.randn_like(out) * 0.1
torch.manual_seed(2)
dy = torch.randn_like(out) * 0.1
out.backward(dy)
golden["moe/out"] = out.detach()
golden["moe/dy"] = dy
golden["moe/dhidden"] = hidden_g.grad.detach().clone()
golden["moe/dgate_up"] = experts.gate_up_proj.grad.detach().clone()
golden["moe/ddown"] = experts.down_proj.grad.detach().clone()

# bf16 variant of the same eager loop (the dtype the HIP path runs in)
torch.manual_seed(3)
experts_bf = m.Qwen3MoeExperts(_Cfg())
with torch.no_grad():
    experts_bf.gate_up_proj.copy_(experts.gate_up_proj.detach())
    experts_bf.down_proj.copy_(experts.down_proj.detach())
experts_bf = experts_bf.to(torch.bfloat16)
out_bf = experts_bf(hidden.to(torch.bfloat16), top_i, top_w.detach().to(torch.bfloat16))
golden["moe/out_bf16"] = out_bf.detach()

# ── RMSNorm ─ ref: patched_modeling_qwen3_moe_gpu.py Qwen3MoeRMSNorm ──
torch.manual_seed(4)
norm = m.Qwen3MoeRMSNorm(64, eps=1e-6)
with torch.no_grad():
    norm.weight.normal_(1.0, 0.1)
x = torch.randn(32, 64, requires_grad=True)
y = norm(x)
dy = torch.randn_like(y)
y.backward(dy)
golden["rmsnorm/x"] = x.detach().clone()
golden["rmsnorm/w"] = norm.weight.detach().clone()
golden["rmsnorm/y"] = y.detach()
golden["rmsnorm/dy"] = dy
golden["rmsnorm/dx"] = x.grad.detach().clone()
golden["rmsnorm/dw"] = norm.weight.grad.detach().clone()

# ── RoPE ─ ref: patched_modeling_qwen3_moe_gpu.py:94-111 ──
torch.manual_seed(5)
B, h, S, D = 1, 4, 16, 32
q = torch.randn(B, h, S, D)
k = torch.randn(B, 2, S, D)
inv = 1.0 / (10000.0 ** (torch.arange(0, D, 2, dtype=torch.float32) / D))
ang = torch.outer(torch.arange(S, dtype=torch.float32), inv)
emb = torch.cat((ang, ang), -1)
cos, sin = emb.cos()[None], emb.sin()[None]
qe, ke = m.apply_rotary_pos_emb(q, k, cos, sin)
for n, t in [("q", q), ("k", k), ("cos", cos), ("sin", sin), ("qe", qe), ("ke", ke)]:
    golden[f"rope/{n}"] = t

# ── cross entropy ─ ref: ops/kernels/cross_entropy/eager.py:22-36 + chunk_loss.py ──
from veomni.ops.kernels.cross_entropy.eager import eager_cross_entropy  # noqa: E402
from veomni.ops.kernels.cross_entropy.chunk_loss import chunk_loss_function  # noqa: E402

torch.manual_seed(6)
Tl, H, V = 40, 32, 97
hs = torch.randn(1, Tl, H) * 0.5
w = torch.randn(V, H) * 0.05
labels = torch.randint(0, V, (1, Tl))
labels[0, 5] = -100
hs_g = hs.clone().requires_grad_(True)
w_g = w.clone().requires_grad_(True)
loss, _ = chunk_loss_function(hs_g, w_g, labels, chunk_size=16)
loss.backward()
golden["ce/hs"] = hs
golden["ce/w"] = w
golden["ce/labels"] = labels
golden["ce/loss"] = loss.detach()
golden["ce/dhs"] = hs_g.grad.detach().clone()
golden["ce/dw"] = w_g.grad.detach().clone()

logits = torch.randn(64, V)
lab2 = torch.randint(0, V, (64,))
lab2[3] = -100
l2, _ = eager_cross_entropy(logits.clone(), lab2, V, (lab2 != -100).sum(), -100)
golden["ce/logits"] = logits
golden["ce/labels2"] = lab2
golden["ce/loss2"] = l2.detach()

# ── load balancing loss ─ ref: load_balancing_loss/eager.py:28-114 ──
from veomni.ops.kernels.load_balancing_loss.eager import load_balancing_loss_pytorch  # noqa: E402

torch.manual_seed(7)
gl = tuple(torch.randn(48, 8) for _ in range(3))
golden["lbl/gate_logits_0"] = gl[0]
golden["lbl/gate_logits_1"] = gl[1]
golden["lbl/gate_logits_2"] = gl[2]
golden["lbl/loss"] = load_balancing_loss_pytorch(gl, 8, 2)
am = torch.ones(4, 12, dtype=torch.int64)
am[:, -3:] = 0
golden["lbl/mask"] = am
golden["lbl/loss_masked"] = load_balancing_loss_pytorch(gl, 8, 2, am)

# ── SP collator ─ ref: data/data_collator.py:317-427 ──
# torchdata is not installed in this container; it is only imported by the
# reference's data_loader (off-path). Stub it so veomni.data imports.
import types  # noqa: E402

if "torchdata" not in sys.modules:
    from torch.utils.data import DataLoader as _DL  # noqa: E402
    from torch.utils.data.distributed import DistributedSampler as _DS  # noqa: E402

    td = types.ModuleType("torchdata")
    sd = types.ModuleType("torchdata.stateful_dataloader")
    sd.__path__ = []  # mark as package so submodule imports resolve
    samp = types.ModuleType("torchdata.stateful_dataloader.sampler")
    sd.StatefulDataLoader = _DL
    samp.StatefulDistributedSampler = _DS
    td.stateful_dataloader = sd
    sys.modules["torchdata"] = td
    sys.modules["torchdata.stateful_dataloader"] = sd
    sys.modules["torchdata.stateful_dataloader.sampler"] = samp

# Bypass veomni/data/__init__.py (pulls torchvision etc. — off-path): register
# a stub parent package pointing at the real directory, then import the
# collator module directly.
if "veomni.data" not in sys.modules:
    pkg = types.ModuleType("veomni.data")
    pkg.__path__ = [os.path.join(REF, "veomni", "data")]
    sys.modules["veomni.data"] = pkg
import importlib  # noqa: E402

SequenceParallelCollator = importlib.import_module(
    "veomni.data.data_collator"
).SequenceParallelCollator

torch.manual_seed(8)
for sp_size in (2, 4):
    for rank in range(sp_size):
        col = object.__new__(SequenceParallelCollator)
        col.sp_size = sp_size
        col.sp_rank = rank
        ids = torch.randint(0, 1000, (1, 30))
        lab = ids.clone()
        shifted = torch.nn.functional.pad(lab[..., 1:], (0, 1), "constant", -100)
        padded = col.sp_padding("labels", shifted, dim=-1, pad_value=-100)
        sliced = col.sp_slice("labels", padded, dim=-1)
        ids_p = col.sp_padding("input_ids", ids, dim=-1, pad_value=0)
        ids_s = col.sp_slice("input_ids", ids_p, dim=-1)
        golden[f"spcol/{sp_size}/{rank}/ids"] = ids
        golden[f"spcol/{sp_size}/{rank}/labels_out"] = sliced
        golden[f"spcol/{sp_size}/{rank}/ids_out"] = ids_s

# ── Packing collator ─ ref: data/data_collator.py:219-317 ──
_dc = importlib.import_module("veomni.data.data_collator")
PackingCollator = _dc.PackingCollator
DEFAULT_INFO = _dc.DEFAULT_DATA_COLLATE_INFO

def _mk_pack_features(lens, seed):
    g = torch.Generator().manual_seed(seed)
    feats = []
    for L in lens:
        ids = torch.randint(0, 500, (L,), generator=g)
        feats.append({
            "input_ids": ids,
            "labels": ids.clone(),
            "attention_mask": torch.ones(L, dtype=torch.int64),
            "position_ids": torch.arange(L),
        })
    return feats

def _mk_vlm_features(lens, seed):
    g = torch.Generator().manual_seed(seed)
    feats = []
    for L in lens:
        ids = torch.randint(0, 500, (L,), generator=g)
        npatch = int(torch.randint(2, 6, (1,), generator=g))
        feats.append({
            "input_ids": ids,
            "labels": ids.clone(),
            "attention_mask": torch.ones(L, dtype=torch.int64),
            "position_ids": torch.arange(L),
            "pixel_values": torch.randn(npatch, 8, generator=g),
            "image_grid_thw": torch.tensor([[1, npatch, 1]]),
        })
    return feats

col = object.__new__(PackingCollator)
col.collate_infos = DEFAULT_INFO.copy()
col.pad_to_length = False
col.seq_classification = False
col.metadata_collate_func = None
col.sp_enabled = True  # skip fa-kwargs (needs full veomni utils on load side)
vout = col(_mk_vlm_features([4, 6], seed=21))
for k in ("input_ids", "pixel_values", "image_grid_thw"):
    golden[f"packvlm/{k}"] = vout[k]

for name, lens, pad_to in (("plain", [5, 7, 3], None), ("padded", [5, 7, 3], 32),
                           ("single", [9], 16)):
    col = object.__new__(PackingCollator)
    col.collate_infos = DEFAULT_INFO.copy()
    col.pad_to_length = pad_to if pad_to else False
    col.seq_classification = False
    col.metadata_collate_func = None
    col.sp_enabled = False
    out = col(_mk_pack_features(lens, seed=11))
    for k in ("input_ids", "labels", "attention_mask", "position_ids",
              "cu_seq_lens_q"):
        golden[f"pack/{name}/{k}"] = out[k]
    golden[f"pack/{name}/max_length_q"] = torch.tensor(out["max_length_q"])
    if "tail_padding_length" in out:
        golden[f"pack/{name}/tail"] = out["tail_padding_length"]

# ── EP permute/unpermute ─ ref: distributed/moe/moe_utils.py:19-99 ──
from veomni.distributed.moe.moe_utils import (  # noqa: E402
    generate_weights_idx,
    permute,
    sort_chunks_by_idxs,
    unpermute,
)

torch.manual_seed(9)
Tt, Ht, Et, kk = 24, 16, 6, 2
toks = torch.randn(Tt, Ht)
sel = torch.randint(0, Et, (Tt, kk))
# ensure distinct experts per token (one_hot sum semantics)
sel[:, 1] = (sel[:, 0] + 1 + sel[:, 1] % (Et - 1)) % Et
rw = torch.softmax(torch.randn(Tt, kk), -1)
mask = torch.nn.functional.one_hot(sel, num_classes=Et).permute(2, 1, 0)
routing_map = mask.sum(dim=1)
perm, sorted_idx = permute(toks, routing_map)
widx = generate_weights_idx(rw, sel, Et)
unperm = unpermute(perm.clone(), widx, toks.shape, sorted_idx, routing_map)
golden["ep/tokens"] = toks
golden["ep/sel"] = sel
golden["ep/rw"] = rw
golden["ep/perm"] = perm
golden["ep/sorted_idx"] = sorted_idx
golden["ep/weights_idx"] = widx
golden["ep/unperm"] = unperm
split_sizes = torch.tensor([4, 2, 6, 12])
order = [2, 0, 3, 1]
golden["ep/sorted_chunks"] = sort_chunks_by_idxs(toks, split_sizes, order)
golden["ep/sorted_chunks_sizes"] = split_sizes
golden["ep/sorted_chunks_order"] = torch.tensor(order)

# ── tiny end-to-end reference model fwd+bwd ─ ref: patched_modeling_qwen3_moe
# _gpu.py (full model), loss via install_loss_mapping("chunk_loss") — the
# reference's default CE backend (cross_entropy/__init__.py:445-516).
from transformers.models.qwen3_moe.configuration_qwen3_moe import Qwen3MoeConfig  # noqa: E402
from veomni.ops.kernels.cross_entropy import install_loss_mapping  # noqa: E402

install_loss_mapping("chunk_loss")
tiny_cfg = dict(
    vocab_size=512, hidden_size=128, intermediate_size=256, num_hidden_layers=2,
    num_attention_heads=4, num_key_value_heads=2, head_dim=32, num_experts=8,
    num_experts_per_tok=2, moe_intermediate_size=64, norm_topk_prob=True,
    rope_theta=1000000.0, rms_norm_eps=1e-6, tie_word_embeddings=False,
    attention_bias=False, output_router_logits=False,
)
torch.manual_seed(21)
ref_model = m.Qwen3MoeForCausalLM(Qwen3MoeConfig(**tiny_cfg, attn_implementation="sdpa"))
with torch.no_grad():
    for name, p in sorted(ref_model.named_parameters(), key=lambda kv: kv[0]):
        gseed = torch.Generator().manual_seed(abs(hash(name)) % (2**31))
        if "norm" in name and name.endswith("weight") and p.dim() == 1:
            p.fill_(1.0)
        else:
            p.copy_(torch.randn(p.shape, generator=gseed) * 0.03)
ids = torch.randint(0, 512, (1, 64), generator=torch.Generator().manual_seed(99))
out = ref_model(input_ids=ids, labels=ids.clone())
out.loss.backward()
golden["e2e/config"] = tiny_cfg
golden["e2e/input_ids"] = ids
golden["e2e/loss"] = out.loss.detach()
golden["e2e/state_dict"] = {k: v.detach().clone() for k, v in ref_model.state_dict().items()}
golden["e2e/grads"] = {
    n: p.grad.detach().clone() for n, p in ref_model.named_parameters() if p.grad is not None
}
with torch.no_grad():
    logits = ref_model(input_ids=ids).logits
golden["e2e/logits"] = logits.detach()

# ── multi-step training trace (loss trajectory incl. optimizer + clip) ──
train_model = m.Qwen3MoeForCausalLM(Qwen3MoeConfig(**tiny_cfg, attn_implementation="sdpa"))
with torch.no_grad():
    for name, p in sorted(train_model.named_parameters(), key=lambda kv: kv[0]):
        gseed = torch.Generator().manual_seed(abs(hash(name)) % (2**31))
        if "norm" in name and name.endswith("weight") and p.dim() == 1:
            p.fill_(1.0)
        else:
            p.copy_(torch.randn(p.shape, generator=gseed) * 0.03)
opt = torch.optim.AdamW(train_model.parameters(), lr=1e-3, betas=(0.9, 0.95),
                        eps=1e-8, weight_decay=0.01)
step_losses = []
for st in range(3):
    sids = torch.randint(0, 512, (1, 64), generator=torch.Generator().manual_seed(1000 + st))
    out2 = train_model(input_ids=sids, labels=sids.clone())
    out2.loss.backward()
    torch.nn.utils.clip_grad_norm_(train_model.parameters(), 1.0)
    opt.step()
    opt.zero_grad(set_to_none=True)
    step_losses.append(out2.loss.detach())
golden["train3/losses"] = torch.stack(step_losses)

out_path = os.path.join(os.path.dirname(__file__), "golden.pt")
torch.save(golden, out_path)
print(f"wrote {out_path} with {len(golden)} entries")
