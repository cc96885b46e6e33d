#!/usr/bin/env python3
"""Training-step benchmark (driver contract — see repo prompt).

Default (no flags): Qwen3-MoE-30B-A3B FSDP2 seq4096 — the config BASELINE's
metric is quoted on ("tokens/sec/node + step MFU, Qwen3-MoE-30B FSDP2
seq4096 at 1/2/4/8 MI355X"). It fits one 288 GB GPU (~254 GiB peak: bf16
params/grads/optimizer states + checkpointed activations), so the same
workload runs at every N (EP = N for N >= 2) and the driver's weak-scaling
efficiency is computed over a consistent series. configs[1] (Llama-3-8B
dense) stays available via --model llama3-8b. A step = one
forward+backward+clip+optimizer over one synthetic packed seq-4096 batch
per rank (weak scaling).

Rank 0 prints ONE JSON line with metric/value/roofline/cpu_baseline.
"""

import argparse
import json
import os
import sys
import time

REPO = os.path.dirname(os.path.abspath(__file__))
if REPO not in sys.path:
    sys.path.insert(0, REPO)

import torch
import torch.distributed as dist


def log(msg):
    if int(os.environ.get("RANK", "0")) == 0:
        print(f"[bench] {msg}", file=sys.stderr, flush=True)


def active_params(cfg):
    """Dense-equivalent active params/token (reference count_flops.py:123-155)."""
    H, L = cfg.hidden_size, cfg.num_hidden_layers
    V = cfg.vocab_size
    hq, hkv, D = cfg.num_attention_heads, cfg.num_key_value_heads, cfg.head_dim
    attn = H * (hq * D) + 2 * H * (hkv * D) + (hq * D) * H
    if cfg.is_moe:
        mlp = 3 * H * cfg.moe_intermediate_size * cfg.num_experts_per_tok
        router = H * cfg.num_experts
    else:
        mlp = 3 * H * cfg.intermediate_size
        router = 0
    emb = V * H  # lm_head
    return L * (attn + mlp + router) + emb


def step_flops(cfg, tokens, seq_len):
    """Analytic fwd+bwd FLOPs per step (factor 6·P·T + attention term,
    reference count_flops.py:123-155,502-528; causal 1/2)."""
    p = active_params(cfg)
    attn = (
        cfg.num_hidden_layers
        * 2 * 2 * seq_len * cfg.num_attention_heads * cfg.head_dim * 0.5
    ) * 3  # qk^T + pv, fwd+2x bwd
    return 6.0 * p * tokens + attn * tokens


def run_cpu_baseline(preset, seq_len, budget_s=15.0):
    """Oracle (host-mirror, eager ops) timed on host cores on a BOUNDED
    sample of the same workload. kind="port" (DESIGN.md §c/§d)."""
    from veomni_amd.models import build_model
    from veomni_amd.models.modeling import bind_ops
    from veomni_amd.data import synthetic_batch

    bind_ops("eager")
    from veomni_amd.models import PRESETS
    # MoE eager on host loops experts per layer — bound the sample tighter
    sample_seq = min(seq_len, 256 if PRESETS[preset].is_moe else 512)
    model = build_model(preset, dtype=torch.bfloat16, device="cpu", empty_init=True)
    model.use_checkpoint = False
    opt = torch.optim.AdamW(model.parameters(), lr=1e-5)
    batch = synthetic_batch(model.config.vocab_size, sample_seq, seed=42, device="cpu")
    t0 = time.time()
    steps = 0
    while time.time() - t0 < budget_s and steps < 8:
        loss, _ = model(**batch)
        loss.backward()
        opt.step()
        opt.zero_grad(set_to_none=True)
        steps += 1
    dt = time.time() - t0
    toks = steps * sample_seq / dt
    return {
        "value": round(toks, 3),
        "unit": "tokens/s",
        "cores": torch.get_num_threads(),
        "kind": "port",
        "sample": f"{steps} steps of {preset} seq{sample_seq} bf16 eager on host ({dt:.1f}s)",
    }


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=8)
    ap.add_argument("--warmup", type=int, default=3)
    ap.add_argument("--model", type=str, default=None)
    ap.add_argument("--seq-len", type=int, default=4096)
    ap.add_argument("--batch", type=int, default=0, help="micro-batch per rank (0 = auto)")
    ap.add_argument("--no-cpu-baseline", action="store_true")
    ap.add_argument("--sp", type=int, default=1,
                    help="Ulysses sequence-parallel degree (divides --gpus)")
    ap.add_argument("--checkpoint", choices=["auto", "on", "off"],
                    default="auto", help="activation checkpointing override")
    ap.add_argument("--sync-ulysses", action="store_true",
                    help="use the synchronous a2a path (A/B for the async overlap)")
    args = ap.parse_args()

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", str(args.gpus)))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    n_gpus = max(world, 1)

    assert torch.cuda.is_available(), "bench needs an MI355X"
    torch.cuda.set_device(local_rank)

    if world > 1:
        dist.init_process_group("nccl")
    # world == 1: no process group at all — gloo's C++ banner pollutes
    # stdout and would break the single-JSON-line contract.

    import veomni_amd.ops  # registrations
    from veomni_amd.ops import HIP_OPS_CONFIG, hip_lib
    from veomni_amd.distributed.loss_utils import mean_global_loss
    from veomni_amd.distributed.fsdp2 import build_parallelize_model
    from veomni_amd.distributed.parallel_state import init_parallel_state
    from veomni_amd.models import PRESETS, build_model
    from veomni_amd.models.modeling import bind_ops
    from veomni_amd.data import synthetic_batch

    preset = args.model or "qwen3-moe-30b"
    from veomni_amd.models import VL_PRESETS, _init_vl_presets
    _init_vl_presets()
    is_vl = preset in VL_PRESETS
    vl_cfg = VL_PRESETS.get(preset)
    cfg = vl_cfg.text if is_vl else PRESETS[preset]
    # auto micro-batch, sized for 288 GB HBM at N=1 (measured peaks):
    # llama-8b 4x4096 tokens without checkpointing; the 30B MoE 8x4096 with
    # checkpointing (236 GiB peak incl. bf16 params/grads/optimizer states —
    # larger expert GEMM groups lift the grouped-GEMM rate ~35%).
    # VL dense: mbs 4 measured 9,797 vs 7,826 tok/s at mbs 1 (155 GiB peak);
    # VL-MoE keeps mbs 1 (30B text states + ViT activations at seq 8192)
    mbs = args.batch or ((1 if cfg.is_moe else 4) if is_vl
                         else (8 if cfg.is_moe else 4))
    if args.sp > 1:
        mbs = 1  # SP rides the packed B==1 path
    if is_vl and args.seq_len == 4096 and args.model:
        # BASELINE config 4 default: image-text seq 8192
        args.seq_len = 8192
    sp = max(args.sp, 1)
    assert n_gpus % sp == 0, "--sp must divide --gpus"
    ep_size = n_gpus if (cfg.is_moe and n_gpus > 1 and sp == 1) else 1
    init_parallel_state(ep_size=ep_size, ulysses_size=sp,
                        async_ulysses=sp > 1 and not args.sync_ulysses,
                        device_type="cuda")
    bind_ops(HIP_OPS_CONFIG)

    log(f"building {preset} on cuda:{local_rank} (ep={ep_size}, ws={world})")
    t_build = time.time()
    if is_vl:
        from veomni_amd.models import build_vl_model

        model = build_vl_model(preset, dtype=torch.bfloat16, device="cuda")
    else:
        model = build_model(preset, dtype=torch.bfloat16, device="cuda")
    # 288 GB HBM3E: dense llama-8b at N=1 holds full activations comfortably
    # (no forward recompute). The 30B MoE keeps checkpointing (scattered
    # expert activations are ~1 GB/layer/rank).
    model.use_checkpoint = (cfg.is_moe or is_vl) if args.checkpoint == "auto" \
        else args.checkpoint == "on"
    model = build_parallelize_model(model)
    def make_opt(kind):
        if kind == "ve":
            from veomni_amd.optim import VeAdamW
            return VeAdamW(model.parameters(), lr=1e-5, betas=(0.9, 0.95))
        return torch.optim.AdamW(model.parameters(), lr=1e-5, betas=(0.9, 0.95),
                                 fused=kind == "fused",
                                 foreach=None if kind == "fused" else True)

    fused_ok = True
    opt = None
    for kind in ("ve", "fused", "foreach"):
        try:
            opt = make_opt(kind)
            for p in model.parameters():
                if p.requires_grad:
                    p.grad = torch.zeros_like(p)
            opt.grad_scale = torch.ones((), dtype=torch.float32, device="cuda")
            opt.step()
            opt.zero_grad(set_to_none=True)
            fused_ok = kind in ("ve", "fused")  # these honor grad_scale
            log(f"optimizer: {kind}")
            break
        except Exception as e:
            log(f"{kind} AdamW unavailable ({e})")
            opt = None
    if opt is None:
        opt = make_opt("foreach")
        fused_ok = False
    log(f"built in {time.time() - t_build:.1f}s; mem {torch.cuda.memory_allocated()/2**30:.1f} GiB")
    torch.cuda.reset_peak_memory_stats()

    seq = args.seq_len
    if is_vl:
        from veomni_amd.data import synthetic_vlm_batch

        batch = synthetic_vlm_batch(vl_cfg, seq, batch=mbs, seed=42 + rank,
                                    device="cuda")
    else:
        # sp-group peers must hold the SAME sequence (each rank keeps a
        # 1/sp slice), so seed by dp-rank
        batch = synthetic_batch(cfg.vocab_size, seq, batch=mbs,
                                seed=42 + rank // sp, device="cuda")
        if sp > 1:
            from veomni_amd.data import sp_collate
            assert mbs == 1, "SP bench path is the packed B==1 path"
            batch = sp_collate(batch)

    n_valid = (batch["labels"] != -100).sum()

    def one_step():
        loss, _ = model(**batch)
        # reference trainer semantics (loss_utils.py:54-96): global per-token
        # mean; a no-op at equal tokens/rank but keeps the parity surface hot
        loss = mean_global_loss(loss, n_valid)
        loss.backward()
        # fused path folds the clip coefficient into AdamW's grad_scale
        model.clip_grad_norm_(1.0, fused_optimizer=opt if fused_ok else None)
        opt.step()
        opt.zero_grad(set_to_none=True)
        return loss

    for i in range(args.warmup):
        loss = one_step()
        log(f"warmup {i}: loss {float(loss.detach()):.4f}")

    # profile one extra (untimed) step for the per-kernel roofline leg
    hip_lib.profile_enable(True)
    one_step()
    prof = hip_lib.profile_summary()
    hip_lib.profile_enable(False)

    log(f"peak mem after warmup: {torch.cuda.max_memory_allocated()/2**30:.1f} GiB")
    if world > 1:
        dist.barrier()
    torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(args.steps):
        one_step()
    torch.cuda.synchronize()
    if world > 1:
        dist.barrier()
    dt = time.time() - t0

    # max over ranks
    if world > 1:
        t = torch.tensor([dt], device="cuda")
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        dt = float(t)

    tokens_per_step = seq * mbs * n_gpus // sp
    toks_per_s = tokens_per_step * args.steps / dt
    ms_per_step = dt / args.steps * 1000.0

    flops = step_flops(cfg, seq, seq) * mbs * n_gpus // sp  # per step whole job
    if is_vl:
        # vision tower contribution: 6 * P_vis * n_patches (dense-equivalent;
        # window-attention quadratic terms are small at these grids)
        p_vis = sum(p.numel() for p in model.visual.parameters())
        n_patches = int(batch["pixel_values"].shape[0])
        flops += 6.0 * p_vis * n_patches * n_gpus
    mfu = flops * args.steps / dt / (n_gpus * 2.5e15)

    # roofline: dominant hand-written kernel
    roofline = None
    if "group_gemm_nk" in prof:
        rec = prof["group_gemm_nk"]
        fl = sum(rec["work"]) / rec["count"]
        ach = fl / (rec["ms_avg"] / 1000.0) / 1e12
        # HBM bytes/launch for the dominant (fc1 mbs8) launch shape, from the
        # committed rocprofv3 TCC passes at that exact shape
        # (profiles/r02_gg_traffic.txt: FETCH_SIZE x2 gfx950 correction +
        # WRITE_SIZE; separate --pmc runs per the slot limits). Only reported
        # for the default 30B workload the passes were collected on.
        traffic = 3.761e9 if (preset == "qwen3-moe-30b" and mbs == 8
                              and seq == 4096) else None
        roofline = {"bound": "mfma", "achieved": round(ach, 1), "peak": 2500.0,
                    "unit": "TFLOP/s", "frac": round(ach / 2500.0, 4),
                    "traffic": traffic,
                    "traffic_source": ("rocprofv3 TCC pass at the fc1 mbs8 "
                                       "launch shape; profiles/r02_gg_traffic.txt"
                                       if traffic else None),
                    "kernel": "vh_group_gemm_nk*_bf16 (shape-dispatched)"}
    elif "ce_fwd" in prof:
        rec = prof["ce_fwd"]
        by = sum(rec["work"]) / rec["count"]
        ach = by / (rec["ms_avg"] / 1000.0) / 1e9
        roofline = {"bound": "hbm", "achieved": round(ach, 1), "peak": 8000.0,
                    "unit": "GB/s", "frac": round(ach / 8000.0, 4), "traffic": None,
                    "kernel": "vh_ce_fwd_bf16"}

    cpu_baseline = None
    if rank == 0 and n_gpus == 1 and not args.no_cpu_baseline and not is_vl:
        log("timing CPU baseline (host cores, bounded sample)")
        try:
            cpu_baseline = run_cpu_baseline(preset, seq)
        except Exception as e:  # report, never break the bench line
            cpu_baseline = {"value": None, "unit": "tokens/s", "cores": None,
                            "kind": "port", "sample": f"failed: {e}"}
        bind_ops(HIP_OPS_CONFIG)

    if rank == 0:
        out = {
            "metric": "tokens/sec/node",
            "value": round(toks_per_s, 1),
            "unit": "tokens/s",
            "n_gpus": n_gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(ms_per_step, 2),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16",
            "data": "synthetic",
            "mfu": round(mfu, 4),
            "config": {
                "workload": f"{preset} FSDP2 bf16 seq{seq}"
                + (f" EP{ep_size}" if ep_size > 1 else ""),
                "model": preset,
                "global_batch": mbs * n_gpus,
                "seq_len": seq,
                "parallelism": f"dp{world // sp}"
                + (f"_sp{sp}" + ("sync" if args.sync_ulysses else "") if sp > 1 else "")
                + (f"_ep{ep_size}" if ep_size > 1 else ""),
                # the aux load-balancing loss IS part of every timed step
                # (reference Qwen3-MoE default; ref load_balancing_loss/eager.py)
                "router_aux_loss_coef": cfg.router_aux_loss_coef,
            },
            "roofline": roofline,
            "cpu_baseline": cpu_baseline,
        }
        print(json.dumps(out), flush=True)

    if world > 1:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
