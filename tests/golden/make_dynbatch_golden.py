#!/usr/bin/env python3
"""Generate golden micro-batch compositions by RUNNING the reference's
dynamic batching (veomni/data/dynamic_batching.py) on deterministic synthetic
sample streams. Run in the build container (needs /root/reference); commits
tests/golden/dynbatch.json for the CPU parity test."""

import importlib.util
import json
import os
import random
import sys
import types

import torch

REF = "/root/reference"


def _load_ref_dynbatch():
    # import the module file directly; stub the veomni package chain so the
    # heavy veomni.data.__init__ (torchdata etc.) never runs
    sys.path.insert(0, REF)
    for name in ("veomni", "veomni.data", "veomni.utils"):
        if name not in sys.modules:
            m = types.ModuleType(name)
            m.__path__ = [os.path.join(REF, *name.split(".")[0:])]
            sys.modules[name] = m
    import logging as pylog

    lg = types.ModuleType("veomni.utils.logging")
    lg.get_logger = lambda *_: pylog.getLogger("ref")
    sys.modules["veomni.utils.logging"] = lg
    spec = importlib.util.spec_from_file_location(
        "veomni.data.dynamic_batching",
        os.path.join(REF, "veomni", "data", "dynamic_batching.py"))
    mod = importlib.util.module_from_spec(spec)
    sys.modules["veomni.data.dynamic_batching"] = mod
    spec.loader.exec_module(mod)
    return mod


def sample_stream(n, seed, max_len=900):
    rng = random.Random(seed)
    out = []
    for i in range(n):
        L = rng.randint(1, max_len)  # length 1 exercises the empty-string drop
        out.append({"input_ids": torch.full((L,), i, dtype=torch.int64),
                    "attention_mask": torch.ones(L, dtype=torch.int64),
                    "sid": i})
    return out


def run_case(mod, case):
    strat = mod.TextBatchingStrategy(
        token_micro_bsz=case["token_micro_bsz"],
        buffer_size=case["buffer_size"],
        bsz_warmup_steps=case.get("bsz_warmup_steps", 0),
        bsz_warmup_init_mbtoken=case.get("bsz_warmup_init_mbtoken", 200),
        physical_token_cap=case.get("physical_token_cap"),
    )
    samples = sample_stream(case["n_samples"], case["seed"])
    loader = mod.DynamicBatchSizeDataLoader(
        samples, strat, collate_fn=None,
        num_micro_batch=case["num_micro_batch"],
        length=case["length"], drop_last=case["drop_last"])
    steps = []
    for group in loader:
        steps.append([[int(s["sid"]) for s in mb]
                      + (["PAD"] if s_is_pad(mb) else []) for mb in group])
        if len(steps) >= case.get("max_steps", 50):
            break
    return steps


def s_is_pad(mb):
    return isinstance(mb, dict) and mb.get("padding_flag")


def main():
    mod = _load_ref_dynbatch()
    cases = [
        dict(name="plain", n_samples=400, seed=1, token_micro_bsz=4096,
             buffer_size=20, num_micro_batch=1, length=30, drop_last=True),
        dict(name="warmup", n_samples=400, seed=2, token_micro_bsz=4096,
             buffer_size=10, bsz_warmup_steps=8, bsz_warmup_init_mbtoken=512,
             num_micro_batch=2, length=12, drop_last=True),
        dict(name="physcap", n_samples=300, seed=3, token_micro_bsz=4096,
             buffer_size=5, physical_token_cap=2048, num_micro_batch=1,
             length=25, drop_last=True),
        # NB: with a re-iterable source the reference wraps epochs until
        # `length` steps, so this exercises the wrap path (the drop_last=False
        # tail-flush branch is only reachable for one-shot iterators, where
        # the reference leaks StopIteration; mirrored structurally).
        dict(name="wrap", n_samples=60, seed=4, token_micro_bsz=8192,
             buffer_size=4, num_micro_batch=2, length=40, drop_last=False,
             max_steps=40),
    ]
    out = {}
    for c in cases:
        # padding groups in the reference are dict micro-batches (no collate):
        # mark them by padding_flag
        strat_steps = run_case(mod, c)
        out[c["name"]] = {"case": {k: v for k, v in c.items()}, "steps": strat_steps}
    path = os.path.join(os.path.dirname(__file__), "dynbatch.json")
    with open(path, "w") as f:
        json.dump(out, f, indent=1)
    print("wrote", path, {k: len(v["steps"]) for k, v in out.items()})


if __name__ == "__main__":
    main()
