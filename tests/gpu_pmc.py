#!/usr/bin/env python3
"""PMC probe target: launches the grouped-GEMM variants a fixed number of
times so rocprofv3 --pmc can attribute counters per kernel.

Usage (on the GPU box, counters in their own pass — never with traces):
  rocprofv3 --pmc SQ_LDS_BANK_CONFLICT SQ_WAIT_ANY SQ_WAIT_INST_ANY \
            SQ_ACTIVE_INST_ANY SQ_WAVE_CYCLES -d out -- python tests/gpu_pmc.py
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from veomni_amd.ops import hip_lib as L


def main():
    dev = "cuda"
    torch.manual_seed(0)
    E, rows, H, I = 16, 32768, 2048, 768
    cumsum = torch.full((E,), rows // E).cumsum(0).to(dev)
    a = (torch.randn(rows, H, device=dev) * 0.3).to(torch.bfloat16)
    w1 = (torch.randn(E, 2 * I, H, device=dev) * 0.3).to(torch.bfloat16)
    g1 = (torch.randn(rows, 2 * I, device=dev) * 0.3).to(torch.bfloat16)

    lib = L.get_lib()
    c1 = torch.empty(rows, 2 * I, dtype=torch.bfloat16, device=dev)
    c2 = torch.empty(rows, H, dtype=torch.bfloat16, device=dev)
    cw = torch.empty(E, 2 * I, H, dtype=torch.bfloat16, device=dev)
    s = L.cur_stream()
    for _ in range(5):
        # fwd on nk256 (current dispatch target)
        lib.vh_group_gemm_nk256_bf16(a.data_ptr(), w1.data_ptr(), c1.data_ptr(),
                                     cumsum.data_ptr(), E, 2 * I, H, rows, s)
        # fc1 dgrad on nk8 (current dispatch target)
        lib.vh_group_gemm_nk8_bf16(g1.data_ptr(), w1.data_ptr(), c2.data_ptr(),
                                   cumsum.data_ptr(), E, H, 2 * I, rows, 0, s)
        # wgrad on mn8 (current dispatch target)
        lib.vh_group_gemm_mn8_bf16(g1.data_ptr(), a.data_ptr(), cw.data_ptr(),
                                   cumsum.data_ptr(), E, 2 * I, H, s)
    torch.cuda.synchronize()
    print("pmc probe done")


if __name__ == "__main__":
    main()
