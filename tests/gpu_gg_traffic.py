#!/usr/bin/env python3
"""Single-kernel launcher for the TCC traffic passes (rocprofv3 --pmc
FETCH_SIZE / WRITE_SIZE in separate passes): runs ONLY nk256s at the exact
N=1 bench fc1 shape (mbs8: 262144 scattered rows, G=128, N=1536, K=2048)."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from veomni_amd.ops import hip_lib as L


def main():
    dev = "cuda"
    torch.manual_seed(0)
    G, rows, N, K = 128, 262144, 1536, 2048
    cumsum = torch.full((G,), rows // G, device=dev).cumsum(0)
    a = (torch.randn(rows, K, device=dev) * 0.3).to(torch.bfloat16)
    w = (torch.randn(G, N, K, device=dev) * 0.3).to(torch.bfloat16)
    c = torch.empty(rows, N, dtype=torch.bfloat16, device=dev)
    lib = L.get_lib()
    for _ in range(5):
        rc = lib.vh_group_gemm_nk256s_bf16(a.data_ptr(), w.data_ptr(), c.data_ptr(),
                                           cumsum.data_ptr(), G, N, K, rows,
                                           L.cur_stream())
        assert rc == 0
    torch.cuda.synchronize()
    print("traffic run done")


if __name__ == "__main__":
    main()
