"""A/B: custom NT GEMM (acco_gemm_nt) vs hipBLASLt (torch.matmul) on the
live projection shapes. Random [-1,1) operands (guide rule 25)."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from acco_amd import ops

SHAPES = [
    ("qkv      ", 8192, 3072, 2048),
    ("o_proj   ", 8192, 2048, 2048),
    ("gate_up  ", 8192, 16384, 2048),
    ("down     ", 8192, 2048, 8192),
]


def bench(fn, iters=20):
    for _ in range(3):
        fn()
    torch.cuda.synchronize()
    t = time.time()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.time() - t) / iters


def main():
    ext = ops.hip_ext()
    torch.manual_seed(0)
    for name, M, N, K in SHAPES:
        A = (torch.rand(M, K, device="cuda") * 2 - 1).bfloat16()
        B = (torch.rand(N, K, device="cuda") * 2 - 1).bfloat16()
        C = ext.gemm_nt(A, B)
        ref = torch.matmul(A.float(), B.float().t())
        err = (C.float() - ref).abs().max().item()
        rel = err / ref.abs().max().item()
        fl = 2.0 * M * N * K
        t_us = bench(lambda: ext.gemm_nt(A, B)) * 1e6
        t_blas = bench(lambda: torch.matmul(A, B.t())) * 1e6
        print(f"{name} M{M} N{N} K{K}: ours {t_us:7.1f} us {fl/t_us/1e6:6.0f} TF "
              f"| hipBLASLt {t_blas:7.1f} us {fl/t_blas/1e6:6.0f} TF "
              f"| maxrelerr {rel:.2e}")


if __name__ == "__main__":
    main()
