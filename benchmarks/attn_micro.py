"""Attention kernel micro-benchmark / profiling target (flagship shape)."""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from acco_amd import ops


def main():
    torch.manual_seed(0)
    B, S, H, Hkv, Dh = 8, 1024, 32, 8, 64
    if "--8b" in sys.argv:
        B, S, H, Hkv, Dh = 4, 512, 32, 8, 128
    q = torch.randn(B, S, H, Dh, device="cuda").bfloat16()
    k = torch.randn(B, S, Hkv, Dh, device="cuda").bfloat16()
    v = torch.randn(B, S, Hkv, Dh, device="cuda").bfloat16()
    do = torch.randn(B, S, H, Dh, device="cuda").bfloat16()
    sc = Dh ** -0.5
    o, l = ops.hip_ext().attn_fwd(q, k, v, sc, 0)
    delta = (do.float() * o.float()).sum(-1).permute(0, 2, 1).contiguous()

    iters = 10
    for _ in range(3):
        ops.hip_ext().attn_fwd(q, k, v, sc, 0)
        ops.hip_ext().attn_bwd(q, k, v, do, l, delta, sc, 0)
    torch.cuda.synchronize()
    t = time.time()
    for _ in range(iters):
        ops.hip_ext().attn_fwd(q, k, v, sc, 0)
        ops.hip_ext().attn_bwd(q, k, v, do, l, delta, sc, 0)
    torch.cuda.synchronize()
    dt = (time.time() - t) / iters
    fl = 2 * 2 * B * H * S * S * Dh * 0.5 * (1 + 2.5)
    print(f"attn fwd+bwd: {dt*1e3:.3f} ms, ~{fl/dt/1e12:.0f} TF/s combined")


if __name__ == "__main__":
    main()
