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

    iters = 20
    for _ in range(3):
        ops.hip_ext().attn_fwd(q, k, v, sc, 0)
        ops.hip_ext().attn_bwd(q, k, v, do, l, delta, sc, 0)
    torch.cuda.synchronize()
    t = time.time()
    for _ in range(iters):
        ops.hip_ext().attn_fwd(q, k, v, sc, 0)
    torch.cuda.synchronize()
    dt_f = (time.time() - t) / iters
    t = time.time()
    for _ in range(iters):
        ops.hip_ext().attn_bwd(q, k, v, do, l, delta, sc, 0)
    torch.cuda.synchronize()
    dt_b = (time.time() - t) / iters
    fl_f = 2 * 2 * B * H * S * S * Dh * 0.5          # causal: half the tiles
    fl_b = fl_f * 2.5
    print(f"attn fwd:  {dt_f*1e3:.3f} ms, ~{fl_f/dt_f/1e12:.0f} TF/s")
    print(f"attn bwd:  {dt_b*1e3:.3f} ms, ~{fl_b/dt_b/1e12:.0f} TF/s")
    dt = dt_f + dt_b
    print(f"attn fwd+bwd: {dt*1e3:.3f} ms, "
          f"~{(fl_f+fl_b)/dt/1e12:.0f} TF/s combined")


if __name__ == "__main__":
    main()
