"""Micro-benchmark for the memory-bound HIP kernels (everything except
attention/GEMM): fused AdamW, RMSNorm, LayerNorm, RoPE, SwiGLU, gelu_new,
fused causal-LM CE, attention Delta.

Each op runs at the flagship llama-1b / gptneo-125m live shape. Prints
achieved effective HBM bandwidth (bytes moved / kernel wall time) so the
profiles/ evidence shows how close each kernel sits to the ~8 TB/s HBM3E
roof.  Also the rocprofv3 --pmc target for per-kernel counters.
"""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from acco_amd import ops

DEV = "cuda"


def timeit(fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t = time.time()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.time() - t) / iters


def report(name, sec, bytes_moved):
    print(f"{name:34s} {sec*1e6:9.1f} us   {bytes_moved/sec/1e12:6.2f} TB/s eff")


def main():
    torch.manual_seed(0)
    ext = ops.hip_ext()
    B, S, D = 8, 1024, 2048          # llama-1b live shape
    R = B * S

    # ---- fused AdamW (one llama-1b com-round bucket at world=1: ~140M)
    n = 140_000_000
    p = torch.randn(n, device=DEV)
    m = torch.zeros(n, device=DEV)
    v = torch.zeros(n, device=DEV)
    g = torch.randn(n, device=DEV, dtype=torch.bfloat16)
    out = torch.empty(n, device=DEV, dtype=torch.bfloat16)
    no_sd = torch.empty(0, device=DEV)

    def adamw(commit):
        ext.fused_adamw(p, g, m, v, 10, 3e-4, 0.9, 0.95, 1e-8, 0.1, 1.0,
                        no_sd, out, commit)

    # commit: read p,m,v fp32 + g bf16; write p,m,v fp32 + out bf16
    report("adamw commit (140M)", timeit(lambda: adamw(True), iters=10),
           n * (12 + 2 + 12 + 2))
    # tentative: read p,m,v,g; write out only
    report("adamw tentative (140M)", timeit(lambda: adamw(False), iters=10),
           n * (12 + 2 + 2))

    # ---- RMSNorm fwd/bwd (llama-1b: R=8192, D=2048)
    x = torch.randn(B, S, D, device=DEV, dtype=torch.bfloat16)
    w = torch.randn(D, device=DEV, dtype=torch.bfloat16)
    y, rstd = ext.rmsnorm_fwd(x, w, 1e-5)
    dy = torch.randn_like(y)
    report("rms_norm fwd (8x1024x2048)",
           timeit(lambda: ext.rmsnorm_fwd(x, w, 1e-5)), R * D * 2 * 2)
    report("rms_norm bwd",
           timeit(lambda: ext.rmsnorm_bwd(dy, x, w, rstd)), R * D * 2 * 3)

    # ---- LayerNorm (gptneo shape 8x2048x768)
    Bn, Sn, Dn = 8, 2048, 768
    xn = torch.randn(Bn, Sn, Dn, device=DEV, dtype=torch.bfloat16)
    wn = torch.randn(Dn, device=DEV, dtype=torch.bfloat16)
    bn = torch.randn(Dn, device=DEV, dtype=torch.bfloat16)
    yn, mu, ivn = ext.layernorm_fwd(xn, wn, bn, 1e-5)
    dyn = torch.randn_like(yn)
    report("layer_norm fwd (8x2048x768)",
           timeit(lambda: ext.layernorm_fwd(xn, wn, bn, 1e-5)),
           Bn * Sn * Dn * 2 * 2)
    report("layer_norm bwd",
           timeit(lambda: ext.layernorm_bwd(dyn, xn, wn, mu, ivn)),
           Bn * Sn * Dn * 2 * 3)

    # ---- RoPE (llama-1b q: [8,1024,32,64])
    H, Hkv, Dh = 32, 8, 64
    q4 = torch.randn(B, S, H, Dh, device=DEV, dtype=torch.bfloat16)
    pos = torch.arange(S, dtype=torch.float32)
    inv = 1.0 / (10000.0 ** (torch.arange(0, Dh, 2, dtype=torch.float32) / Dh))
    fr = torch.outer(pos, inv)
    cos = torch.cat([fr.cos(), fr.cos()], -1).to(DEV).contiguous()
    sin = torch.cat([fr.sin(), fr.sin()], -1).to(DEV).contiguous()
    report("rope fwd (8x1024x32x64)",
           timeit(lambda: ext.rope_fwd(q4, cos, sin, False)),
           B * S * H * Dh * 2 * 2)

    # ---- SwiGLU packed (llama-1b MLP: inter 8192)
    I = 8192
    gu = torch.randn(R, 2 * I, device=DEV, dtype=torch.bfloat16)
    report("swiglu packed fwd (8192x2*8192)",
           timeit(lambda: ext.swiglu_packed_fwd(gu)), R * I * 2 * 3)
    dz = torch.randn(R, I, device=DEV, dtype=torch.bfloat16)
    report("swiglu packed bwd",
           timeit(lambda: ext.swiglu_packed_bwd(dz, gu)), R * I * 2 * 5)

    # ---- gelu_new (gptneo MLP 4*768)
    xg = torch.randn(Bn * Sn, 4 * Dn, device=DEV, dtype=torch.bfloat16)
    report("gelu_new fwd (16384x3072)",
           timeit(lambda: ext.gelu_fwd(xg)), xg.numel() * 2 * 2)
    report("gelu_new bwd",
           timeit(lambda: ext.gelu_bwd(xg, xg)), xg.numel() * 2 * 3)

    # ---- fused causal-LM CE (llama-1b head: vocab 50304)
    V = 50304
    logits = torch.randn(B, S, V, device=DEV, dtype=torch.bfloat16)
    labels = torch.randint(0, V, (B, S), device=DEV)
    acc, lse = ext.ce_fwd(logits, labels)
    report("ce fwd (8x1024x50304)",
           timeit(lambda: ext.ce_fwd(logits, labels)), B * S * V * 2)
    report("ce bwd",
           timeit(lambda: ext.ce_bwd(logits, labels, lse, acc, 1.0)),
           B * S * V * 2 * 2)

    # ---- attention Delta
    o = torch.randn(B, S, H, Dh, device=DEV, dtype=torch.bfloat16)
    do = torch.randn_like(o)
    report("attn delta (8x1024x32x64)",
           timeit(lambda: ext.attn_delta(do, o)), o.numel() * 2 * 2)


if __name__ == "__main__":
    main()
