import sys, time
sys.path.insert(0, "/root/repo")
import torch
from acco_amd import ops
ext = ops.hip_ext()

def t(fn, iters=20):
    for _ in range(5): fn()
    torch.cuda.synchronize(); t0 = time.time()
    for _ in range(iters): fn()
    torch.cuda.synchronize(); return (time.time() - t0) / iters

for (R, I, tag) in [(2048, 14336, "8b"), (8192, 8192, "1b"), (2048, 16384, "8b-pow2I")]:
    gu = torch.randn(R, 2 * I, device="cuda", dtype=torch.bfloat16)
    g = torch.randn(R, I, device="cuda", dtype=torch.bfloat16)
    u = torch.randn_like(g)
    bytes_ = R * I * 2 * 3
    s = t(lambda: ext.swiglu_packed_fwd(gu))
    c = t(lambda: ext.swiglu_fwd(g, u))
    print(f"{tag:9s} R={R} I={I} packed {s*1e6:7.1f}us {bytes_/s/1e12:5.2f} TB/s | contig {c*1e6:7.1f}us {bytes_/c/1e12:5.2f} TB/s")
