"""Race probe for the 8-phase gemm256 kernel (r2): error-block mapping."""
import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from learningorchestra_amd.ops import functional as F


def probe(M, N=256, K=512, iters=20):
    torch.manual_seed(4)
    A = torch.randn(M, K, device="cuda").to(torch.bfloat16)
    B = torch.randn(N, K, device="cuda").to(torch.bfloat16)
    ref = (A.float() @ B.float().t())
    allblocks = set()
    bad = 0
    for i in range(iters):
        out = F.gemm(A, B, tb=True)
        err = (out.float() - ref).abs()
        mask = err > 2.0
        if mask.any():
            bad += 1
            rows = mask.any(1).nonzero().flatten()
            allblocks |= set((rows // 256).tolist())
    nb = (M + 255) // 256
    bl = sorted(allblocks)
    print(f"M={M} ({nb} blocks): bad {bad}/{iters}; "
          f"err blocks n={len(bl)} min={bl[0] if bl else '-'} "
          f"max={bl[-1] if bl else '-'} sample={bl[:10]}", flush=True)


if __name__ == "__main__":
    for M in (32768, 33024, 40960, 51200, 65536):
        probe(M)


def granular():
    print("CUs:", torch.cuda.get_device_properties(0).multi_processor_count,
          flush=True)
    for nb in (161, 168, 176, 180, 184, 188, 192, 193, 196, 200):
        probe(nb * 256, iters=12)


def verify_fix():
    print("CUs:", torch.cuda.get_device_properties(0).multi_processor_count,
          flush=True)
    # fast-path region: high-iteration soak
    for M in (40960, 45056, 48128):
        probe(M, K=1024, iters=40)
    # drain region
    for M in (49152, 51200, 65536, 102400):
        probe(M, K=512, iters=30)
        probe(M, K=1024, iters=10)
