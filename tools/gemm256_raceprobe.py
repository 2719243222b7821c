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


def stale_lds_probe():
    """Distinguish a stale-LDS dependence from a timing race. Prediction if
    some LDS region is consumed before being written this launch: after a
    DIFFERENT-shape gemm256 poisons the LDS, the first iterations fail and
    later ones self-heal (stale bytes == this shape's bytes again); with a
    poison run before EVERY iteration, every iteration fails. A timing race
    would give randomly scattered bad iterations instead."""
    import torch

    def run_case(M, poison_each, iters=16, K=512, N=256):
        torch.manual_seed(4)
        A = torch.randn(M, K, device="cuda").to(torch.bfloat16)
        B = torch.randn(N, K, device="cuda").to(torch.bfloat16)
        ref = (A.float() @ B.float().t())
        Ap = torch.randn(33024, K, device="cuda").to(torch.bfloat16)
        refp = None
        F.gemm(Ap, B, tb=True)                  # poison once (129 blocks)
        bad = []
        for i in range(iters):
            if poison_each:
                F.gemm(Ap, B, tb=True)
            out = F.gemm(A, B, tb=True)
            err = (out.float() - ref).abs().max().item()
            if err > 2.0:
                bad.append(i)
        print(f"M={M} poison_each={poison_each}: bad iters {bad}", flush=True)

    for rep in range(2):
        run_case(40960, False)
    for rep in range(2):
        run_case(40960, True)
