#!/usr/bin/env python3
"""MFMA GEMM microbenchmark: our gfx950 kernel vs rocBLAS (torch.matmul)
on square + engine shapes. TFLOP/s, uniform random operands (guide §5.4
rule 25: never quote zero-fill numbers)."""
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
from learningorchestra_amd.ops import functional as F

SHAPES = [
    ("square-2048", 2048, 2048, 2048, False, True),
    ("square-4096", 4096, 4096, 4096, False, True),
    ("square-8192", 8192, 8192, 8192, False, True),
    ("rn50-c3s1", 401408, 64, 576, False, True),     # 3x3 conv stage1
    ("rn50-1x1-s3", 6272, 2048, 512, False, True),
    ("fc1-mnist", 8192, 256, 1024, False, True),
    ("dx-nn-4096", 4096, 4096, 4096, False, False),
]


def bench(fn, iters=10, warm=3):
    for _ in range(warm):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    assert torch.cuda.is_available()
    out = {}
    for name, M, N, K, ta, tb in SHAPES:
        A = torch.randn(M, K, device="cuda").bfloat16()
        B = (torch.randn(N, K) if tb else torch.randn(K, N)).cuda().bfloat16()
        C = torch.empty(M, N, device="cuda", dtype=torch.bfloat16)
        dt = bench(lambda: F.gemm(A, B, ta=ta, tb=tb, out=C))
        tf = 2.0 * M * N * K / dt / 1e12
        B_kn = B.t().contiguous() if tb else B
        dt_ref = bench(lambda: torch.matmul(A, B_kn))
        tf_ref = 2.0 * M * N * K / dt_ref / 1e12
        out[name] = {"lo_tflops": round(tf, 1), "rocblas_tflops": round(tf_ref, 1),
                     "ratio": round(tf / tf_ref, 3)}
        print(name, out[name], flush=True)
    print(json.dumps(out))


if __name__ == "__main__":
    main()
