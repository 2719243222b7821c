import os, sys, time
sys.path.insert(0, "/root/repo")
import torch
from learningorchestra_amd.ops import functional as F

def bench(fn, iters=20, warm=4):
    for _ in range(warm): fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6

# ResNet-50 B512 dX GEMM shapes (plain F,T: dy @ wt): (M, N=in_c, K=out_c)
shapes = [
    ("s0 1x1 c1dx", 512*56*56, 64, 64),
    ("s0 c3 dcol",  512*56*56, 576, 64),
    ("s0 1x1 c3dx", 512*56*56, 64, 256),
    ("s1 1x1 dx",   512*28*28, 128, 512),
    ("s1 c3 dcol",  512*28*28, 1152, 128),
    ("s2 1x1 dx",   512*14*14, 256, 1024),
    ("s2 c3 dcol",  512*14*14, 2304, 256),
    ("s3 1x1 dx",   512*7*7, 512, 2048),
    ("fwd-noepi s3", 512*7*7, 2048, 512),
    ("s3 c3 dcol",  512*7*7, 4608, 512),
    ("mnist fc1 dx", 32768, 1024, 256),
    ("mnist fc2 dx", 32768, 256, 16),
    ("tc fc dx",    4096, 384, 16),
]
for name, M, N, K in shapes:
    A = torch.randn(M, K, device="cuda").to(torch.bfloat16)
    B = torch.randn(N, K, device="cuda").to(torch.bfloat16)
    out = torch.empty(M, N, device="cuda", dtype=torch.bfloat16)
    t_ours = bench(lambda: F.gemm(A, B, tb=True, out=out))
    t_rb = bench(lambda: torch.matmul(A, B.t(), out=out))
    tf = 2*M*N*K/1e12
    print(f"{name:14s} M={M:8d} N={N:5d} K={K:5d}: ours {t_ours:7.0f}us ({tf/t_ours*1e6:5.0f}TF)  rocBLAS {t_rb:7.0f}us ({tf/t_rb*1e6:5.0f}TF)  ratio {t_rb/t_ours:.2f}")
