"""Microbench: fused conv_dx vs NT-GEMM(dcol)+col2im on the MNIST conv2 shape."""
import sys, time, torch
sys.path.insert(0, ".")
from learningorchestra_amd.ops import functional as F

def t(fn, iters=20):
    for _ in range(3):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e3

B, H, W, C, KH, S, P, outC = 32768, 12, 12, 32, 5, 1, 0, 64
OH = OW = (H - KH) // S + 1
kpad = KH * KH * C
dy2 = torch.randn(B * OH * OW, outC, device="cuda").to(torch.bfloat16)
wt = torch.randn(kpad, outC, device="cuda").to(torch.bfloat16)
dx = torch.empty(B, H, W, C, device="cuda", dtype=torch.bfloat16)
dcol = torch.empty(B * OH * OW, kpad, device="cuda", dtype=torch.bfloat16)

ms_f = t(lambda: F.conv2d_dx_fused(dy2, wt, B, H, W, C, KH, KH, S, S, P, P, out=dx))
def ref():
    F.gemm(dy2, wt, tb=True, out=dcol)
    F.col2im(dcol, B, H, W, C, KH, KH, S, S, P, P, out=dx)
ms_r = t(ref)
gb = (dy2.numel() + wt.numel()) * 2 / 1e9 + dx.numel() * 2 / 1e9
tf = 2 * (B * OH * OW) * kpad * outC / 1e12
print(f"fused   {ms_f:8.3f} ms  {tf/ms_f*1e3:7.1f} TF  {gb/ms_f*1e3:6.2f} TB/s-min")
print(f"ref     {ms_r:8.3f} ms  {tf/ms_r*1e3:7.1f} TF")
