"""Numerics + timing for the g2fast conv-dW path (BK % (OH*OW) == 0)."""
import time, torch
import learningorchestra_amd.ops.functional as F

def check(B, H, W, C, KH, outC, label):
    OH = OW = H - KH + 1
    kpad = ((KH * KH * C + 7) // 8) * 8
    torch.manual_seed(3)
    dy2 = torch.randn(B * OH * OW, outC, device="cuda").to(torch.bfloat16)
    x = torch.randn(B, H, W, C, device="cuda").to(torch.bfloat16)
    gw = torch.empty(outC, kpad, device="cuda", dtype=torch.float32)
    F.conv2d_dw_implicit(dy2, x, gw, KH, KH, 1, 1, 0, 0, 8)
    # reference via materialized col + fp32 matmul
    col = F.im2col(x, KH, KH, 1, 1, 0, 0, kpad)
    ref = dy2.float().t() @ col.float()
    err = (gw - ref).abs().max().item()
    rel = err / ref.abs().max().item()
    # timing
    for _ in range(3):
        F.conv2d_dw_implicit(dy2, x, gw, KH, KH, 1, 1, 0, 0, 8)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(10):
        F.conv2d_dw_implicit(dy2, x, gw, KH, KH, 1, 1, 0, 0, 8)
    torch.cuda.synchronize()
    us = (time.perf_counter() - t0) / 10 * 1e6
    print(f"{label}: maxerr {err:.4f} rel {rel:.2e} time {us:.0f} us", flush=True)

check(32768, 12, 12, 32, 5, 64, "mnist-conv2 (fast path, R=64|BK)")
check(4096, 14, 14, 32, 7, 64, "R=64 k7 (fast path)")
check(4096, 12, 12, 32, 6, 64, "R=49 (generic path)")
check(2048, 18, 18, 16, 3, 32, "R=256 (generic, BK%256!=0... 64%256!=0)")
