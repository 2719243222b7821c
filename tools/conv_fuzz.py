"""Randomized shape fuzz for the fused conv kernel family vs references.

Run on a GPU box:  python tools/conv_fuzz.py [n_iters] [seed]
Draws random shapes, keeps the ones each kernel's launcher accepts, and
compares against the im2col/GEMM (fwd, dW) and dcol/col2im (dX) paths.
"""
import random
import sys

import torch

sys.path.insert(0, ".")
from learningorchestra_amd.ops import functional as F  # noqa: E402


def rnd_shapes(rng):
    C = rng.choice([1, 16, 32, 64, 96, 128])
    KH = rng.choice([1, 2, 3, 5, 7])
    KW = rng.choice([1, KH])
    S = rng.choice([1, 1, 1, 2])
    P = rng.choice([0, 0, 1, 2, KH // 2])
    H = rng.choice([8, 12, 15, 16, 24, 28, 56, 100, 256])
    W = 1 if KW == 1 and rng.random() < 0.5 else H
    outC = rng.choice([8, 16, 24, 32, 48, 64, 128])
    B = rng.choice([1, 2, 3, 5, 8])
    if H + 2 * P < KH or W + 2 * P < KW:
        return None
    return B, H, W, C, KH, KW, S, P, outC


def main(iters=60, seed=0):
    rng = random.Random(seed)
    torch.manual_seed(seed)
    tried = {"fwd": 0, "dx": 0, "dw1": 0, "fwd1d": 0, "dx1d": 0}
    for it in range(iters):
        sh = rnd_shapes(rng)
        if sh is None:
            continue
        B, H, W, C, KH, KW, S, P, outC = sh
        OH = (H + 2 * P - KH) // S + 1
        OW = (W + 2 * P - KW) // S + 1
        if OH <= 0 or OW <= 0 or B * OH * OW > 2_000_000:
            continue
        kdim = KH * KW * C
        kpad = (kdim + 7) // 8 * 8
        x = (torch.randn(B, H, W, C, device="cuda") * 0.5).to(torch.bfloat16)
        w = torch.zeros(outC, kpad, device="cuda", dtype=torch.bfloat16)
        w[:, :kdim] = (torch.randn(outC, kdim, device="cuda") * 0.2).to(torch.bfloat16)
        bias = torch.randn(outC, device="cuda", dtype=torch.float32)
        dy2 = (torch.randn(B * OH * OW, outC, device="cuda") * 0.5).to(torch.bfloat16)
        wt = w.t().contiguous()
        tag = f"it{it} B{B} {H}x{W}x{C} k{KH}x{KW} s{S} p{P} ->{outC}"

        # --- fwd (conv_fwd_small) --------------------------------------
        y = torch.empty(B * OH * OW, outC, device="cuda", dtype=torch.bfloat16)
        if F.conv2d_fwd_small(x, w, KH, KW, S, S, P, P, bias=bias, relu=True,
                              out=y):
            tried["fwd"] += 1
            col = F.im2col(x, KH, KW, S, S, P, P, kpad)
            ref = F.gemm(col, w, tb=True, bias=bias, relu=True)
            torch.testing.assert_close(y.float(), ref.float(), atol=0.1,
                                       rtol=5e-2, msg=lambda m: f"fwd {tag}\n{m}")

        # --- fwd 1-D (P applies to h only in the 1-D kernels; the fuzz
        # dy2/x use the 2-D geometry, so only P==0 is comparable) ---------
        if W == 1 and KW == 1 and S == 1 and P == 0:
            y1 = torch.empty(B * OH, outC, device="cuda", dtype=torch.bfloat16)
            if F.conv1d_fwd(x, w, KH, P, bias=bias, relu=False, out=y1):
                tried["fwd1d"] += 1
                col = F.im2col(x, KH, 1, 1, 1, P, 0, kpad)
                ref = F.gemm(col, w, tb=True, bias=bias)
                torch.testing.assert_close(y1.float(), ref.float(), atol=0.1,
                                           rtol=5e-2,
                                           msg=lambda m: f"fwd1d {tag}\n{m}")

        # --- dX (conv_dx) ----------------------------------------------
        dx = torch.empty(B, H, W, C, device="cuda", dtype=torch.bfloat16)
        if F.conv2d_dx_fused(dy2, wt, B, H, W, C, KH, KW, S, S, P, P, out=dx):
            tried["dx"] += 1
            dcol32 = (dy2.float() @ wt.float().t()).cpu()
            ref32 = F.col2im(dcol32, B, H, W, C, KH, KW, S, S, P, P,
                             out=torch.empty(B, H, W, C))
            err = (dx.float().cpu() - ref32).abs().max().item()
            scale = ref32.abs().max().item() + 1.0
            assert err < 0.03 * scale + 0.05, f"dx {tag}: err {err} scale {scale}"

        # --- dX 1-D -----------------------------------------------------
        if W == 1 and KW == 1 and S == 1 and P == 0:
            dx1 = torch.empty(B, H, 1, C, device="cuda", dtype=torch.bfloat16)
            if F.conv1d_dx(dy2, wt, KH, P, out=dx1):
                tried["dx1d"] += 1
                dcol32 = (dy2.float() @ wt.float().t()).cpu()
                ref32 = F.col2im(dcol32, B, H, 1, C, KH, 1, 1, 1, P, 0,
                                 out=torch.empty(B, H, 1, C))
                err = (dx1.float().cpu() - ref32).abs().max().item()
                scale = ref32.abs().max().item() + 1.0
                assert err < 0.03 * scale + 0.05, f"dx1d {tag}: err {err}"

        # --- dW C=1 ------------------------------------------------------
        if C == 1 and outC <= 32:
            gw = torch.empty(outC, kpad, device="cuda", dtype=torch.float32)
            if F.conv2d_dw_c1(dy2, x, gw, KH, KW, S, S, P, P):
                tried["dw1"] += 1
                col = F.im2col(x, KH, KW, S, S, P, P, kpad)
                ref = F.gemm(dy2, col, ta=True, splits=4)
                torch.testing.assert_close(gw[:, :kdim], ref[:, :kdim],
                                           atol=ref.abs().max().item() * 2e-2 + 0.5,
                                           rtol=3e-2,
                                           msg=lambda m: f"dw1 {tag}\n{m}")
    print("FUZZ OK", tried)


if __name__ == "__main__":
    main(int(sys.argv[1]) if len(sys.argv) > 1 else 60,
         int(sys.argv[2]) if len(sys.argv) > 2 else 0)
