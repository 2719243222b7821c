import sys, torch
sys.path.insert(0, ".")
from learningorchestra_amd.ops import functional as F
B, H, W, C, KH, outC = 32768, 12, 12, 32, 5, 64
OH = OW = H - KH + 1
kpad = KH * KH * C
dy2 = torch.randn(B * OH * OW, outC, device="cuda").to(torch.bfloat16)
wt = torch.randn(kpad, outC, device="cuda").to(torch.bfloat16)
dx = torch.empty(B, H, W, C, device="cuda", dtype=torch.bfloat16)
for _ in range(5):
    F.conv2d_dx_fused(dy2, wt, B, H, W, C, KH, KH, 1, 1, 0, 0, out=dx)
torch.cuda.synchronize()
print("done")
