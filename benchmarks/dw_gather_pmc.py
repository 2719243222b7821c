import sys, torch
sys.path.insert(0, ".")
from learningorchestra_amd.ops import functional as F
# MNIST conv2 dW shape: dy2 [2.1M, 64], x [32768,12,12,32], gw [64, 800]
B, H, W, C, KH, outC = 32768, 12, 12, 32, 5, 64
OH = OW = H - KH + 1
kpad = KH * KH * C
dy2 = torch.randn(B * OH * OW, outC, device="cuda").to(torch.bfloat16)
x = torch.randn(B, H, W, C, device="cuda").to(torch.bfloat16)
gw = torch.empty(outC, kpad, device="cuda", dtype=torch.float32)
for _ in range(5):
    F.conv2d_dw_implicit(dy2, x, gw, KH, KH, 1, 1, 0, 0, 8)
torch.cuda.synchronize()
print("done")
