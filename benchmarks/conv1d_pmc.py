"""PMC probe for conv1d_dx2 / conv1d_fwd2 (r2): what stalls the ~330 us
plateau (traffic floor ~70 us, MFMA floor ~41 us)."""
import os
import sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from learningorchestra_amd.ops import functional as F

B, H, C, outC, KH = 4096, 256, 128, 128, 3
OH = H - KH + 1
dy2 = torch.randn(B * OH, outC, device="cuda").to(torch.bfloat16)
wt = torch.randn(KH * C, outC, device="cuda").to(torch.bfloat16)
dx = torch.empty(B, H, 1, C, device="cuda", dtype=torch.bfloat16)
for _ in range(6):
    F.conv1d_dx(dy2, wt, KH, 0, out=dx)
torch.cuda.synchronize()
print("done")
