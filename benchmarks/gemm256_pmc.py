import sys, torch
sys.path.insert(0, ".")
from learningorchestra_amd.ops import functional as F
N = 4096
A = torch.randn(N, N, device="cuda").to(torch.bfloat16)
B = torch.randn(N, N, device="cuda").to(torch.bfloat16)
C = torch.empty(N, N, device="cuda", dtype=torch.bfloat16)
for _ in range(8):
    F.gemm(A, B, tb=True, out=C)
torch.cuda.synchronize()
print("done")
