import torch, time
from learningorchestra_amd.ops import functional as F
A = torch.randn(4096, 4096, device="cuda").bfloat16()
B = torch.randn(4096, 4096, device="cuda").bfloat16()
C = torch.empty(4096, 4096, device="cuda", dtype=torch.bfloat16)
for _ in range(3):
    F.gemm(A, B, tb=True, out=C)
torch.cuda.synchronize()
for _ in range(5):
    F.gemm(A, B, tb=True, out=C)
torch.cuda.synchronize()
print("done")
