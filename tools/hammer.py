"""GPU contention generator: big GEMM + conv loops until killed."""
import torch, sys
import learningorchestra_amd.ops.functional as F
torch.manual_seed(7)
A = torch.randn(8192, 2048, device="cuda").to(torch.bfloat16)
Bm = torch.randn(4096, 2048, device="cuda").to(torch.bfloat16)
x = torch.randn(16384, 28, 28, 1, device="cuda").to(torch.bfloat16)
import time
t_end = time.time() + float(sys.argv[1]) if len(sys.argv) > 1 else time.time() + 120
while time.time() < t_end:
    F.gemm(A, Bm, tb=True)
    torch.cuda.synchronize()
