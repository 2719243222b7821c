"""Race screen for the new 8-phase 256^2 GEMM (sync-structure change -> guide
two-lane discipline: multi-run refcheck at several sizes)."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
from learningorchestra_amd.ops import functional as F

torch.manual_seed(0)
worst = 0.0
for trial in range(6):
    for Msz, Nsz, Ksz in [(256, 256, 256), (512, 512, 512), (512, 256, 320),
                          (1024, 1024, 1024), (2048, 2048, 2048),
                          (4096, 4096, 4096), (256, 512, 4096)]:
        A = torch.randn(Msz, Ksz, device="cuda").bfloat16()
        B = torch.randn(Nsz, Ksz, device="cuda").bfloat16()
        out = F.gemm(A, B, tb=True)
        ref = (A.float() @ B.float().t())
        rel = ((out.float() - ref).norm() / ref.norm()).item()
        worst = max(worst, rel)
        assert rel < 2e-2, (trial, Msz, Nsz, Ksz, rel)
    # bias+relu epilogue variant
    A = torch.randn(512, 1024, device="cuda").bfloat16()
    B = torch.randn(256, 1024, device="cuda").bfloat16()
    bias = torch.randn(256, device="cuda")
    out = F.gemm(A, B, tb=True, bias=bias, relu=True)
    ref = torch.relu(A.float() @ B.float().t() + bias)
    rel = ((out.float() - ref).norm() / ref.norm()).item()
    worst = max(worst, rel)
    assert rel < 2e-2, ("biasrelu", rel)
print("RACESCREEN OK, worst rel:", worst)
