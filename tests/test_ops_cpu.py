"""CPU reference-path tests for the op layer (the same references GPU
numerics tests compare the HIP kernels against)."""
import math

import pytest
import torch

from learningorchestra_amd.ops import functional as F


def test_gemm_combos_cpu():
    g = torch.Generator().manual_seed(0)
    A = torch.randn(33, 24, generator=g).bfloat16()
    B = torch.randn(24, 17, generator=g).bfloat16()
    ref = A.float() @ B.float()
    out = F.gemm(A, B)
    assert torch.allclose(out.float(), ref, atol=1e-1)
    # ta / tb
    out2 = F.gemm(A.t().contiguous(), B, ta=True)
    assert torch.allclose(out2.float(), ref, atol=1e-1)
    out3 = F.gemm(A, B.t().contiguous(), tb=True)
    assert torch.allclose(out3.float(), ref, atol=1e-1)


def test_gemm_bias_relu_cpu():
    A = torch.randn(8, 16).bfloat16()
    B = torch.randn(16, 4).bfloat16()
    bias = torch.randn(4)
    out = F.gemm(A, B, bias=bias, relu=True)
    ref = torch.relu(A.float() @ B.float() + bias)
    assert torch.allclose(out.float(), ref, atol=1e-1)
    assert (out.float() >= 0).all()


def test_im2col_col2im_adjoint():
    """<im2col(x), d> == <x, col2im(d)> — the transpose relationship that
    makes conv backward-data correct."""
    torch.manual_seed(0)
    B, H, W, C, k = 2, 8, 8, 4, 3
    kpad = ((k * k * C + 7) // 8) * 8
    x = torch.randn(B, H, W, C).bfloat16()
    col = F.im2col(x, k, k, 1, 1, 1, 1, kpad)
    OH = H + 2 - k + 1
    assert col.shape == (B * OH * OH, kpad)
    d = torch.randn(col.shape).bfloat16()
    dx = F.col2im(d, B, H, W, C, k, k, 1, 1, 1, 1)
    lhs = (col.float() * d.float()).sum()
    rhs = (x.float() * dx.float()).sum()
    assert abs(lhs - rhs) / (abs(lhs) + 1e-6) < 0.05


def test_maxpool_roundtrip():
    torch.manual_seed(1)
    x = torch.randn(2, 8, 8, 16).bfloat16()
    out, idx = F.maxpool2d(x, 2, 2, 2, 2)
    assert out.shape == (2, 4, 4, 16)
    ref = torch.nn.functional.max_pool2d(
        x.permute(0, 3, 1, 2).float(), 2).permute(0, 2, 3, 1)
    assert torch.allclose(out.float(), ref, atol=1e-2)
    dy = torch.randn_like(out)
    dx = F.maxpool2d_bwd(dy, idx, 8, 8, 2, 2, 2, 2)
    # each output grad lands on exactly one input position
    assert torch.allclose(dx.float().sum(), dy.float().sum(), rtol=0.05)


def test_softmax_ce_cpu():
    torch.manual_seed(2)
    M, C, CV = 64, 16, 10
    logits = torch.randn(M, C).bfloat16()
    labels = torch.randint(0, CV, (M,))
    dl, loss, correct = F.softmax_ce(logits, labels, cvalid=CV, gscale=1.0 / M)
    ref_loss = torch.nn.functional.cross_entropy(
        logits[:, :CV].float(), labels, reduction="sum")
    assert abs(loss.item() - ref_loss.item()) / ref_loss.item() < 0.02
    # grad rows sum to ~0 and padded cols are 0
    assert dl[:, CV:].float().abs().max() == 0
    assert dl.float().sum(1).abs().max() < 1e-2
    assert 0 <= correct.item() <= M


def test_sgd_step_cpu():
    n = 16
    master = torch.ones(n)
    grad = torch.full((n,), 2.0)
    mom = torch.zeros(n)
    mirror = torch.empty(n, dtype=torch.bfloat16)
    F.sgd_step(master, grad, mom, mirror, lr=0.1, mu=0.9, wd=0.0)
    assert torch.allclose(master, torch.full((n,), 1.0 - 0.2))
    F.sgd_step(master, grad, mom, mirror, lr=0.1, mu=0.9, wd=0.0)
    # mom = 0.9*2 + 2 = 3.8 ; master = 0.8 - 0.38
    assert torch.allclose(master, torch.full((n,), 0.42), atol=1e-6)
    assert torch.allclose(mirror.float(), master, atol=0.01)


def test_colsum_argmax_cpu():
    x = torch.randn(100, 24).bfloat16()
    assert torch.allclose(F.colsum(x), x.float().sum(0), atol=1e-2)
    am = F.argmax_rows(x, cvalid=10)
    assert (am.long() == x[:, :10].float().argmax(1)).all()
