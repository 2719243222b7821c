"""MNIST LeNet-style CNN — the flagship BASELINE.json config
("MNIST LeNet-style CNN train bf16 on 1 MI355X").

Equivalent of the TF-model path the reference drove via binary_executor
(SURVEY §3.3), built on the explicit-backward NHWC engine with every hot op a
gfx950 HIP kernel: im2col + MFMA GEMM convs (fused bias+ReLU epilogue),
maxpool, MFMA FC layers, fused softmax-CE, fused SGD.

Topology (28x28x1 input): conv 5x5x32 -> pool2 -> conv 5x5x64 -> pool2 ->
fc 1024->256 (ReLU) -> fc 256->16 (10 valid classes, padded head).
"""
from __future__ import annotations

import os

from ..engine.layers import (Conv2dNHWC, Flatten, Linear, MaxPool2dNHWC,
                             SequentialClassifier)

NUM_CLASSES = 10
CPAD = 16  # classifier head padded for 16-B-aligned GEMM rows


def build_mnist_cnn(device="cpu", seed: int = 0,
                    channels=(32, 64), fc_width: int = 256) -> SequentialClassifier:
    c1, c2 = channels
    # Both convs default to implicit (no col matrix): conv2 runs the
    # small-image fused fwd + gather dW, conv1 the C=1 fused fwd + fused dW
    # (conv_fwd_small / conv_dw_c1) — measured 3.31M -> 5.43M samples/s at
    # B=32768 over the materialized-col baseline (PERFORMANCE.md).
    # LO_IMPLICIT_CONV=0 / LO_IMPLICIT_CONV1=0 restore the col paths.
    imp = os.environ.get("LO_IMPLICIT_CONV", "1") == "1"
    imp1 = os.environ.get("LO_IMPLICIT_CONV1", "1") == "1"
    layers = [
        Conv2dNHWC("conv1", 1, c1, 5, 5, relu=True, first=True,
                   implicit=imp1),                                  # 28 -> 24
        MaxPool2dNHWC(2),                                           # 24 -> 12
        Conv2dNHWC("conv2", c1, c2, 5, 5, relu=True, implicit=imp),  # 12 -> 8
        MaxPool2dNHWC(2),                                           # 8 -> 4
        Flatten(),
        Linear("fc1", 4 * 4 * c2, fc_width, relu=True),
        Linear("fc2", fc_width, CPAD, relu=False),
    ]
    return SequentialClassifier(layers, NUM_CLASSES, device=device, seed=seed)
