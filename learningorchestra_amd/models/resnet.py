"""ResNet-50 on the explicit-backward NHWC engine — BASELINE.json config 5
("ResNet-50 Model-load + fine-tune on synthetic 224x224, DP 1/2/4/8 GPU
scaling curve").

Standard bottleneck-v1 topology; every hot op is a gfx950 HIP kernel:
* convs: 1x1/s1 as direct MFMA GEMM (no im2col), others im2col + MFMA GEMM,
  backward-data via stride-general gather col2im;
* BN+ReLU fused fwd/bwd (batchnorm.hip); forward joins fused into bn3,
  backward join add fused into conv1's dX GEMM epilogue;
* 3x3/s2 maxpool with padding, global average pool, padded-head linear,
  fused softmax-CE (wave path, 1000 classes).
"""
from __future__ import annotations

from typing import List, Optional, Tuple

import torch

from ..engine.arena import ParamArena
from ..engine.layers import (AvgPoolGlobal, BatchNormReLU, Conv2dNHWC, Linear,
                             MaxPool2dNHWC)
from ..ops import functional as F

NUM_CLASSES = 1000


class Bottleneck:
    expansion = 4

    def __init__(self, name: str, in_c: int, mid: int, stride: int):
        out_c = mid * self.expansion
        self.out_c = out_c
        self.conv1 = Conv2dNHWC(f"{name}.c1", in_c, mid, 1, 1, relu=False, bias=False)
        self.bn1 = BatchNormReLU(f"{name}.bn1", mid)
        self.conv2 = Conv2dNHWC(f"{name}.c2", mid, mid, 3, 3, stride=stride,
                                pad=1, relu=False, bias=False)
        self.bn2 = BatchNormReLU(f"{name}.bn2", mid)
        self.conv3 = Conv2dNHWC(f"{name}.c3", mid, out_c, 1, 1, relu=False, bias=False)
        self.bn3 = BatchNormReLU(f"{name}.bn3", out_c, relu=False)
        self.downsample: Optional[Tuple[Conv2dNHWC, BatchNormReLU]] = None
        if stride != 1 or in_c != out_c:
            self.downsample = (
                Conv2dNHWC(f"{name}.ds", in_c, out_c, 1, 1, stride=stride,
                           relu=False, bias=False),
                BatchNormReLU(f"{name}.dsbn", out_c, relu=False))
        self._z = None
        self._dsum = None

    def layers(self):
        out = [self.conv1, self.bn1, self.conv2, self.bn2, self.conv3, self.bn3]
        if self.downsample:
            out += list(self.downsample)
        return out

    def param_names(self):
        names = []
        for lay in self.layers():
            names += lay.param_names()
        return names

    def build(self, arena: ParamArena) -> None:
        for lay in self.layers():
            lay.build(arena)

    @staticmethod
    def _conv_bn(conv, bn, x):
        # the conv GEMM epilogue fills bn's sum/sumsq scratch (fused bn_stats)
        h = conv.forward(x, stats=bn.scratch(x.device) if x.is_cuda else None)
        return bn.forward(h, stats_ready=conv.stats_filled)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if self.downsample:
            idt = self._conv_bn(self.downsample[0], self.downsample[1], x)
        else:
            idt = x
        h = self._conv_bn(self.conv1, self.bn1, x)
        h = self._conv_bn(self.conv2, self.bn2, h)
        # bottleneck join fused into bn3: z = relu(bn3(conv3) + idt)
        c3 = self.conv3.forward(h, stats=self.bn3.scratch(h.device)
                                if h.is_cuda else None)
        z = self.bn3.forward(c3, stats_ready=self.conv3.stats_filled,
                             residual=idt)
        self._z = z
        if self._dsum is None or self._dsum.shape != z.shape:
            self._dsum = torch.empty_like(z)
        return z

    def backward(self, dz: torch.Tensor) -> torch.Tensor:
        dsum = F.relu_bwd(dz, self._z, out=self._dsum)
        dh = self.bn3.backward(dsum)
        dh = self.conv3.backward(dh)
        dh = self.bn2.backward(dh)
        dh = self.conv2.backward(dh)
        dh = self.bn1.backward(dh)
        if self.downsample:
            d_idt = self.downsample[0].backward(self.downsample[1].backward(dsum))
        else:
            d_idt = dsum
        # branch-grad sum fused into conv1's dX GEMM epilogue (C = A@B + D)
        return self.conv1.backward(dh, dx_addend=d_idt)


class ResNet:
    def __init__(self, block_counts=(3, 4, 6, 3), num_classes: int = NUM_CLASSES,
                 device="cpu", seed: int = 0, width: int = 64):
        self.device = torch.device(device)
        self.num_classes = num_classes
        self.cpad = (num_classes + 7) // 8 * 8
        self.arena = ParamArena(device)

        # stem takes 8 input channels: RGB is zero-padded 3->8 in forward so
        # the 7x7 im2col runs the vectorized path (the C=3 scalar im2col was
        # 3 ms/step, 3.6% — the padded channels are all-zero, their weights
        # receive zero gradient and stay at init)
        self.stem_conv = Conv2dNHWC("stem", 8, width, 7, 7, stride=2, pad=3,
                                    relu=False, first=True, bias=False)
        self.stem_bn = BatchNormReLU("stem.bn", width)
        self.stem_pool = MaxPool2dNHWC(3, stride=2, pad=1)
        self.blocks: List[Bottleneck] = []
        in_c = width
        mids = [width, width * 2, width * 4, width * 8]
        for si, (n, mid) in enumerate(zip(block_counts, mids)):
            for bi in range(n):
                stride = 2 if (si > 0 and bi == 0) else 1
                blk = Bottleneck(f"s{si}b{bi}", in_c, mid, stride)
                self.blocks.append(blk)
                in_c = mid * Bottleneck.expansion
        self.avgpool = AvgPoolGlobal()
        self.fc = Linear("fc", in_c, self.cpad)

        self.stem_conv.build(self.arena)
        self.stem_bn.build(self.arena)
        for blk in self.blocks:
            blk.build(self.arena)
        self.fc.build(self.arena)
        self.arena.finalize(seed)

        self.loss_sum = torch.zeros(1, dtype=torch.float32, device=self.device)
        self.correct = torch.zeros(1, dtype=torch.int32, device=self.device)
        self._dlogits = None

    def _bn_layers(self):
        out = [self.stem_bn]
        for blk in self.blocks:
            out += [l for l in blk.layers() if isinstance(l, BatchNormReLU)]
        return out

    def set_training(self, training: bool) -> None:
        for bn in self._bn_layers():
            bn.training = training

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if x.shape[-1] != 8:  # zero-pad RGB to the stem's 8 channels
            x = torch.nn.functional.pad(x, (0, 8 - x.shape[-1]))
        h = self.stem_conv.forward(x, stats=self.stem_bn.scratch(x.device)
                                   if x.is_cuda else None)
        h = self.stem_bn.forward(h, stats_ready=self.stem_conv.stats_filled)
        h = self.stem_pool.forward(h)
        for blk in self.blocks:
            h = blk.forward(h)
        h = self.avgpool.forward(h)
        return self.fc.forward(h)

    def train_step(self, x: torch.Tensor, y: torch.Tensor,
                   gscale: Optional[float] = None, grad_hook=None):
        if gscale is None:
            gscale = 1.0 / x.shape[0]
        logits = self.forward(x)
        if self._dlogits is None or self._dlogits.shape != logits.shape:
            self._dlogits = torch.empty_like(logits)
        self.loss_sum.zero_()
        self.correct.zero_()
        F.softmax_ce(logits, y, self._dlogits, self.loss_sum, self.correct,
                     cvalid=self.num_classes, gscale=gscale)
        dy = self.fc.backward(self._dlogits)
        if grad_hook:
            grad_hook(self.fc.param_names())
        dy = self.avgpool.backward(dy)
        for blk in reversed(self.blocks):
            dy = blk.backward(dy)
            if grad_hook:
                grad_hook(blk.param_names())
        dy = self.stem_pool.backward(dy)
        dy = self.stem_bn.backward(dy)
        self.stem_conv.backward(dy)
        if grad_hook:
            grad_hook(self.stem_bn.param_names() + self.stem_conv.param_names())
        return self.loss_sum, self.correct

    def post_opt_step(self) -> None:
        self.stem_conv.post_opt_step()
        for blk in self.blocks:
            for lay in blk.layers():
                lay.post_opt_step()
        self.fc.post_opt_step()

    @torch.no_grad()
    def predict(self, x: torch.Tensor) -> torch.Tensor:
        return F.argmax_rows(self.forward(x), self.num_classes)

    # -- checkpoint (arena params + BN running stats) ------------------------
    def state_dict(self):
        sd = self.arena.state_dict()
        for bn in self._bn_layers():
            sd.update(bn.extra_state())
        return sd

    def load_state_dict(self, sd):
        self.arena.load_state_dict({k: v for k, v in sd.items()
                                    if k in self.arena._offsets})
        for bn in self._bn_layers():
            bn.load_extra_state(sd)
        self.post_opt_step()


def build_resnet50(device="cpu", seed: int = 0, num_classes: int = NUM_CLASSES,
                   **kw) -> ResNet:
    """ResNet-50: bottleneck counts (3,4,6,3). The reference loaded
    tf.keras.applications.ResNet50 via the model verb (model_image/
    model.py:136-142); this is the MI355X-native equivalent."""
    return ResNet((3, 4, 6, 3), num_classes=num_classes, device=device,
                  seed=seed, **kw)


def build_resnet18ish(device="cpu", seed: int = 0, **kw) -> ResNet:
    """Small bottleneck net for tests (counts 1,1,1,1)."""
    return ResNet((1, 1, 1, 1), device=device, seed=seed, **kw)
