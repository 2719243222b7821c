"""Vision model zoo — keras-applications-surface wrappers (the reference
instantiated ``tensorflow.keras.applications.ResNet50`` through the model
verb, model_image/model.py:136-142; translate_module_path routes that here).
"""
from __future__ import annotations

from typing import Optional

import numpy as np
import torch

from ..engine.trainer import Trainer, make_sgd
from .resnet import build_resnet50


class ResNet50:
    """MI355X-native ResNet-50 with the keras-ish fit/predict/evaluate
    surface (weights are random-init; there is no network for pretrained
    checkpoints — load them via load_state_dict if provided on disk)."""

    def __init__(self, weights: Optional[str] = None, classes: int = 1000,
                 seed: int = 0, lr: float = 0.05, momentum: float = 0.9,
                 batch_size: int = 64, device: Optional[str] = None):
        self.device = device or ("cuda" if torch.cuda.is_available() else "cpu")
        self.model = build_resnet50(self.device, seed=seed, num_classes=classes)
        self.trainer = Trainer(self.model, make_sgd(self.model, lr=lr,
                                                    momentum=momentum),
                               device=self.device)
        self.batch_size = batch_size
        if weights and weights != "imagenet":
            sd = torch.load(weights, map_location="cpu", weights_only=True)
            self.model.load_state_dict(sd)

    def _to_x(self, x) -> torch.Tensor:
        x = torch.as_tensor(np.asarray(x, dtype=np.float32))
        if x.dim() == 4 and x.shape[1] == 3:       # NCHW -> NHWC
            x = x.permute(0, 2, 3, 1).contiguous()
        return x.to(self.device, torch.bfloat16)

    def fit(self, x=None, y=None, epochs: int = 1,
            batch_size: Optional[int] = None, verbose: int = 0) -> dict:
        xt = self._to_x(x)
        yt = torch.as_tensor(np.asarray(y), dtype=torch.long, device=self.device)
        bs = min(batch_size or self.batch_size, xt.shape[0])
        self.model.set_training(True)
        losses = []
        for _ in range(int(epochs)):
            perm = torch.randperm(xt.shape[0], device=self.device)
            total = seen = 0.0
            for i in range(0, xt.shape[0] - bs + 1, bs):
                sel = perm[i:i + bs]
                loss, _ = self.trainer.step(xt[sel].contiguous(),
                                            yt[sel].contiguous())
                total += loss * bs
                seen += bs
            losses.append(total / max(seen, 1))
        return {"loss": losses}

    @torch.no_grad()
    def predict(self, x, batch_size: Optional[int] = None) -> np.ndarray:
        xt = self._to_x(x)
        bs = batch_size or self.batch_size
        self.model.set_training(False)
        outs = []
        for i in range(0, xt.shape[0], bs):
            outs.append(self.model.predict(xt[i:i + bs].contiguous())
                        .cpu().numpy())
        self.model.set_training(True)
        return np.concatenate(outs) if outs else np.empty(0, dtype=np.int32)

    @torch.no_grad()
    def evaluate(self, x, y, batch_size: Optional[int] = None) -> dict:
        preds = self.predict(x, batch_size)
        yt = np.asarray(y)
        return {"accuracy": float((preds == yt.astype(preds.dtype)).mean()),
                "n": int(len(preds))}

    def state_dict(self):
        return self.model.state_dict()

    def load_state_dict(self, sd):
        self.model.load_state_dict(sd)
