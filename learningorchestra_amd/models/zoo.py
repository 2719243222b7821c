"""Native model zoo — the instantiable classes the model verb's reflective
runtime resolves (the reference instantiated tf.keras classes the same way,
model_image/model.py:112-156; ``tensorflow.keras...`` module paths translate
here via models/__init__.translate_module_path).

Each model exposes the keras-ish fit / predict / evaluate surface the
binary-executor verbs (train/tune/evaluate/predict) call by method name, and
runs on the MI355X engine (HIP kernels on GPU, torch fp32 reference on CPU).
"""
from __future__ import annotations

from typing import Optional, Sequence

import numpy as np
import torch

from ..engine.trainer import Trainer, make_sgd
from .mnist_cnn import build_mnist_cnn
from .tabular import (GaussianNBClassifier, LogisticRegressionClassifier,  # noqa: F401 - re-export
                      MLPClassifier)
from .trees import (DecisionTreeClassifier, GBTClassifier,  # noqa: F401 - re-export
                    RandomForestClassifier)


def _default_device(device: Optional[str]) -> str:
    if device:
        return device
    return "cuda" if torch.cuda.is_available() else "cpu"


class MnistCNN:
    """LeNet-style CNN for 28x28x1 images (the flagship engine model)."""

    def __init__(self, seed: int = 0, channels: Sequence[int] = (32, 64),
                 fc_width: int = 256, lr: float = 0.05, momentum: float = 0.9,
                 batch_size: int = 512, device: Optional[str] = None):
        self.device = _default_device(device)
        # rebuild spec: the artifact store persists this + state_dict so any
        # DDP rank can reconstruct the model on its own device
        self.lo_spec = {"seed": seed, "channels": list(channels),
                        "fc_width": fc_width, "lr": lr, "momentum": momentum,
                        "batch_size": batch_size}
        self.lr, self.momentum, self.batch_size = lr, momentum, batch_size
        self.model = build_mnist_cnn(self.device, seed=seed,
                                     channels=tuple(channels),
                                     fc_width=fc_width)
        self.trainer = Trainer(self.model,
                               make_sgd(self.model, lr=lr, momentum=momentum),
                               device=self.device)
        self.history: dict = {}

    # -- data shaping --------------------------------------------------------
    def _to_x(self, x) -> torch.Tensor:
        if hasattr(x, "to_numpy"):
            x = x.to_numpy()
        x = torch.as_tensor(np.asarray(x, dtype=np.float32))
        if x.dim() == 2 and x.shape[1] == 784:
            x = x.view(-1, 28, 28, 1)
        elif x.dim() == 3:
            x = x.unsqueeze(-1)
        elif x.dim() == 4 and x.shape[1] == 1:   # NCHW -> NHWC
            x = x.permute(0, 2, 3, 1).contiguous()
        return x.to(self.device, torch.bfloat16)

    def _to_y(self, y) -> torch.Tensor:
        if hasattr(y, "to_numpy"):
            y = y.to_numpy()
        return torch.as_tensor(np.asarray(y), dtype=torch.long,
                               device=self.device)

    # -- keras-ish surface ---------------------------------------------------
    def fit(self, x=None, y=None, epochs: int = 1,
            batch_size: Optional[int] = None, verbose: int = 0) -> dict:
        xt, yt = self._to_x(x), self._to_y(y)
        bs = min(batch_size or self.batch_size, xt.shape[0])
        n = xt.shape[0]
        losses, accs = [], []
        for _ in range(int(epochs)):
            perm = torch.randperm(n, device=self.device)
            ep_loss = ep_corr = seen = 0.0
            for i in range(0, n - bs + 1, bs):
                sel = perm[i:i + bs]
                loss, acc = self.trainer.step(xt[sel].contiguous(),
                                              yt[sel].contiguous())
                ep_loss += loss * bs
                ep_corr += acc * bs
                seen += bs
            losses.append(ep_loss / max(seen, 1))
            accs.append(ep_corr / max(seen, 1))
        self.history = {"loss": losses, "accuracy": accs}
        return self.history

    @torch.no_grad()
    def predict(self, x, batch_size: int = 8192) -> np.ndarray:
        xt = self._to_x(x)
        outs = []
        for i in range(0, xt.shape[0], batch_size):
            outs.append(self.model.predict(xt[i:i + batch_size].contiguous())
                        .cpu().numpy())
        return np.concatenate(outs) if outs else np.empty(0, dtype=np.int32)

    @torch.no_grad()
    def evaluate(self, x, y, batch_size: int = 8192) -> dict:
        preds = self.predict(x, batch_size)
        yt = np.asarray(y) if not hasattr(y, "to_numpy") else y.to_numpy()
        acc = float((preds == yt.astype(preds.dtype)).mean()) if len(preds) else 0.0
        return {"accuracy": acc, "n": int(len(preds))}

    # -- persistence hooks (ArtifactStore) -----------------------------------
    def state_dict(self):
        return self.model.state_dict()

    def load_state_dict(self, sd):
        self.model.load_state_dict(sd)


class TextCNNClassifier:
    """Keras-surface wrapper over the TextCNN engine model (IMDb config)."""

    def __init__(self, vocab: int = 20000, emb_dim: int = 128,
                 filters: int = 128, kernel_sizes=(3, 4, 5),
                 num_classes: int = 2, seed: int = 0, lr: float = 0.05,
                 batch_size: int = 512, device: Optional[str] = None):
        from .textcnn import build_textcnn
        self.device = _default_device(device)
        self.lo_spec = {"vocab": vocab, "emb_dim": emb_dim,
                        "filters": filters,
                        "kernel_sizes": list(kernel_sizes),
                        "num_classes": num_classes, "seed": seed, "lr": lr,
                        "batch_size": batch_size}
        self.model = build_textcnn(self.device, seed=seed, vocab=vocab,
                                   emb_dim=emb_dim, filters=filters,
                                   kernel_sizes=tuple(kernel_sizes),
                                   num_classes=num_classes)
        self.trainer = Trainer(self.model, make_sgd(self.model, lr=lr),
                               device=self.device)
        self.batch_size = batch_size

    def _to_ids(self, x) -> torch.Tensor:
        if hasattr(x, "to_numpy"):
            x = x.to_numpy()
        return torch.as_tensor(np.asarray(x), dtype=torch.long,
                               device=self.device)

    def fit(self, x=None, y=None, epochs: int = 1,
            batch_size: Optional[int] = None, verbose: int = 0) -> dict:
        ids = self._to_ids(x)
        yt = torch.as_tensor(np.asarray(y), dtype=torch.long,
                             device=self.device)
        bs = min(batch_size or self.batch_size, ids.shape[0])
        losses = []
        for _ in range(int(epochs)):
            perm = torch.randperm(ids.shape[0], device=self.device)
            tot = seen = 0.0
            for i in range(0, ids.shape[0] - bs + 1, bs):
                sel = perm[i:i + bs]
                loss, _ = self.trainer.step(ids[sel].contiguous(),
                                            yt[sel].contiguous())
                tot += loss * bs
                seen += bs
            losses.append(tot / max(seen, 1))
        return {"loss": losses}

    @torch.no_grad()
    def predict(self, x, batch_size: int = 4096) -> np.ndarray:
        ids = self._to_ids(x)
        outs = []
        for i in range(0, ids.shape[0], batch_size):
            outs.append(self.model.predict(ids[i:i + batch_size].contiguous())
                        .cpu().numpy())
        return np.concatenate(outs) if outs else np.empty(0, dtype=np.int32)

    @torch.no_grad()
    def evaluate(self, x, y, batch_size: int = 4096) -> dict:
        preds = self.predict(x, batch_size)
        yt = np.asarray(y)
        return {"accuracy": float((preds == yt.astype(preds.dtype)).mean()),
                "n": int(len(preds))}

    def state_dict(self):
        return self.model.state_dict()

    def load_state_dict(self, sd):
        self.model.load_state_dict(sd)
