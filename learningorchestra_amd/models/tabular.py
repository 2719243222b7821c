"""Tabular classifier family for the builder verb — the MLlib classifier set
{lr, dt, rf, gb, nb} (reference builder.py:55-61) on MI355X-native engines.

* lr — logistic regression on the explicit-backward engine (MFMA GEMM +
  fused softmax-CE + fused SGD kernels on GPU; torch fp32 reference on CPU);
* nb — Gaussian naive Bayes (closed-form per-class mean/var — the
  "segmented reduction" of SURVEY §2.9, done as batched tensor reductions);
* dt / rf / gb — histogram tree ensembles (trees.py, tree_hist HIP kernel).

All expose the sklearn-ish fit / predict / predict_proba surface the
reflective executors and builder drive.
"""
from __future__ import annotations

import math
from typing import Optional

import numpy as np
import torch

from ..engine.layers import Linear, SequentialClassifier
from ..engine.trainer import Trainer, make_sgd
from .trees import DecisionTreeClassifier, GBTClassifier, RandomForestClassifier


def _pick_device(device: Optional[str]) -> str:
    if device:
        return device
    return "cuda" if torch.cuda.is_available() else "cpu"


def _pad8(n: int) -> int:
    return (n + 7) // 8 * 8


class LogisticRegressionClassifier:
    """Multinomial LR = one padded Linear layer + fused softmax-CE, trained
    with mini-batch momentum SGD on the flat-arena engine."""

    def __init__(self, epochs: int = 40, batch_size: int = 8192, lr: float = 0.1,
                 device: Optional[str] = None, seed: int = 0,
                 standardize: bool = True):
        self.epochs = epochs
        self.batch_size = batch_size
        self.lr = lr
        self.device = _pick_device(device)
        self.seed = seed
        self.standardize = standardize
        self.model: Optional[SequentialClassifier] = None
        self._mu = self._sigma = None
        self.classes_: Optional[np.ndarray] = None

    def _prep(self, X) -> torch.Tensor:
        X = torch.as_tensor(np.asarray(X, dtype=np.float32))
        X = torch.nan_to_num(X, nan=0.0)
        if self.standardize and self._mu is not None:
            X = (X - self._mu) / self._sigma
        # pad feature dim for 16-B-aligned GEMM rows
        fpad = _pad8(X.shape[1])
        if fpad != X.shape[1]:
            X = torch.nn.functional.pad(X, (0, fpad - X.shape[1]))
        return X.to(self.device, torch.bfloat16)

    def fit(self, X, y) -> "LogisticRegressionClassifier":
        Xt = torch.as_tensor(np.asarray(X, dtype=np.float32))
        Xt = torch.nan_to_num(Xt, nan=0.0)
        if self.standardize:
            self._mu = Xt.mean(0, keepdim=True)
            self._sigma = Xt.std(0, keepdim=True).clamp(min=1e-6)
        yt = np.asarray(y)
        self.classes_ = np.unique(yt)
        ymap = {c: i for i, c in enumerate(self.classes_)}
        yi = torch.as_tensor([ymap[v] for v in yt], dtype=torch.long,
                             device=self.device)
        ncls = len(self.classes_)
        cpad = max(8, _pad8(ncls))
        Xb = self._prep(X)
        fdim = Xb.shape[1]
        self.model = SequentialClassifier(
            [Linear("lr.fc", fdim, cpad)], ncls, device=self.device,
            seed=self.seed)
        trainer = Trainer(self.model, make_sgd(self.model, lr=self.lr,
                                               momentum=0.9), self.device)
        n = Xb.shape[0]
        g = torch.Generator(device="cpu").manual_seed(self.seed)
        bs = min(self.batch_size, n)
        for _ in range(self.epochs):
            perm = torch.randperm(n, generator=g).to(self.device)
            for i in range(0, n - bs + 1, bs):
                sel = perm[i:i + bs]
                trainer.step_async(Xb[sel].contiguous(), yi[sel].contiguous())
        return self

    def _logits(self, X) -> torch.Tensor:
        return self.model.forward(self._prep(X)).float()

    def predict(self, X):
        with torch.no_grad():
            am = self._logits(X)[:, :len(self.classes_)].argmax(1).cpu().numpy()
        return self.classes_[am]

    def predict_proba(self, X):
        with torch.no_grad():
            p = torch.softmax(self._logits(X)[:, :len(self.classes_)], 1)
        return p.cpu().numpy()

    def score(self, X, y):
        return float((self.predict(X) == np.asarray(y)).mean())


class GaussianNBClassifier:
    """Gaussian naive Bayes via per-class mean/variance reductions."""

    def __init__(self, device: Optional[str] = None, var_smoothing: float = 1e-9):
        self.device = _pick_device(device)
        self.var_smoothing = var_smoothing

    def fit(self, X, y) -> "GaussianNBClassifier":
        X = torch.as_tensor(np.asarray(X, dtype=np.float32), device=self.device)
        X = torch.nan_to_num(X, nan=0.0)
        yt = np.asarray(y)
        self.classes_ = np.unique(yt)
        ymap = {c: i for i, c in enumerate(self.classes_)}
        yi = torch.as_tensor([ymap[v] for v in yt], device=self.device)
        C, F = len(self.classes_), X.shape[1]
        onehot = torch.nn.functional.one_hot(yi, C).float()          # [N,C]
        counts = onehot.sum(0).clamp(min=1)                           # [C]
        self.mu = (onehot.T @ X) / counts.unsqueeze(1)                # [C,F]
        ex2 = (onehot.T @ X.square()) / counts.unsqueeze(1)
        self.var = (ex2 - self.mu.square()).clamp(min=0) + self.var_smoothing
        self.prior = (counts / counts.sum()).log()
        return self

    def _joint(self, X) -> torch.Tensor:
        X = torch.as_tensor(np.asarray(X, dtype=np.float32), device=self.device)
        X = torch.nan_to_num(X, nan=0.0)
        diff = X.unsqueeze(1) - self.mu.unsqueeze(0)                  # [N,C,F]
        ll = -0.5 * ((diff.square() / self.var).sum(-1)
                     + self.var.log().sum(-1) + self.mu.shape[1] * math.log(2 * math.pi))
        return ll + self.prior

    def predict(self, X):
        return self.classes_[self._joint(X).argmax(1).cpu().numpy()]

    def predict_proba(self, X):
        return torch.softmax(self._joint(X), 1).cpu().numpy()


def make_classifier(kind: str, device: Optional[str] = None, **kw):
    kind = kind.lower()
    if kind == "lr":
        return LogisticRegressionClassifier(device=device, **kw)
    if kind == "nb":
        return GaussianNBClassifier(device=device, **kw)
    if kind == "dt":
        return DecisionTreeClassifier(device=device, **kw)
    if kind == "rf":
        return RandomForestClassifier(device=device, **kw)
    if kind == "gb":
        return GBTClassifier(device=device, **kw)
    if kind == "mlp":
        return MLPClassifier(device=device, **kw)
    raise ValueError(f"unknown classifier '{kind}' (use lr/dt/rf/gb/nb/mlp)")


class MLPClassifier:
    """Configurable multi-layer perceptron on the engine (native analog of
    sklearn.neural_network.MLPClassifier / a keras Dense stack): ReLU hidden
    layers + padded softmax head, every op a gfx950 kernel on GPU."""

    def __init__(self, hidden=(256, 128), epochs: int = 20,
                 batch_size: int = 8192, lr: float = 0.05, momentum: float = 0.9,
                 device: Optional[str] = None, seed: int = 0,
                 standardize: bool = True):
        self.hidden = tuple(hidden)
        self.epochs = epochs
        self.batch_size = batch_size
        self.lr, self.momentum = lr, momentum
        self.device = _pick_device(device)
        self.seed = seed
        self.standardize = standardize
        self.model: Optional[SequentialClassifier] = None
        self._mu = self._sigma = None
        self.classes_: Optional[np.ndarray] = None

    def _prep(self, X) -> torch.Tensor:
        X = torch.as_tensor(np.asarray(X, dtype=np.float32))
        X = torch.nan_to_num(X, nan=0.0)
        if self.standardize and self._mu is not None:
            X = (X - self._mu) / self._sigma
        fpad = _pad8(X.shape[1])
        if fpad != X.shape[1]:
            X = torch.nn.functional.pad(X, (0, fpad - X.shape[1]))
        return X.to(self.device, torch.bfloat16)

    def fit(self, X, y, epochs: Optional[int] = None) -> "MLPClassifier":
        Xt = torch.as_tensor(np.asarray(X, dtype=np.float32))
        Xt = torch.nan_to_num(Xt, nan=0.0)
        if self.standardize:
            self._mu = Xt.mean(0, keepdim=True)
            self._sigma = Xt.std(0, keepdim=True).clamp(min=1e-6)
        yt = np.asarray(y)
        self.classes_ = np.unique(yt)
        ymap = {c: i for i, c in enumerate(self.classes_)}
        yi = torch.as_tensor([ymap[v] for v in yt], dtype=torch.long,
                             device=self.device)
        Xb = self._prep(X)
        dims = [Xb.shape[1], *[_pad8(h) for h in self.hidden]]
        ncls = len(self.classes_)
        cpad = max(8, _pad8(ncls))
        layers = [Linear(f"mlp.l{i}", dims[i], dims[i + 1], relu=True)
                  for i in range(len(dims) - 1)]
        layers.append(Linear("mlp.head", dims[-1], cpad))
        self.model = SequentialClassifier(layers, ncls, device=self.device,
                                          seed=self.seed)
        trainer = Trainer(self.model, make_sgd(self.model, lr=self.lr,
                                               momentum=self.momentum),
                          self.device)
        n = Xb.shape[0]
        bs = min(self.batch_size, n)
        g = torch.Generator(device="cpu").manual_seed(self.seed)
        for _ in range(int(epochs or self.epochs)):
            perm = torch.randperm(n, generator=g).to(self.device)
            for i in range(0, n - bs + 1, bs):
                sel = perm[i:i + bs]
                trainer.step_async(Xb[sel].contiguous(), yi[sel].contiguous())
        return self

    def predict(self, X):
        with torch.no_grad():
            logits = self.model.forward(self._prep(X)).float()
        return self.classes_[logits[:, :len(self.classes_)].argmax(1).cpu().numpy()]

    def predict_proba(self, X):
        with torch.no_grad():
            logits = self.model.forward(self._prep(X)).float()
        return torch.softmax(logits[:, :len(self.classes_)], 1).cpu().numpy()

    def score(self, X, y):
        return float((self.predict(X) == np.asarray(y)).mean())
