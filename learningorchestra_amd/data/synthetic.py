"""Synthetic datasets for the BASELINE.json configs.

There is no network access for real datasets (Titanic/MNIST/IMDb,
/root/reference/README.md:63), so every benchmark/test runs on synthetic data
of the SAME SHAPE with random-init weights, as BASELINE.md specifies.
"""
from __future__ import annotations

import io
from typing import Tuple

import torch


def mnist_batch(batch: int, device="cpu", dtype=torch.float32, nhwc: bool = True,
                seed: int = None, generator=None) -> Tuple[torch.Tensor, torch.Tensor]:
    """Synthetic MNIST: images [B,28,28,1] (NHWC) in [0,1), labels [B] in [0,10)."""
    g = generator
    if g is None and seed is not None:
        g = torch.Generator(device="cpu").manual_seed(seed)
    shape = (batch, 28, 28, 1) if nhwc else (batch, 1, 28, 28)
    x = torch.rand(shape, generator=g, dtype=torch.float32)
    y = torch.randint(0, 10, (batch,), generator=g)
    return x.to(device=device, dtype=dtype), y.to(device)


def imdb_batch(batch: int, seq_len: int = 256, vocab: int = 20000, device="cpu",
               seed: int = None) -> Tuple[torch.Tensor, torch.Tensor]:
    """Synthetic IMDb: token ids [B,S], binary labels [B]."""
    g = torch.Generator(device="cpu").manual_seed(seed) if seed is not None else None
    x = torch.randint(0, vocab, (batch, seq_len), generator=g)
    y = torch.randint(0, 2, (batch,), generator=g)
    return x.to(device), y.to(device)


def imagenet_batch(batch: int, device="cpu", dtype=torch.float32, nhwc: bool = True,
                   seed: int = None) -> Tuple[torch.Tensor, torch.Tensor]:
    """Synthetic 224x224 images for the ResNet-50 fine-tune config."""
    g = torch.Generator(device="cpu").manual_seed(seed) if seed is not None else None
    shape = (batch, 224, 224, 3) if nhwc else (batch, 3, 224, 224)
    x = torch.rand(shape, generator=g, dtype=torch.float32)
    y = torch.randint(0, 1000, (batch,), generator=g)
    return x.to(device=device, dtype=dtype), y.to(device)


def tabular(rows: int, features: int = 28, seed: int = 0,
            device="cpu") -> Tuple[torch.Tensor, torch.Tensor]:
    """Synthetic tabular matrix for the GBT/RF configs (10M-row class)."""
    g = torch.Generator(device="cpu").manual_seed(seed)
    x = torch.randn(rows, features, generator=g)
    w = torch.randn(features, generator=g)
    logits = x @ w + 0.5 * torch.sin(x[:, 0] * 3.0)
    y = (logits > 0).to(torch.float32)
    return x.to(device), y.to(device)


def tabular_multiclass(rows: int, features: int = 64, n_classes: int = 10,
                       seed: int = 0, device="cpu"
                       ) -> Tuple[torch.Tensor, torch.Tensor]:
    """Multiclass blobs (MNIST-like 10-class tabular shape): class-dependent
    feature means + unit noise — learnable by gini trees but not trivial."""
    g = torch.Generator(device="cpu").manual_seed(seed)
    y = torch.randint(0, n_classes, (rows,), generator=g)
    centers = torch.randn(n_classes, features, generator=g) * 1.6
    x = centers[y] + torch.randn(rows, features, generator=g)
    return x.to(device), y.to(device)


def titanic_csv(rows: int = 891, seed: int = 0) -> str:
    """A Titanic-shaped CSV (same columns as the canonical dataset) as text,
    for the Dataset->Transform->LogisticRegression plumbing config."""
    import random
    rng = random.Random(seed)
    buf = io.StringIO()
    buf.write("PassengerId,Survived,Pclass,Name,Sex,Age,SibSp,Parch,Ticket,Fare,Cabin,Embarked\n")
    for i in range(1, rows + 1):
        pclass = rng.choice([1, 2, 3])
        sex = rng.choice(["male", "female"])
        age = round(rng.uniform(1, 80), 1) if rng.random() > 0.2 else ""
        sibsp = rng.randint(0, 4)
        parch = rng.randint(0, 4)
        fare = round(rng.uniform(5, 500) / pclass, 4)
        surv = 1 if (sex == "female" or pclass == 1) and rng.random() > 0.35 else (
            1 if rng.random() > 0.8 else 0)
        emb = rng.choice(["S", "C", "Q"])
        buf.write(f"{i},{surv},{pclass},\"Passenger {i}\",{sex},{age},{sibsp},"
                  f"{parch},T{i:06d},{fare},,{emb}\n")
    return buf.getvalue()
