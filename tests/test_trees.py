"""Histogram-tree family tests (CPU reference path; GPU kernel covered in
test_gpu_trees.py)."""
import numpy as np
import pytest
import torch

from learningorchestra_amd.data.synthetic import tabular
from learningorchestra_amd.models.trees import (DecisionTreeClassifier,
                                                GBTClassifier,
                                                RandomForestClassifier,
                                                TreeLearner, build_histograms,
                                                quantize)


def test_quantize_bins():
    X, _ = tabular(5000, 4, seed=0)
    binned, edges = quantize(X)
    assert binned.dtype == torch.uint8 and binned.shape == X.shape
    assert edges.shape == (4, 254)
    # bins roughly uniform for continuous data
    counts = torch.bincount(binned[:, 0].long(), minlength=255).float()
    assert counts[counts > 0].std() / counts.mean() < 2.0


def test_histogram_totals():
    torch.manual_seed(0)
    N, F = 2000, 3
    binned = torch.randint(0, 255, (N, F), dtype=torch.uint8)
    node_of = torch.randint(0, 4, (N,), dtype=torch.int32)
    node_of[:100] = -1  # inactive
    grad = torch.randn(N)
    hess = torch.rand(N)
    hist = build_histograms(binned, node_of, grad, hess, 4)
    active = node_of >= 0
    assert abs(hist[..., 0].sum().item() / F - grad[active].sum().item()) < 1e-2
    # per-node totals
    for nd in range(4):
        m = node_of == nd
        assert abs(hist[nd, 0, :, 0].sum().item() - grad[m].sum().item()) < 1e-2


def test_single_tree_learns_split():
    # y = 1[x0 > 0]: one split on feature 0 should nail it
    X, _ = tabular(4000, 3, seed=1)
    y = (X[:, 0] > 0).float()
    binned, edges = quantize(X)
    p = y.mean()
    grad = torch.full_like(y, p) - y
    hess = torch.ones_like(y)
    tree = TreeLearner(max_depth=2, lr=1.0).fit(binned, grad, hess)
    assert tree.feature[0].item() == 0  # root splits on x0
    pred = tree.predict_binned(binned)
    acc = (((p + pred) > 0.5).float() == y).float().mean().item()
    assert acc > 0.95


@pytest.mark.parametrize("cls,kw", [
    (GBTClassifier, {"n_trees": 20, "max_depth": 4}),
    (RandomForestClassifier, {"n_trees": 10, "max_depth": 6}),
    (DecisionTreeClassifier, {"max_depth": 8}),
])
def test_classifier_beats_majority(cls, kw):
    X, y = tabular(8000, 8, seed=2)
    Xtr, ytr = X[:6000].numpy(), y[:6000].numpy()
    Xte, yte = X[6000:].numpy(), y[6000:].numpy()
    clf = cls(device="cpu", **kw).fit(Xtr, ytr)
    acc = (clf.predict(Xte).astype(int) == yte.astype(int)).mean()
    majority = max(yte.mean(), 1 - yte.mean())
    assert acc > majority + 0.05, (acc, majority)
    proba = clf.predict_proba(Xte)
    assert proba.shape == (2000, 2)
    assert np.allclose(proba.sum(1), 1.0, atol=1e-3)
