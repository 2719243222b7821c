#!/usr/bin/env python3
"""BASELINE.json config 4: Spark-MLlib-style GBTClassifier on 10M-row
synthetic tabular, tree-histogram kernel on MI355X.

A "step" is one boosting iteration (one tree: per-level tree_hist builds +
split/partition). Reports rows/sec for the histogram-build hot op and
whole-fit boosting throughput.
"""
import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--rows", type=int, default=10_000_000)
    ap.add_argument("--features", type=int, default=28)
    ap.add_argument("--trees", type=int, default=20)
    ap.add_argument("--depth", type=int, default=5)
    args = ap.parse_args()

    from learningorchestra_amd.data.synthetic import tabular
    from learningorchestra_amd.models.trees import (GBTClassifier,
                                                    build_histograms, quantize)

    device = "cuda" if torch.cuda.is_available() else "cpu"
    X, y = tabular(args.rows, args.features, seed=0)
    X, y = X.to(device), y.to(device)

    # --- isolated histogram-build benchmark (the hot op) -------------------
    binned, _ = quantize(X)
    node_of = torch.randint(0, 8, (args.rows,), dtype=torch.int32, device=device)
    grad = torch.randn(args.rows, device=device)
    hess = torch.rand(args.rows, device=device)
    for _ in range(3):
        build_histograms(binned, node_of, grad, hess, 8)
    if device == "cuda":
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    iters = 10
    for _ in range(iters):
        build_histograms(binned, node_of, grad, hess, 8)
    if device == "cuda":
        torch.cuda.synchronize()
    hist_dt = (time.perf_counter() - t0) / iters
    hist_rows_per_sec = args.rows / hist_dt

    # --- full GBT fit ------------------------------------------------------
    clf = GBTClassifier(n_trees=args.trees, max_depth=args.depth, device=device)
    t0 = time.perf_counter()
    clf.fit(X, y)
    if device == "cuda":
        torch.cuda.synchronize()
    fit_dt = time.perf_counter() - t0
    acc = float((torch.as_tensor(clf.predict(X[:1_000_000])).to(device).float()
                 == y[:1_000_000]).float().mean())

    print(json.dumps({
        "metric": "rows/sec (tree histogram build, 8 nodes x 28 feat x 256 bins)",
        "value": hist_rows_per_sec,
        "hist_ms": hist_dt * 1000,
        "fit_seconds": fit_dt,
        "boost_iters_per_sec": args.trees / fit_dt,
        "rows": args.rows, "features": args.features,
        "trees": args.trees, "depth": args.depth,
        "train_accuracy_1M": acc,
        "device": device,
    }))


if __name__ == "__main__":
    main()
