#!/usr/bin/env python3
"""Flagship benchmark: MNIST-CNN bf16 training throughput (BASELINE.json
headline metric: "samples/sec MNIST-CNN train at 1/2/4/8 MI355X").

Single node, one process per GPU over RCCL (torch.distributed "nccl" backend
on ROCm), weak scaling: per-GPU batch fixed, whole-job samples/sec reported.

  python bench.py --gpus 1 --steps 50 --warmup 10
  python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
      --master-addr 127.0.0.1 bench.py --gpus 8 --steps 50 --warmup 10

Synthetic MNIST-shaped data (28x28x1, random), random-init weights — there
is no dataset network access. Every hot op in the timed region
is a hand-written gfx950 HIP kernel (learningorchestra_amd/csrc); the full
step (fwd+bwd+all-reduce+fused SGD) is inside the timing, captured in a
hipGraph when world_size == 1.
"""
from __future__ import annotations

import argparse
import json
import math
import os
import time

import torch


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=50)
    ap.add_argument("--warmup", type=int, default=10)
    ap.add_argument("--batch", type=int, default=0, help="per-GPU batch (0 = model default)")
    ap.add_argument("--model", default="mnist-cnn",
                    choices=["mnist-cnn", "textcnn", "resnet50", "gbt"],
                    help="flagship = mnist-cnn (BASELINE.json headline)")
    ap.add_argument("--rows", type=int, default=10_000_000,
                    help="gbt: total rows across all ranks (BASELINE cfg 4)")
    ap.add_argument("--no-graph", action="store_true")
    # 0.01: stable for 300+ step soaks on the fixed synthetic batch. 0.05
    # diverges past ~200 iterations and 0.02 is borderline (atomics make
    # the trajectory nondeterministic run to run; observed one divergence
    # in three 300-step soaks). Throughput is lr-independent.
    ap.add_argument("--lr", type=float, default=0.01)
    args = ap.parse_args()

    from learningorchestra_amd.parallel import (barrier, get_rank,
                                                get_world_size,
                                                init_distributed)
    from learningorchestra_amd.engine.trainer import Trainer, make_sgd
    from learningorchestra_amd.models.mnist_cnn import build_mnist_cnn
    from learningorchestra_amd.models.textcnn import build_textcnn
    from learningorchestra_amd.data.synthetic import (imagenet_batch,
                                                      imdb_batch, mnist_batch)

    local_rank = init_distributed()
    world = get_world_size()
    rank = get_rank()
    use_gpu = torch.cuda.is_available()
    device = f"cuda:{local_rank}" if use_gpu else "cpu"
    if use_gpu:
        torch.cuda.set_device(local_rank)

    if args.model == "gbt":
        # BASELINE config 4: GBT on the 10M-row synthetic tabular matrix;
        # a "step" is one boosting iteration (tree fit on the row shard,
        # per-level histograms all-reduced across ranks)
        _bench_gbt(args, rank, world, device, use_gpu)
        return

    if args.model == "mnist-cnn":
        batch = args.batch or 32768
        model = build_mnist_cnn(device, seed=0)
        model_desc = "MNIST-CNN (LeNet-style, conv32-conv64-fc256)"
        extra_cfg = {"image": "28x28x1"}
    elif args.model == "textcnn":
        batch = args.batch or 4096
        model = build_textcnn(device, seed=0)
        model_desc = "TextCNN (IMDb sentiment, emb128, filters 128 x k3/4/5)"
        extra_cfg = {"seq_len": 256, "vocab": 20000}
    else:
        from learningorchestra_amd.models.resnet import build_resnet50
        batch = args.batch or 512
        model = build_resnet50(device, seed=0)
        model_desc = "ResNet-50 (bottleneck v1, 1000 classes)"
        extra_cfg = {"image": "224x224x3"}
    # graph capture at every world size: world==1 captures the whole step;
    # world>1 runs split-graph mode (fwd+bwd graph -> flat RCCL all-reduce
    # outside the graphs -> optimizer graph), so the multi-GPU step is not
    # launch-bound (r1 VERDICT weak #4)
    use_graph = use_gpu and not args.no_graph
    trainer = Trainer(model, make_sgd(model, lr=args.lr), device=device,
                      use_graph=use_graph)
    # the trainer disables graphs when ranks time-share one GPU (preemption
    # corruption, see engine/trainer.py) — report the EFFECTIVE state
    use_graph = trainer.use_graph

    if args.model == "mnist-cnn":
        x, y = mnist_batch(batch, device=device, dtype=torch.bfloat16,
                           seed=1234 + rank)
    elif args.model == "textcnn":
        x, y = imdb_batch(batch, seq_len=256, vocab=20000, seed=1234 + rank)
        x, y = x.to(device), y.to(device)
    else:
        x, y = imagenet_batch(batch, device=device, dtype=torch.bfloat16,
                              seed=1234 + rank)

    # LO_BENCH_TRACE=1: per-50-step |w|/|g| maxima on stderr — diagnostic
    # for the rare non-finite loss seen in 300-step soaks (PERFORMANCE.md
    # "Open issue"); prints outside the timed region only when enabled
    trace = os.environ.get("LO_BENCH_TRACE") == "1" and rank == 0

    def _trace(tag):
        if trace:
            import sys
            if use_gpu:
                torch.cuda.synchronize()
            a = model.arena
            print(f"[trace {tag}] |w|max {float(a.master.abs().max()):.4e} "
                  f"|g|max {float(a.grad.abs().max()):.4e} "
                  f"loss_sum {float(model.loss_sum):.4e}", file=sys.stderr,
                  flush=True)

    for _ in range(args.warmup):
        trainer.step_async(x, y)
    _trace("warmup")

    barrier()
    if use_gpu:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for i in range(args.steps):
        trainer.step_async(x, y)
        if trace and i % 50 == 49:
            _trace(f"step{i + 1}")
    if use_gpu:
        torch.cuda.synchronize()
    barrier()
    elapsed = time.perf_counter() - t0

    # MAX over ranks (slowest rank defines the job)
    if world > 1:
        import torch.distributed as dist
        t = torch.tensor([elapsed], dtype=torch.float64,
                         device=device if use_gpu else "cpu")
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = t.item()

    loss = model.loss_sum.item() / batch
    # NaN/Infinity are not valid JSON (strict parsers reject the line)
    loss = round(loss, 4) if math.isfinite(loss) else None
    samples = batch * world * args.steps
    value = samples / elapsed
    if rank == 0:
        print(json.dumps({
            "metric": "samples/sec",
            "value": value,
            "unit": "samples/sec",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": elapsed / args.steps * 1000.0,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16",
            "data": "synthetic",
            "config": {"model": model_desc,
                       "global_batch": batch * world,
                       "per_gpu_batch": batch,
                       **extra_cfg,
                       "parallelism": f"dp{world}",
                       "graph_capture": use_graph,
                       "final_loss": loss},
        }))


def _bench_gbt(args, rank: int, world: int, device: str,
               use_gpu: bool) -> None:
    import torch.distributed as dist

    from learningorchestra_amd.data.synthetic import tabular
    from learningorchestra_amd.models.trees import TreeLearner, quantize
    from learningorchestra_amd.parallel import barrier

    n_local = args.rows // world
    X, y = tabular(n_local, 28, seed=100 + rank)
    X, y = X.to(device), y.to(device).float()
    binned, _ = quantize(X)
    learner = TreeLearner(max_depth=5, lr=0.2)
    raw = torch.zeros_like(y)
    trees = []

    def boost_step():
        p = torch.sigmoid(raw)
        tree = learner.fit(binned, p - y, p * (1 - p))
        trees.append(tree)
        raw.add_(tree.predict_binned(binned))

    for _ in range(args.warmup):
        boost_step()
    barrier()
    if use_gpu:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        boost_step()
    if use_gpu:
        torch.cuda.synchronize()
    barrier()
    elapsed = time.perf_counter() - t0
    if world > 1:
        t = torch.tensor([elapsed], dtype=torch.float64,
                         device=device if use_gpu else "cpu")
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = t.item()
    p = torch.sigmoid(raw)
    acc = float(((p > 0.5).float() == y).float().mean())
    if rank == 0:
        print(json.dumps({
            "metric": "rows/sec (GBT boosting, 10M-row synthetic tabular)",
            "value": n_local * world * args.steps / elapsed,
            "unit": "rows/sec",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": elapsed / args.steps * 1000.0,
            "higher_is_better": True,
            "scaling": "strong",
            "vs_baseline": None,
            "dtype": "fp32",
            "data": "synthetic",
            "config": {"model": "GBTClassifier (histogram trees, depth 5)",
                       "rows": n_local * world, "features": 28,
                       "parallelism": f"dp{world}",
                       "train_accuracy": round(acc, 4)},
        }))


if __name__ == "__main__":
    main()
