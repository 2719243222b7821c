"""Multi-process data-parallel tests on CPU (gloo, world_size 2) — the same
code path that runs RCCL over xGMI on the 8-GPU node."""
import os

import pytest
import torch
import torch.multiprocessing as mp


def _worker(rank: int, world: int, port: int, q):
    os.environ.update({"RANK": str(rank), "WORLD_SIZE": str(world),
                       "LOCAL_RANK": str(rank),
                       "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(port)})
    import torch.distributed as dist
    from learningorchestra_amd.parallel import (all_reduce_grads,
                                                init_distributed)
    from learningorchestra_amd.models.mnist_cnn import build_mnist_cnn
    from learningorchestra_amd.engine.trainer import Trainer, make_sgd
    from learningorchestra_amd.data.synthetic import mnist_batch
    try:
        init_distributed(backend="gloo")
        model = build_mnist_cnn("cpu", seed=11, channels=(4, 4), fc_width=16)
        trainer = Trainer(model, make_sgd(model, lr=0.05), device="cpu")
        # each rank gets DIFFERENT data; grads all-reduce -> identical params
        x, y = mnist_batch(16, dtype=torch.bfloat16, seed=100 + rank)
        for _ in range(3):
            trainer.step_async(x, y)
        checksum = model.arena.master.double().sum().item()
        sig = model.arena.master[:64].clone()
        gathered = [torch.empty_like(sig) for _ in range(world)]
        dist.all_gather(gathered, sig)
        same = all(torch.equal(gathered[0], g) for g in gathered)
        q.put((rank, checksum, same))
        dist.destroy_process_group()
    except Exception as exc:  # pragma: no cover
        q.put((rank, f"ERROR: {exc!r}", False))


@pytest.mark.timeout(240)
@pytest.mark.parametrize("world,port", [(2, 29511), (4, 29515)])
def test_ddp_gloo_params_stay_identical(world, port):
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_worker, args=(r, world, port, q))
             for r in range(world)]
    for p in procs:
        p.start()
    results = [q.get(timeout=220) for _ in range(world)]
    for p in procs:
        p.join(timeout=30)
    for rank, checksum, same in results:
        assert not isinstance(checksum, str), checksum
        assert same, "params diverged across ranks"
    # every rank ended with bit-identical master weights
    assert len({c for _, c, _ in results}) == 1


def test_single_process_allreduce_noop():
    from learningorchestra_amd.parallel import all_reduce_grads, get_world_size
    assert get_world_size() == 1
    g = torch.ones(8)
    assert all_reduce_grads(g) is None
    assert torch.equal(g, torch.ones(8))


def _tree_worker(rank: int, world: int, port: int, q):
    os.environ.update({"RANK": str(rank), "WORLD_SIZE": str(world),
                       "LOCAL_RANK": str(rank),
                       "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(port)})
    import torch.distributed as dist
    from learningorchestra_amd.parallel import init_distributed
    from learningorchestra_amd.data.synthetic import tabular
    from learningorchestra_amd.models.trees import GBTClassifier
    try:
        init_distributed(backend="gloo")
        X, y = tabular(8000, 8, seed=2)           # every rank builds same data
        sh = slice(rank * 4000, (rank + 1) * 4000)  # then trains its shard
        clf = GBTClassifier(n_trees=8, max_depth=4, device="cpu")
        clf.fit(X[sh].numpy(), y[sh].numpy())
        # identical trees across ranks -> identical predictions
        preds = torch.as_tensor(clf.predict(X[:512].numpy()).astype("int64"))
        gathered = [torch.empty_like(preds) for _ in range(world)]
        dist.all_gather(gathered, preds)
        same = all(torch.equal(gathered[0], g) for g in gathered)
        acc = (preds.numpy() == y[:512].numpy().astype(int)).mean()
        q.put((rank, float(acc), same))
        dist.destroy_process_group()
    except Exception as exc:  # pragma: no cover
        q.put((rank, f"ERROR: {exc!r}", False))


@pytest.mark.timeout(180)
def test_distributed_gbt_identical_trees():
    """Data-parallel GBT: per-level histogram all-reduce -> every rank grows
    the same ensemble (the 8-GPU BASELINE config's mechanism, on gloo)."""
    world = 2
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_tree_worker, args=(r, world, 29613, q))
             for r in range(world)]
    for p in procs:
        p.start()
    results = [q.get(timeout=170) for _ in range(world)]
    for p in procs:
        p.join(timeout=30)
    for rank, acc, same in results:
        assert not isinstance(acc, str), acc
        assert same, "trees diverged across ranks"
        assert acc > 0.7, acc


def _textcnn_worker(rank: int, world: int, port: int, q):
    os.environ.update({"RANK": str(rank), "WORLD_SIZE": str(world),
                       "LOCAL_RANK": str(rank),
                       "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(port)})
    import torch.distributed as dist
    from learningorchestra_amd.parallel import init_distributed
    from learningorchestra_amd.models.textcnn import TextCNN
    from learningorchestra_amd.engine.trainer import Trainer, make_sgd
    from learningorchestra_amd.data.synthetic import imdb_batch
    try:
        init_distributed(backend="gloo")
        model = TextCNN(vocab=200, emb_dim=16, filters=16,
                        kernel_sizes=(2, 3), device="cpu", seed=7)
        trainer = Trainer(model, make_sgd(model, lr=0.05), device="cpu")
        ids, y = imdb_batch(8, seq_len=32, vocab=200, seed=50 + rank)
        for _ in range(2):
            trainer.step_async(ids, y)
        sig = model.arena.master[:64].clone()
        gathered = [torch.empty_like(sig) for _ in range(world)]
        dist.all_gather(gathered, sig)
        same = all(torch.equal(gathered[0], g) for g in gathered)
        q.put((rank, float(model.arena.master.double().sum()), same))
        dist.destroy_process_group()
    except Exception as exc:  # pragma: no cover
        q.put((rank, f"ERROR: {exc!r}", False))


@pytest.mark.timeout(180)
def test_ddp_gloo_textcnn_embedding_grads_sync():
    """TextCNN DDP: embedding + multi-branch conv grads all-reduce through
    the same flat-arena hooks -> identical params on every rank."""
    world = 2
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_textcnn_worker, args=(r, world, 29721, q))
             for r in range(world)]
    for p in procs:
        p.start()
    results = [q.get(timeout=170) for _ in range(world)]
    for p in procs:
        p.join(timeout=30)
    for rank, checksum, same in results:
        assert not isinstance(checksum, str), checksum
        assert same, "textcnn params diverged across ranks"


def test_bench_torchrun_world4_cpu(tmp_path):
    """The EXACT command shape the round driver uses for the N-GPU scaling
    ladder, at world 4 on CPU/gloo: one JSON line, whole-job aggregate,
    dp4 parallelism reported."""
    import json
    import os
    import subprocess
    import sys

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    from learningorchestra_amd.parallel.launch import free_port
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "4", "--master-addr", "127.0.0.1",
         "--master-port", str(free_port()), "bench.py", "--gpus", "4",
         "--steps", "2", "--warmup", "1", "--batch", "64"],
        capture_output=True, text=True, timeout=420, cwd=repo)
    assert r.returncode == 0, r.stdout[-2000:] + r.stderr[-2000:]
    line = [ln for ln in r.stdout.splitlines() if ln.startswith("{")][-1]
    doc = json.loads(line)
    assert doc["n_gpus"] == 4
    assert doc["config"]["parallelism"] == "dp4"
    assert doc["config"]["global_batch"] == 4 * 64
    assert doc["value"] > 0
    # full driver contract: every required key present, strict-JSON parseable
    for key in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
                "dtype", "data", "config"):
        assert key in doc, key
    json.loads(line, parse_constant=lambda c: (_ for _ in ()).throw(
        ValueError(f"non-strict JSON constant {c} in bench line")))
