"""Per-rank worker for API-launched data-parallel training.

Launched as ``python -m torch.distributed.run --nnodes=1 --nproc-per-node N
--master-addr 127.0.0.1 -m learningorchestra_amd.parallel.train_worker
--spec <spec.json>`` by parallel/launch.py (which the train verb's scheduler
job spawns). This replaces the reference's Spark worker fan-out — builder
jobs ran on 3 Spark workers (/root/reference/docker-compose.yml:157-163);
here a train/torch request with ``"gpus": N`` runs one process per GPU with
RCCL gradient all-reduce over xGMI.

Every rank: rebuild the parent model artifact on its own device, resolve the
``$``-parameter protocol against the (flushed) document store, shard the rows
``rank::world``, run ``fit`` (the engine Trainer all-reduces grads when
distributed is initialized). Rank 0 persists the fitted instance and writes a
result JSON the API process folds into the execution document.
"""
from __future__ import annotations

import argparse
import json
import os
import time


def _shard(value, rank: int, world: int):
    """Row-shard DataFrames / arrays / tensors rank::world."""
    if world <= 1 or value is None:
        return value
    try:
        import pandas as pd
        if isinstance(value, pd.DataFrame) or isinstance(value, pd.Series):
            return value.iloc[rank::world]
    except ImportError:
        pass
    import numpy as np
    import torch
    if isinstance(value, (np.ndarray, torch.Tensor)) and value.ndim >= 1:
        return value[rank::world]
    return value


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--spec", required=True)
    args = ap.parse_args()
    with open(args.spec) as fh:
        spec = json.load(fh)

    import torch

    from ..config import get_config
    from ..executor.parameters import Parameters
    from ..storage import ArtifactStore, Data, Metadata, connect
    from .ddp import barrier, get_rank, get_world_size, init_distributed

    local_rank = init_distributed()
    rank = get_rank()
    world = get_world_size()
    use_gpu = torch.cuda.is_available()
    if use_gpu:
        n_dev = torch.cuda.device_count()
        device = f"cuda:{local_rank % n_dev}"  # 2-ranks-on-1-GPU RCCL works
        torch.cuda.set_device(device)
    else:
        device = "cpu"

    cfg = get_config()  # LO_* env forwarded by the launcher
    db = connect(cfg)
    artifacts = ArtifactStore(os.path.join(cfg.data_root, "binaries"))
    data = Data(db, artifacts)

    parent = spec["artifact"]
    instance = artifacts.load(parent["name"], parent["type"], device=device)

    params = Parameters(data, cfg.allow_user_code).treat(
        spec.get("method_parameters", {}))
    for key in ("x", "y"):
        if key in params:
            params[key] = _shard(params[key], rank, world)

    t0 = time.time()
    method = getattr(instance, spec.get("method", "fit"))
    result = method(**params)
    if use_gpu:
        torch.cuda.synchronize()
    barrier()
    duration = time.time() - t0

    if rank == 0:
        save_as = spec.get("save_as")
        if save_as:
            artifacts.save(instance, save_as["name"], save_as["type"])
        out = {
            "worldSize": world,
            "device": device,
            "durationSeconds": round(duration, 4),
            "result": result if isinstance(result, (dict, list, str, int,
                                                    float, bool, type(None)))
            else repr(result),
        }
        if spec.get("result_path"):
            tmp = spec["result_path"] + ".tmp"
            with open(tmp, "w") as fh:
                json.dump(out, fh, default=str)  # numpy scalars etc.
            os.replace(tmp, spec["result_path"])
        # metadata is written by the API process when the job completes
        _ = Metadata  # imported for parity; reserved for future direct writes
    barrier()


if __name__ == "__main__":
    main()
