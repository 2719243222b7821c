"""`python -m learningorchestra_amd` — the single-node launcher.

Replaces the reference's ./run.sh + docker stack deploy: one
process owns the node's GPUs; multi-GPU training jobs are spawned per-request
by the scheduler (train verb with "gpus": N).

    python -m learningorchestra_amd                # serve the REST API
    python -m learningorchestra_amd --build        # compile gfx950 kernels
    python -m learningorchestra_amd --port 8080
"""
from __future__ import annotations

import argparse


def main() -> None:
    ap = argparse.ArgumentParser(prog="learningorchestra_amd")
    ap.add_argument("--build", action="store_true",
                    help="compile the gfx950 HIP extension in-tree and exit")
    ap.add_argument("--host", default=None)
    ap.add_argument("--port", type=int, default=None)
    args = ap.parse_args()
    if args.build:
        from .build_ext import build
        build()
        return
    import uvicorn

    from .api.app import create_app
    from .config import get_config
    cfg = get_config()
    uvicorn.run(create_app(), host=args.host or cfg.host,
                port=args.port or cfg.port)


if __name__ == "__main__":
    main()
