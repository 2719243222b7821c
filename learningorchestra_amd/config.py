"""Typed configuration (replaces the reference's env-vars + per-service Constants
classes, e.g. /root/reference/microservices/binary_executor_image/constants.py:1-79
and docker-compose env anchors docker-compose.yml:20-24).

One config object, sourced from environment variables with sane single-node
defaults. The reference spread this over 9 Dockerfiles; here it is one place.
"""
from __future__ import annotations

import os
from dataclasses import dataclass, field


def _env(name: str, default: str) -> str:
    return os.environ.get(name, default)


@dataclass
class Config:
    # --- storage -----------------------------------------------------------
    # Root directory for all persisted state (document store + artifacts).
    # The reference used a 3-member MongoDB replica set + named Docker volumes
    # (docker-compose.yml:42-90,325-333); single-node MI355X keeps one root.
    data_root: str = field(default_factory=lambda: _env(
        "LO_DATA_ROOT", os.path.join(os.path.expanduser("~"), ".learningorchestra_amd")))
    # Optional real MongoDB URI. If set (and pymongo can connect) the document
    # store uses it; otherwise the embedded Mongo-compatible store is used.
    mongo_uri: str = field(default_factory=lambda: _env("LO_MONGO_URI", ""))
    database_name: str = field(default_factory=lambda: _env("LO_DATABASE_NAME", "database"))

    # --- API server --------------------------------------------------------
    host: str = field(default_factory=lambda: _env("LO_HOST", "0.0.0.0"))
    port: int = field(default_factory=lambda: int(_env("LO_PORT", "80")))
    api_prefix: str = "/api/learningOrchestra/v1"

    # --- paging (reference: database_api_image/constants.py:41) ------------
    limit_param_max: int = 100
    metadata_row_id: int = 0

    # --- gateway-style response cache (krakend.json:1769-1770: 300 s
    # cache_ttl on the reference's gateway). DELIBERATE deviation: default 0
    # (off) — a 300 s GET cache makes the finished-flag poll contract serve
    # stale "finished: false" for minutes, which the reference shipped as a
    # quirk; enable with LO_CACHE_TTL for reference-faithful behavior.
    # Mutations invalidate the cache either way.
    cache_ttl: float = field(default_factory=lambda: float(_env("LO_CACHE_TTL", "0")))

    # --- executor ----------------------------------------------------------
    max_jobs: int = field(default_factory=lambda: int(_env("LO_MAX_JOBS", "8")))
    # exec() of user-supplied code (builder modelingCode, function/python,
    # '#' parameters) is part of the reference API (builder.py:99,
    # code_execution.py:185) and is gated behind this flag. IMPORTANT SCOPE
    # NOTE: LO_ALLOW_USER_CODE=0 does NOT make the API safe for
    # untrusted callers — the model/explore/transform/binary verbs remain a
    # reflective importlib surface by reference design (any module path +
    # callable with caller kwargs). The deployment assumption is a trusted
    # cluster, exactly as the reference's (which shipped 4 ungated exec()s).
    allow_user_code: bool = field(default_factory=lambda: _env("LO_ALLOW_USER_CODE", "1") == "1")

    # --- compute -----------------------------------------------------------
    device: str = field(default_factory=lambda: _env("LO_DEVICE", "auto"))  # auto|cuda|cpu
    dtype: str = field(default_factory=lambda: _env("LO_DTYPE", "bf16"))

    def resolve_device(self) -> str:
        if self.device != "auto":
            return self.device
        try:
            import torch
            return "cuda" if torch.cuda.is_available() else "cpu"
        except Exception:
            return "cpu"


_config: Config | None = None


def get_config() -> Config:
    global _config
    if _config is None:
        _config = Config()
    return _config


def set_config(cfg: Config) -> None:
    global _config
    _config = cfg
