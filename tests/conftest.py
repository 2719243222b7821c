import os
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: test requires an AMD GPU (MI355X); skipped on CPU-only hosts")


def pytest_collection_modifyitems(config, items):
    try:
        import torch
        has_gpu = torch.cuda.is_available()
    except Exception:
        has_gpu = False
    if has_gpu:
        return
    skip_gpu = pytest.mark.skip(reason="no GPU available")
    for item in items:
        if "gpu" in item.keywords:
            item.add_marker(skip_gpu)


@pytest.fixture()
def tmp_config(tmp_path, monkeypatch):
    """Fresh Config rooted in a tmp dir."""
    from learningorchestra_amd.config import Config, set_config
    cfg = Config(data_root=str(tmp_path), mongo_uri="")
    set_config(cfg)
    yield cfg
    set_config(None)


@pytest.fixture()
def db(tmp_config):
    from learningorchestra_amd.storage.docstore import DocumentStore
    return DocumentStore(None)  # in-memory


@pytest.fixture()
def artifacts(tmp_config, tmp_path):
    from learningorchestra_amd.storage.artifacts import ArtifactStore
    return ArtifactStore(str(tmp_path / "binaries"))
