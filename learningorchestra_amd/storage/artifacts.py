"""Artifact (binary) store — models and stage outputs.

The reference persists model/stage binaries on shared Docker volumes keyed
``/binaries/{service_type}/{filename}``, saving Keras models via
``instance.save()`` and everything else with dill, and reading by trying dill
first then keras (/root/reference/microservices/binary_executor_image/
utils.py:195-233). The MI355X rebuild keeps the same path scheme under the
single data root and the same save/load polymorphism, with torch-native
modules saved as a ``state_dict`` + constructor spec (so checkpoints are
portable and resumable) and arbitrary Python objects via dill.
"""
from __future__ import annotations

import json
import os
import re
from typing import Any, Optional

import dill

_NAME_RE = re.compile(r"^[A-Za-z0-9][A-Za-z0-9_.\- ]*$")


def check_name(name: str) -> str:
    """Reject artifact/dataset names that could escape the data root
    (path separators, '..' traversal, hidden/empty names). Raised at the
    API boundary too; this is defense in depth."""
    if (not isinstance(name, str) or not _NAME_RE.match(name)
            or ".." in name or len(name) > 200):
        raise ValueError(f"invalid name {name!r}: names must match "
                         "[A-Za-z0-9][A-Za-z0-9_.- ]* with no '..'")
    return name


class ArtifactStore:
    def __init__(self, root: Optional[str] = None):
        if root is None:
            from ..config import get_config
            root = os.path.join(get_config().data_root, "binaries")
        self._root = root
        os.makedirs(root, exist_ok=True)

    def _dir(self, service_type: str) -> str:
        d = os.path.join(self._root, service_type.replace("/", "_"))
        os.makedirs(d, exist_ok=True)
        return d

    def path(self, name: str, service_type: str) -> str:
        check_name(name)
        d = self._dir(service_type)
        p = os.path.join(d, name)
        # belt-and-braces: the final path must stay under the store root
        if os.path.commonpath([os.path.realpath(os.path.dirname(p)),
                               os.path.realpath(self._root)]) \
                != os.path.realpath(self._root):
            raise ValueError(f"name {name!r} escapes the artifact root")
        return p

    # -- save ---------------------------------------------------------------
    def save(self, instance: Any, name: str, service_type: str) -> str:
        """Persist ``instance``. Torch modules and engine models that expose
        ``lo_spec`` + ``state_dict`` -> spec.json + state.pt (portable:
        rebuildable on ANY device — each DDP rank reconstructs on its own
        GPU); everything else -> dill (mirrors utils.py:195-208
        keras-then-dill)."""
        base = self.path(name, service_type)
        try:
            import torch
            spec = getattr(instance, "lo_spec", None)
            speccable = (isinstance(instance, torch.nn.Module)
                         or (isinstance(spec, dict)
                             and hasattr(instance, "state_dict")))
            if speccable:
                os.makedirs(base, exist_ok=True)
                with open(os.path.join(base, "spec.json"), "w") as fh:
                    json.dump({
                        "format": "torch_module",
                        "class_module": type(instance).__module__,
                        "class_name": type(instance).__name__,
                        "spec": spec,
                    }, fh)
                sd = {k: (v.cpu() if isinstance(v, torch.Tensor) else v)
                      for k, v in instance.state_dict().items()}
                torch.save(sd, os.path.join(base, "state.pt"))
                return base
        except ImportError:
            pass
        with open(base + ".dill", "wb") as fh:
            dill.dump(instance, fh)
        return base + ".dill"

    # -- load ---------------------------------------------------------------
    def load(self, name: str, service_type: str,
             device: Optional[str] = None) -> Any:
        """Read back an artifact: dill first, then spec directory (mirrors
        utils.py:210-221 dill-then-keras). ``device``: override the rebuild
        device when the class constructor accepts one (the DDP worker path)."""
        base = self.path(name, service_type)
        if os.path.exists(base + ".dill"):
            with open(base + ".dill", "rb") as fh:
                return dill.load(fh)
        if os.path.isdir(base) and os.path.exists(os.path.join(base, "spec.json")):
            import importlib
            import inspect
            import torch
            with open(os.path.join(base, "spec.json")) as fh:
                meta = json.load(fh)
            module = importlib.import_module(meta["class_module"])
            cls = getattr(module, meta["class_name"])
            spec = dict(meta.get("spec") or {})
            if device is not None:
                try:
                    if "device" in inspect.signature(cls).parameters:
                        spec["device"] = device
                except (TypeError, ValueError):
                    pass
            instance = cls(**spec)
            state = torch.load(os.path.join(base, "state.pt"),
                               map_location="cpu", weights_only=True)
            instance.load_state_dict(state)
            return instance
        raise FileNotFoundError(f"no artifact '{name}' of type '{service_type}'")

    def exists(self, name: str, service_type: str) -> bool:
        base = self.path(name, service_type)
        return os.path.exists(base + ".dill") or os.path.isdir(base)

    def delete(self, name: str, service_type: str) -> None:
        base = self.path(name, service_type)
        if os.path.exists(base + ".dill"):
            os.remove(base + ".dill")
        elif os.path.isdir(base):
            import shutil
            shutil.rmtree(base)

    # -- raw files (dataset/generic, database_api database.py:69-76) ---------
    def save_raw(self, stream, name: str, service_type: str = "dataset/generic",
                 chunk_size: int = 1 << 20) -> str:
        path = self.path(name, service_type)
        with open(path, "wb") as fh:
            for chunk in stream:
                if chunk:
                    fh.write(chunk)
        return path

    def open_raw(self, name: str, service_type: str = "dataset/generic"):
        return open(self.path(name, service_type), "rb")
