"""Dataset <-> in-memory resolution (the reference's ``Data`` classes,
e.g. /root/reference/microservices/binary_executor_image/utils.py:257-352).

Resolves a name to content for the ``$``-parameter protocol:
* a dataset collection -> pandas DataFrame (metadata row dropped), with the
  tensor fast path for numeric datasets;
* a stored binary -> the loaded instance.
"""
from __future__ import annotations

from typing import Any, List, Optional

from .metadata import METADATA_ROW_ID, Metadata


class Data:
    def __init__(self, database, artifacts):
        self._db = database
        self._artifacts = artifacts
        self._metadata = Metadata(database)

    def get_type(self, name: str) -> Optional[str]:
        return self._metadata.get_type(name)

    def get_dataset_content(self, name: str, fields: Optional[List[str]] = None,
                            limit: Optional[int] = None):
        """Dataset collection -> pandas DataFrame (rows with _id > 0),
        mirroring utils.py:318-332 which reads the full collection."""
        import pandas as pd
        cursor = self._db[name].find({"_id": {"$ne": METADATA_ROW_ID}}).sort("_id", 1)
        if limit:
            cursor = cursor.limit(limit)
        rows = []
        for doc in cursor:
            doc.pop("_id", None)
            rows.append(doc)
        df = pd.DataFrame(rows)
        if fields:
            df = df[fields]
        return df

    def get_dataset_tensor(self, name: str, fields: Optional[List[str]] = None,
                           dtype=None):
        """Numeric dataset -> torch tensor (the GPU-loader hot path the
        reference lacked — it re-read Mongo row-by-row)."""
        import torch
        df = self.get_dataset_content(name, fields)
        import numpy as np
        arr = df.to_numpy(dtype=np.float32, na_value=float("nan"))
        t = torch.from_numpy(arr)
        return t.to(dtype) if dtype is not None else t

    def get_object(self, name: str) -> Any:
        """Resolve a name: stored binary if present, else dataset DataFrame."""
        meta = self._metadata.get_metadata(name)
        if meta is not None:
            stype = meta.get("type", "")
            if self._artifacts.exists(name, stype):
                return self._artifacts.load(name, stype)
            return self.get_dataset_content(name)
        # no metadata: try every known artifact type dir
        raise KeyError(f"unknown dataset or artifact '{name}'")

    def get_module_and_class(self, name: str):
        """Walk the parent chain to the owning model's module/class
        (binary_executor_image/utils.py:257-276)."""
        doc = self._metadata.walk_to_model(name)
        if doc is None:
            return None, None
        return doc.get("modulePath"), doc.get("className", doc.get("class"))
