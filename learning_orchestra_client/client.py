from __future__ import annotations

import json
import time
from typing import Any, Dict, List, Optional

PREFIX = "/api/learningOrchestra/v1"


class LearningOrchestraError(RuntimeError):
    def __init__(self, status: int, message: str):
        super().__init__(f"HTTP {status}: {message}")
        self.status = status


class Context:
    """Global connection context (reference: ``Context(cluster_ip)``)."""

    _current: Optional["Context"] = None

    def __init__(self, cluster_ip: Optional[str] = None, _session=None):
        if _session is not None:
            self._session = _session
            self._base = ""
        else:
            import requests
            if not cluster_ip.startswith("http"):
                cluster_ip = f"http://{cluster_ip}"
            self._session = requests.Session()
            self._base = cluster_ip.rstrip("/")
        Context._current = self

    @classmethod
    def in_process(cls, runtime=None) -> "Context":
        """Embedded mode: run the API app inside this process."""
        from fastapi.testclient import TestClient
        from learningorchestra_amd.api.app import Runtime, create_app
        app = create_app(runtime or Runtime())
        return cls(_session=TestClient(app))

    @classmethod
    def current(cls) -> "Context":
        if cls._current is None:
            raise RuntimeError("call Context(cluster_ip) first")
        return cls._current

    # -- HTTP helpers --------------------------------------------------------
    def _req(self, method: str, path: str, **kw):
        resp = getattr(self._session, method)(self._base + path, **kw)
        body = {}
        try:
            body = resp.json()
        except Exception:
            pass
        if resp.status_code >= 400:
            raise LearningOrchestraError(resp.status_code,
                                         str(body.get("result", resp.text)))
        return body.get("result"), resp.status_code

    def post(self, path: str, payload: Dict[str, Any]):
        return self._req("post", path, json=payload)[0]

    def patch(self, path: str, payload: Dict[str, Any]):
        return self._req("patch", path, json=payload)[0]

    def get(self, path: str, params: Optional[Dict[str, Any]] = None):
        return self._req("get", path, params=params)[0]

    def delete(self, path: str):
        return self._req("delete", path)[0]


class _VerbClient:
    verb: str = ""
    tool: str = ""

    def __init__(self, context: Optional[Context] = None):
        self._ctx = context or Context.current()

    @property
    def _path(self) -> str:
        return f"{PREFIX}/{self.verb}/{self.tool}"

    def search_all(self) -> List[Dict[str, Any]]:
        return self._ctx.get(self._path)

    def search(self, name: str, query: Dict[str, Any] = None, limit: int = 10,
               skip: int = 0) -> List[Dict[str, Any]]:
        return self._ctx.get(f"{self._path}/{name}",
                             {"query": json.dumps(query or {}),
                              "limit": limit, "skip": skip})

    def metadata(self, name: str) -> Dict[str, Any]:
        return self._ctx.get(f"{self._path}/{name}/metadata")

    def delete(self, name: str):
        return self._ctx.delete(f"{self._path}/{name}")

    def wait(self, name: str, timeout: float = 300.0) -> Dict[str, Any]:
        """Observe-verb wait: block until the artifact's finished flag."""
        return Observe(self._ctx).wait(name, timeout)


class DatasetCsv(_VerbClient):
    verb, tool = "dataset", "csv"

    def insert(self, dataset_name: str, url: str) -> str:
        return self._ctx.post(self._path, {"datasetName": dataset_name,
                                           "datasetURI": url})

    def insert_sync(self, dataset_name: str, url: str,
                    timeout: float = 600.0) -> Dict[str, Any]:
        self.insert(dataset_name, url)
        return self.wait(dataset_name, timeout)


class DatasetGeneric(DatasetCsv):
    verb, tool = "dataset", "generic"


class Model(_VerbClient):
    verb = "model"

    def __init__(self, context: Optional[Context] = None, tool: str = "torch"):
        super().__init__(context)
        self.tool = tool

    def create(self, model_name: str, module_path: str, class_name: str,
               class_parameters: Dict[str, Any] = None,
               description: str = "") -> str:
        return self._ctx.post(self._path, {
            "modelName": model_name, "modulePath": module_path,
            "class": class_name, "classParameters": class_parameters or {},
            "description": description})

    def update(self, model_name: str, class_parameters: Dict[str, Any]):
        return self._ctx.patch(f"{self._path}/{model_name}",
                               {"classParameters": class_parameters})


class _BinaryVerb(_VerbClient):
    def __init__(self, context: Optional[Context] = None, tool: str = "torch"):
        super().__init__(context)
        self.tool = tool

    def create(self, name: str, parent_name: str, method: str,
               method_parameters: Dict[str, Any] = None,
               model_name: Optional[str] = None, description: str = "") -> str:
        return self._ctx.post(self._path, {
            "name": name, "parentName": parent_name,
            "modelName": model_name or parent_name, "method": method,
            "methodParameters": method_parameters or {},
            "description": description})

    def update(self, name: str, method_parameters: Dict[str, Any]):
        return self._ctx.patch(f"{self._path}/{name}",
                               {"methodParameters": method_parameters})


class Train(_BinaryVerb):
    verb = "train"


class Tune(_BinaryVerb):
    verb = "tune"


class Evaluate(_BinaryVerb):
    verb = "evaluate"


class Predict(_BinaryVerb):
    verb = "predict"


class TransformProjection(_VerbClient):
    verb, tool = "transform", "projection"

    def create(self, input_dataset: str, output_dataset: str,
               fields: List[str]) -> str:
        return self._ctx.post(self._path, {
            "inputDatasetName": input_dataset,
            "outputDatasetName": output_dataset, "names": fields})


class TransformDataType(_VerbClient):
    verb, tool = "transform", "dataType"

    def convert(self, dataset_name: str, types: Dict[str, str]) -> str:
        return self._ctx.patch(self._path, {"datasetName": dataset_name,
                                            "types": types})


class ExploreHistogram(_VerbClient):
    verb, tool = "explore", "histogram"

    def create(self, input_dataset: str, output_dataset: str,
               fields: List[str]) -> str:
        return self._ctx.post(self._path, {
            "inputDatasetName": input_dataset,
            "outputDatasetName": output_dataset, "names": fields})


class BuilderSparkMl(_VerbClient):
    verb, tool = "builder", "sparkml"

    def build(self, train_dataset: str, test_dataset: str, modeling_code: str,
              classifiers: List[str]) -> List[str]:
        return self._ctx.post(self._path, {
            "trainDatasetName": train_dataset, "testDatasetName": test_dataset,
            "modelingCode": modeling_code, "classifiersList": classifiers})


class FunctionPython(_VerbClient):
    verb, tool = "function", "python"

    def run(self, name: str, code: str, parameters: Dict[str, Any] = None,
            description: str = "") -> str:
        return self._ctx.post(self._path, {
            "name": name, "function": code,
            "functionParameters": parameters or {},
            "description": description})


class Observe(_VerbClient):
    """The Observe verb — server-side long-poll on the finished flag (the
    reference implemented this client-side as collection polling)."""
    verb = "observe"

    def observe(self, name: str) -> Dict[str, Any]:
        return self._ctx.get(f"{PREFIX}/observe/{name}")

    def wait(self, name: str, timeout: float = 300.0) -> Dict[str, Any]:
        deadline = time.time() + timeout
        while True:
            remain = max(1.0, min(60.0, deadline - time.time()))
            doc = self._ctx.get(f"{PREFIX}/observe/{name}/wait",
                                {"timeoutSeconds": remain})
            if doc and doc.get("finished"):
                if doc.get("exception"):
                    raise LearningOrchestraError(
                        500, f"pipeline '{name}' failed: {doc['exception']}")
                return doc
            if time.time() >= deadline:
                raise TimeoutError(f"'{name}' not finished after {timeout}s")

    def watch(self, name: str, verb: str = "train", tool: str = "torch",
              poll: float = 0.25, timeout: float = 300.0):
        """Yield result-collection documents as they are written (the
        reference pip client watched the result collection with Mongo
        change streams); generator ends when the pipeline's finished flag
        is set. ``verb``/``tool`` name the collection's poll URI."""
        deadline = time.time() + timeout
        seen = 1                       # _id 0 is the metadata document
        path = f"{PREFIX}/{verb}/{tool}/{name}"
        while True:
            try:
                rows = self._ctx.get(path, {
                    "query": json.dumps({"_id": {"$gte": seen}}),
                    "limit": 100, "skip": 0}) or []
            except LearningOrchestraError:
                rows = []              # collection not created yet
            for doc in sorted((d for d in rows
                               if isinstance(d.get("_id"), int)),
                              key=lambda d: d["_id"]):
                seen = doc["_id"] + 1
                yield doc
            meta = self.observe(name)
            if meta and meta.get("finished"):
                return
            if time.time() >= deadline:
                raise TimeoutError(f"'{name}' not finished after {timeout}s")
            time.sleep(poll)
