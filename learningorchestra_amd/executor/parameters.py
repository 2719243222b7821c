"""The ``$``/``#`` parameter-resolution protocol.

The reference triplicates a ``Parameters`` class into model, database_executor,
binary_executor and code_executor (e.g. /root/reference/microservices/
binary_executor_image/binary_execution.py:18-89). Protocol, implemented once:

* ``"$name"``       -> the content of dataset/artifact ``name`` (DataFrame for
                       datasets, loaded instance for binaries);
* ``"$name.attr"``  -> attribute / key / column of that loaded object;
* ``"#python-expr"``-> the expression is evaluated and the resulting object is
                       used (gated by ``Config.allow_user_code``);
* lists are mapped element-wise (binary_execution.py:21-28);
* anything else passes through verbatim.
"""
from __future__ import annotations

from typing import Any, Dict

DATASET_MARKER = "$"
CODE_MARKER = "#"


class UserCodeDisabled(PermissionError):
    pass


class Parameters:
    def __init__(self, data, allow_user_code: bool = True, extra_globals: Dict[str, Any] = None):
        self._data = data
        self._allow_user_code = allow_user_code
        self._extra_globals = extra_globals or {}

    def treat(self, parameters: Dict[str, Any]) -> Dict[str, Any]:
        return {name: self._treat_value(v) for name, v in (parameters or {}).items()}

    def treat_value(self, value: Any) -> Any:
        return self._treat_value(value)

    def _treat_value(self, value: Any) -> Any:
        if isinstance(value, list):
            return [self._treat_value(v) for v in value]
        if isinstance(value, dict):
            return {k: self._treat_value(v) for k, v in value.items()}
        if not isinstance(value, str):
            return value
        if value.startswith(DATASET_MARKER):
            return self._resolve_dataset(value[1:])
        if value.startswith(CODE_MARKER):
            return self._eval_code(value[1:])
        return value

    def _resolve_dataset(self, spec: str) -> Any:
        name, _, attr_path = spec.partition(".")
        obj = self._data.get_object(name)
        if not attr_path:
            return obj
        for attr in attr_path.split("."):
            obj = self._get_attr(obj, attr)
        return obj

    @staticmethod
    def _get_attr(obj: Any, attr: str) -> Any:
        # attribute, then mapping key / DataFrame column (model.py:32-50
        # resolves "$name.attr" as object attributes; DataFrames commonly
        # want column access)
        if hasattr(obj, attr):
            return getattr(obj, attr)
        try:
            return obj[attr]
        except Exception:
            raise AttributeError(f"object of type {type(obj).__name__} has no "
                                 f"attribute or key '{attr}'")

    def _eval_code(self, code: str) -> Any:
        """``"#expr"`` -> evaluated object (reference model.py:52-64 exec()s
        with tensorflow in scope; here the native model zoo + torch are)."""
        if not self._allow_user_code:
            raise UserCodeDisabled(
                "user-code parameters ('#...') are disabled (LO_ALLOW_USER_CODE=0)")
        ctx: Dict[str, Any] = dict(self._extra_globals)
        _install_default_globals(ctx)
        # the reference exec()s "instance = <code>" (model.py:58-62); eval is
        # the same surface for expressions, exec fallback for statements
        try:
            return eval(code, ctx)  # noqa: S307 - documented trusted-cluster surface
        except SyntaxError:
            exec(code, ctx)  # noqa: S102
            if "instance" in ctx:
                return ctx["instance"]
            if "response" in ctx:
                return ctx["response"]
            raise ValueError("statement-style '#' parameter must set "
                             "'instance' (or 'response')")


def _install_default_globals(ctx: Dict[str, Any]) -> None:
    import builtins
    ctx.setdefault("__builtins__", builtins)
    for mod in ("numpy", "pandas", "torch"):
        try:
            ctx.setdefault(mod.split(".")[0], __import__(mod))
        except ImportError:
            pass
    try:
        import learningorchestra_amd.models as lo_models
        ctx.setdefault("lo_models", lo_models)
    except ImportError:
        pass
