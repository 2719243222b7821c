"""Tune verb — native hyper-parameter search over the model zoo.

The reference's Tune verb drove sklearn GridSearchCV-style classes through
the binary executor (SURVEY §2.1 binary_executor); that path still works for
the scikitlearn tool. This module adds the MI355X-native equivalent for the
torch tool: grid/random search schedulers that fit zoo models per candidate
(each trial on the GPU engine) and keep the best by a scored metric.

    POST /model/torch    {"class": "GridSearch", "modulePath":
        "learningorchestra_amd.models.tuning", "classParameters": {
            "modulePath": "learningorchestra_amd.models.zoo",
            "className": "MnistCNN",
            "paramGrid": {"lr": [0.01, 0.05], "fc_width": [128, 256]}}}
    POST /train/torch    {"name": "tuned", "parentName": ..., "method": "fit",
                          "methodParameters": {"x": "$ds", "y": "$ds.label"}}
"""
from __future__ import annotations

import importlib
import itertools
import random
from typing import Any, Dict, List, Optional


class GridSearch:
    def __init__(self, modulePath: str, className: str,
                 paramGrid: Dict[str, List[Any]],
                 fixedParameters: Optional[Dict[str, Any]] = None,
                 metric: str = "accuracy", epochs: int = 1,
                 validationSplit: float = 0.2):
        self.module_path = modulePath
        self.class_name = className
        self.param_grid = paramGrid
        self.fixed = fixedParameters or {}
        self.metric = metric
        self.epochs = epochs
        self.validation_split = validationSplit
        self.results_: List[Dict[str, Any]] = []
        self.best_params_: Optional[Dict[str, Any]] = None
        self.best_score_: float = float("-inf")
        self.best_estimator_ = None

    def _candidates(self):
        keys = list(self.param_grid)
        for combo in itertools.product(*(self.param_grid[k] for k in keys)):
            yield dict(zip(keys, combo))

    def _make(self, params: Dict[str, Any]):
        module = importlib.import_module(self.module_path)
        cls = getattr(module, self.class_name)
        return cls(**{**self.fixed, **params})

    def _split(self, x, y):
        import numpy as np
        n = len(y)
        n_val = max(1, int(n * self.validation_split))
        rng = np.random.RandomState(0)
        idx = rng.permutation(n)
        tr, va = idx[n_val:], idx[:n_val]
        xa, ya = np.asarray(x), np.asarray(y)
        return xa[tr], ya[tr], xa[va], ya[va]

    def fit(self, x=None, y=None) -> "GridSearch":
        xtr, ytr, xva, yva = self._split(x, y)
        for params in self._candidates():
            est = self._make(params)
            est.fit(xtr, ytr, epochs=self.epochs) if _accepts_epochs(est) \
                else est.fit(xtr, ytr)
            score = self._score(est, xva, yva)
            self.results_.append({"params": params, "score": score})
            if score > self.best_score_:
                self.best_score_, self.best_params_ = score, params
                self.best_estimator_ = est
        return self

    def _score(self, est, xva, yva) -> float:
        if hasattr(est, "evaluate"):
            out = est.evaluate(xva, yva)
            if isinstance(out, dict):
                return float(out.get(self.metric, next(iter(out.values()))))
            return float(out)
        import numpy as np
        return float((np.asarray(est.predict(xva)).astype(int)
                      == np.asarray(yva).astype(int)).mean())

    def predict(self, x):
        return self.best_estimator_.predict(x)

    def evaluate(self, x, y):
        return {"best_score": self.best_score_, "best_params": self.best_params_,
                "holdout": self._score(self.best_estimator_, x, y)}

    def summary(self) -> Dict[str, Any]:
        return {"results": self.results_, "bestParams": self.best_params_,
                "bestScore": self.best_score_}


class RandomSearch(GridSearch):
    def __init__(self, modulePath: str, className: str,
                 paramGrid: Dict[str, List[Any]], nIter: int = 10,
                 seed: int = 0, **kw):
        super().__init__(modulePath, className, paramGrid, **kw)
        self.n_iter = nIter
        self.seed = seed

    def _candidates(self):
        rng = random.Random(self.seed)
        keys = list(self.param_grid)
        for _ in range(self.n_iter):
            yield {k: rng.choice(self.param_grid[k]) for k in keys}


def _accepts_epochs(est) -> bool:
    import inspect
    try:
        return "epochs" in inspect.signature(est.fit).parameters
    except (TypeError, ValueError):
        return False
