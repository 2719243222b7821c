"""learning_orchestra_client — the Python client (pythonClient parity).

The reference ships a pip package of the same name whose scripts start with::

    from learning_orchestra_client import *
    Context("xx.xx.xxx.xxx")

(/root/reference/README.md:92-103). This client keeps that entry point and
offers one class per verb/tool pair with create / search / read / delete and
the Observe wait contract, speaking the same REST surface the MI355X server
exposes. Two transports:

* ``Context("host[:port]")``      — HTTP to a running server;
* ``Context.in_process()``        — embedded single-node mode: spins the
  FastAPI app in-process (no server needed), same API.
"""
from .client import (BuilderSparkMl, Context, DatasetCsv, DatasetGeneric,  # noqa: F401
                     ExploreHistogram, Evaluate, FunctionPython, Model,
                     Observe, Predict, Train, TransformDataType,
                     TransformProjection, Tune)

__all__ = ["Context", "DatasetCsv", "DatasetGeneric", "Model", "Train",
           "Tune", "Evaluate", "Predict", "TransformProjection",
           "TransformDataType", "ExploreHistogram", "BuilderSparkMl",
           "FunctionPython", "Observe"]
