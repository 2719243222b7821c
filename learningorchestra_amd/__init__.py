"""learningorchestra_amd — an MI355X-native ML pipeline orchestration engine.

A from-scratch rebuild of the capabilities of learningOrchestra
(https://github.com/learningOrchestra/learningOrchestra): the same REST API
surface (``/api/learningOrchestra/v1/{verb}/{tool}``), the same 11 pipeline
verbs (Dataset, Model, Transform, Explore, Tune, Train, Evaluate, Predict,
Builder, Observe, Function), the same MongoDB-style metadata / lineage /
``finished``-flag polling contract — but architected AMD-first:

* one in-process executor (replacing the reference's ~10 Flask microservices,
  ``/root/reference/microservices/*``) built on PyTorch-ROCm;
* every train/predict hot-path op (dense + conv GEMM, softmax-cross-entropy,
  SGD/Adam, pooling, tree-histogram build) is a hand-written CDNA4 HIP kernel
  (MFMA + LDS tiling, gfx950);
* multi-GPU scaling is data-parallel RCCL all-reduce over xGMI on one 8-GPU
  node (replacing the reference's Spark worker fan-out).
"""

__version__ = "0.1.0"

from .config import Config, get_config  # noqa: F401
