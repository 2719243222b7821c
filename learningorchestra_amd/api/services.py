"""Verb services that aren't plain reflective executions: projection,
dataType, histogram, and the Spark-ML-style builder.

In the reference these were separate Swarm microservices running Spark jobs
(/root/reference/microservices/{projection,data_type_handler,histogram,
builder}_image); here they are in-process pipelines over the document store
and the MI355X engine.
"""
from __future__ import annotations

import time
from typing import Any, Dict, List, Optional

from ..executor.execution import ValidationError
from ..storage.metadata import METADATA_ROW_ID, Metadata


class ProjectionService:
    """transform/projection: column projection of a dataset
    (reference projection.py:20-48 ran a Spark select; here a streamed
    batched copy)."""

    def __init__(self, database, scheduler):
        self._db = database
        self._metadata = Metadata(database)
        self._scheduler = scheduler

    def create(self, input_name: str, output_name: str, fields: List[str]) -> None:
        parent = self._metadata.get_metadata(input_name)
        if parent is None:
            raise ValidationError(f"dataset '{input_name}' not found")
        if not parent.get("finished", False):
            raise ValidationError(f"dataset '{input_name}' is not finished")
        if self._metadata.exists(output_name):
            raise ValidationError(f"'{output_name}' already exists")
        known = parent.get("fields") or []
        bad = [f for f in fields if known and f not in known]
        if bad:
            raise ValidationError(f"invalid fields: {bad}")
        self._metadata.create_file(output_name, "transform/projection",
                                   parentName=input_name, fields=fields)

        def pipeline():
            src = self._db[input_name]
            dst = self._db[output_name]
            batch = []
            for doc in src.find({"_id": {"$ne": METADATA_ROW_ID}}).sort("_id", 1):
                row = {"_id": doc["_id"]}
                for f in fields:
                    row[f] = doc.get(f)
                batch.append(row)
                if len(batch) >= 4096:
                    dst.insert_many(batch)
                    batch = []
            if batch:
                dst.insert_many(batch)
            self._metadata.update_finished_flag(output_name, True)

        self._scheduler.submit(output_name, pipeline)


class DataTypeService:
    """transform/dataType: in-place field type conversion (string <-> number),
    the reference's data_type_update.py:15-59 (empty string -> None, float ->
    int when integral)."""

    VALID_TYPES = ("string", "number")

    def __init__(self, database, scheduler):
        self._db = database
        self._metadata = Metadata(database)
        self._scheduler = scheduler

    def convert(self, dataset_name: str, field_types: Dict[str, str]) -> None:
        meta = self._metadata.get_metadata(dataset_name)
        if meta is None:
            raise ValidationError(f"dataset '{dataset_name}' not found")
        bad = [t for t in field_types.values() if t not in self.VALID_TYPES]
        if bad:
            raise ValidationError(f"invalid types {bad}; use {self.VALID_TYPES}")
        known = meta.get("fields") or []
        missing = [f for f in field_types if known and f not in known]
        if missing:
            raise ValidationError(f"unknown fields: {missing}")
        self._metadata.update_finished_flag(dataset_name, False)

        def pipeline():
            col = self._db[dataset_name]
            for doc in col.find({"_id": {"$ne": METADATA_ROW_ID}}):
                updates = {}
                for field, ftype in field_types.items():
                    v = doc.get(field)
                    if v is None:
                        continue
                    if ftype == "number" and isinstance(v, str):
                        if v == "":
                            updates[field] = None
                        else:
                            try:
                                f = float(v)
                                updates[field] = int(f) if f.is_integer() else f
                            except ValueError:
                                updates[field] = None
                    elif ftype == "string" and not isinstance(v, str):
                        updates[field] = str(v)
                if updates:
                    col.update_one({"_id": doc["_id"]}, {"$set": updates})
            self._metadata.update_finished_flag(dataset_name, True)

        self._scheduler.submit(dataset_name, pipeline)


class HistogramService:
    """explore/histogram: per-field value counts — the reference's Mongo
    $group pipeline (histogram.py:31-32), one result document per field."""

    def __init__(self, database, scheduler):
        self._db = database
        self._metadata = Metadata(database)
        self._scheduler = scheduler

    def create(self, input_name: str, output_name: str, fields: List[str]) -> None:
        parent = self._metadata.get_metadata(input_name)
        if parent is None:
            raise ValidationError(f"dataset '{input_name}' not found")
        if self._metadata.exists(output_name):
            raise ValidationError(f"'{output_name}' already exists")
        self._metadata.create_file(output_name, "explore/histogram",
                                   parentName=input_name, fields=fields)

        def pipeline():
            col = self._db[input_name]
            dst = self._db[output_name]
            next_id = 1
            for field in fields:
                counts = col.aggregate([
                    {"$match": {"_id": {"$ne": METADATA_ROW_ID}}},
                    {"$group": {"_id": f"${field}", "count": {"$sum": 1}}},
                ])
                values = {str(c["_id"]): c["count"] for c in counts}
                dst.insert_one({"_id": next_id, "field": field, "values": values})
                next_id += 1
            self._metadata.update_finished_flag(output_name, True)

        self._scheduler.submit(output_name, pipeline)


class BuilderService:
    """builder/sparkml: the whole-pipeline verb — user preprocessing code +
    N classifiers, one prediction dataset per classifier with fitTime / F1 /
    accuracy recorded (reference builder.py:30-194).

    The Spark-MLlib classifier set {lr, dt, rf, gb, nb} maps to the
    MI355X-native tabular engines in models/tabular.py / models/trees.py.
    """

    CLASSIFIERS = ("lr", "dt", "rf", "gb", "nb", "mlp")

    def __init__(self, database, artifacts, scheduler, allow_user_code=True,
                 device: Optional[str] = None):
        self._db = database
        self._artifacts = artifacts
        self._metadata = Metadata(database)
        self._scheduler = scheduler
        self._allow_user_code = allow_user_code
        self._device = device

    def result_name(self, test_dataset: str, classifier: str) -> str:
        # reference names result collections testDataset+classifier
        # (builder_image/utils.py:42-44)
        return f"{test_dataset}{classifier}"

    def create(self, train_dataset: str, test_dataset: str, modeling_code: str,
               classifiers: List[str]) -> List[str]:
        if not self._allow_user_code:
            raise ValidationError("builder modelingCode requires LO_ALLOW_USER_CODE=1")
        bad = [c for c in classifiers if c.lower() not in self.CLASSIFIERS]
        if bad:
            raise ValidationError(
                f"invalid classifiers {bad}; valid: {list(self.CLASSIFIERS)}")
        for ds in (train_dataset, test_dataset):
            if not self._metadata.is_finished(ds):
                raise ValidationError(f"dataset '{ds}' missing or unfinished")
        classifiers = [c.lower() for c in classifiers]
        names = []
        for c in classifiers:
            name = self.result_name(test_dataset, c)
            # the reference DELETES any existing result on POST
            # (builder_image/utils.py:71) — kept deliberately
            self._db.drop_collection(name)
            self._metadata.create_file(name, "builder/sparkml",
                                       parentName=train_dataset,
                                       classifier=c)
            names.append(name)

        def pipeline():
            self._run(train_dataset, test_dataset, modeling_code, classifiers)

        self._scheduler.submit(f"builder:{test_dataset}", pipeline,
                               device=self._device)
        return names

    # -- the actual pipeline -------------------------------------------------
    def _run(self, train_dataset: str, test_dataset: str, modeling_code: str,
             classifiers: List[str]) -> None:
        import traceback

        import pandas as pd

        from ..storage import Data
        data = Data(self._db, self._artifacts)
        try:
            ctx = self._exec_modeling_code(data, train_dataset, test_dataset,
                                           modeling_code)
        except BaseException:
            tb = traceback.format_exc()
            for c in classifiers:
                self._metadata.update_finished_flag(
                    self.result_name(test_dataset, c), True, exception=tb)
            return

        for c in classifiers:
            name = self.result_name(test_dataset, c)
            try:
                self._run_classifier(c, ctx, name)
            except BaseException:
                self._metadata.update_finished_flag(
                    name, True, exception=traceback.format_exc())

    def _exec_modeling_code(self, data, train_dataset: str, test_dataset: str,
                            code: str) -> Dict[str, Any]:
        """Run the user's preprocessing code. Contract (reference
        builder.py:84-105): the code sees `training_df` / `testing_df` and
        must produce `features_training`, `features_testing`,
        `features_evaluation` (label column named 'label')."""
        import numpy as np
        import pandas as pd
        ctx: Dict[str, Any] = {
            "np": np, "numpy": np, "pd": pd, "pandas": pd,
            "training_df": data.get_dataset_content(train_dataset),
            "testing_df": data.get_dataset_content(test_dataset),
        }
        exec(code, ctx)  # noqa: S102 - the documented builder surface
        for req in ("features_training", "features_evaluation", "features_testing"):
            if req not in ctx:
                raise ValidationError(f"modelingCode must define '{req}'")
        return ctx

    def _run_classifier(self, kind: str, ctx: Dict[str, Any], name: str) -> None:
        import numpy as np

        from ..models.tabular import make_classifier
        Xtr, ytr = _to_xy(ctx["features_training"])
        Xev, yev = _to_xy(ctx["features_evaluation"])
        Xte, _ = _to_xy(ctx["features_testing"], need_label=False)

        clf = make_classifier(kind, device=self._device)
        t0 = time.time()
        clf.fit(Xtr, ytr)
        fit_time = time.time() - t0
        pred_ev = clf.predict(Xev)
        metrics = _classification_metrics(yev, pred_ev)
        pred_te = clf.predict(Xte)
        proba = clf.predict_proba(Xte) if hasattr(clf, "predict_proba") else None

        dst = self._db[name]
        docs = []
        for i in range(len(pred_te)):
            row = {"_id": i + 1, "prediction": int(pred_te[i])}
            if proba is not None:
                row["probability"] = [float(p) for p in proba[i]]
            docs.append(row)
        if docs:
            dst.insert_many(docs)
        self._metadata.update_fields(name, fitTime=fit_time, **metrics)
        self._metadata.update_finished_flag(name, True)


def _to_xy(df, need_label: bool = True):
    import numpy as np
    import pandas as pd
    if isinstance(df, pd.DataFrame):
        if "label" in df.columns:
            y = df["label"].to_numpy(dtype=np.float32)
            X = df.drop(columns=["label"]).to_numpy(dtype=np.float32)
        else:
            if need_label:
                raise ValidationError("features dataframe needs a 'label' column")
            y = None
            X = df.to_numpy(dtype=np.float32)
        return X, y
    arr = np.asarray(df, dtype=np.float32)
    return arr, None


def _classification_metrics(y_true, y_pred) -> Dict[str, float]:
    """accuracy + macro-F1 (the reference records the
    MulticlassClassificationEvaluator's f1 and accuracy, builder.py:127-139)."""
    import numpy as np
    y_true = np.asarray(y_true).astype(int)
    y_pred = np.asarray(y_pred).astype(int)
    acc = float((y_true == y_pred).mean()) if len(y_true) else 0.0
    f1s = []
    for cls in np.unique(y_true):
        tp = int(((y_pred == cls) & (y_true == cls)).sum())
        fp = int(((y_pred == cls) & (y_true != cls)).sum())
        fn = int(((y_pred != cls) & (y_true == cls)).sum())
        prec = tp / (tp + fp) if tp + fp else 0.0
        rec = tp / (tp + fn) if tp + fn else 0.0
        f1s.append(2 * prec * rec / (prec + rec) if prec + rec else 0.0)
    return {"accuracy": acc, "f1": float(np.mean(f1s)) if f1s else 0.0}
