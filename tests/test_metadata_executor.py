"""Metadata contract, $/# parameter protocol, scheduler, reflective pipelines."""
import time

import pytest

from learningorchestra_amd.executor.execution import Execution, ValidationError
from learningorchestra_amd.executor.parameters import Parameters, UserCodeDisabled
from learningorchestra_amd.executor.scheduler import JobScheduler
from learningorchestra_amd.storage import Data, Metadata


def wait_finished(metadata, name, timeout=10.0):
    deadline = time.time() + timeout
    while time.time() < deadline:
        if metadata.is_finished(name):
            return metadata.get_metadata(name)
        time.sleep(0.01)
    raise TimeoutError(f"{name} never finished")


def test_metadata_finished_flag_contract(db):
    md = Metadata(db)
    doc = md.create_file("ds", "dataset/csv", url="file:///x.csv")
    assert doc["finished"] is False and doc["_id"] == 0
    assert not md.is_finished("ds")
    md.update_file_headers("ds", ["a", "b"])
    md.update_finished_flag("ds", True)
    got = md.get_metadata("ds")
    assert got["finished"] is True and got["fields"] == ["a", "b"]


def test_execution_documents_monotonic(db):
    md = Metadata(db)
    md.create_file("m", "model/torch")
    i1 = md.create_execution_document("m", "first", {"a": 1})
    i2 = md.create_execution_document("m", "second", {"a": 2})
    assert (i1, i2) == (1, 2)
    doc = db["m"].find_one({"_id": 2})
    assert doc["description"] == "second" and doc["exception"] is None


def test_lineage_walk_to_model(db):
    md = Metadata(db)
    md.create_file("model0", "model/torch", modulePath="learningorchestra_amd.models",
                   className="MnistCNN")
    md.create_file("trained", "train/torch", parentName="model0")
    md.create_file("preds", "predict/torch", parentName="trained")
    owner = md.walk_to_model("preds")
    assert owner["datasetName"] == "model0"
    chain = md.lineage("preds")
    assert [c["datasetName"] for c in chain] == ["preds", "trained", "model0"]


def test_catalog_by_type(db):
    md = Metadata(db)
    md.create_file("a", "dataset/csv")
    md.create_file("b", "model/torch")
    assert {d["datasetName"] for d in md.catalog()} == {"a", "b"}
    assert [d["datasetName"] for d in md.catalog("model/torch")] == ["b"]


def test_parameters_dollar_and_hash(db, artifacts):
    db["nums"].insert_many([{"_id": 0, "type": "dataset/csv", "finished": True},
                            {"_id": 1, "x": 1.0, "y": 2.0},
                            {"_id": 2, "x": 3.0, "y": 4.0}])
    data = Data(db, artifacts)
    params = Parameters(data)
    out = params.treat({"df": "$nums", "col": "$nums.x", "lit": 5,
                        "expr": "#1 + 2", "lst": ["$nums.y", "plain"]})
    assert list(out["df"].columns) == ["x", "y"]
    assert list(out["col"]) == [1.0, 3.0]
    assert out["lit"] == 5 and out["expr"] == 3
    assert list(out["lst"][0]) == [2.0, 4.0] and out["lst"][1] == "plain"


def test_parameters_user_code_gate(db, artifacts):
    params = Parameters(Data(db, artifacts), allow_user_code=False)
    with pytest.raises(UserCodeDisabled):
        params.treat({"bad": "#__import__('os').getpid()"})


def test_scheduler_records_exception(db):
    md = Metadata(db)
    md.create_file("boom", "train/torch")
    sched = JobScheduler(md)

    def fail():
        raise RuntimeError("kaboom")

    job = sched.submit("boom", fail)
    with pytest.raises(RuntimeError):
        job.wait(5)
    doc = md.get_metadata("boom")
    assert doc["finished"] is True and "kaboom" in doc["exception"]


def test_model_verb_sklearn_roundtrip(db, artifacts):
    """Model verb against sklearn (the reference's scikit-learn tool surface,
    model_image/model.py:112-156) — instantiate, persist, reload."""
    ex = Execution(db, artifacts)
    ex.create_model("lr0", "model/scikitlearn", "sklearn.linear_model",
                    "LogisticRegression", {"max_iter": 50})
    wait_finished(ex.metadata, "lr0")
    inst = artifacts.load("lr0", "model/scikitlearn")
    assert type(inst).__name__ == "LogisticRegression"
    assert inst.max_iter == 50


def test_binary_executor_train_predict_chain(db, artifacts):
    """train/* persists the fitted model; predict stores the result
    (binary_execution.py:147-189)."""
    import numpy as np
    ex = Execution(db, artifacts)
    ex.create_model("lr0", "model/scikitlearn", "sklearn.linear_model",
                    "LogisticRegression", {"max_iter": 200})
    wait_finished(ex.metadata, "lr0")

    x = np.array([[0.0], [1.0], [2.0], [3.0]])
    y = np.array([0, 0, 1, 1])
    ex.create_binary_execution("lr_t", "train/scikitlearn", "lr0", "fit",
                               {"X": "#[[0.0],[1.0],[2.0],[3.0]]",
                                "y": "#[0,0,1,1]"})
    wait_finished(ex.metadata, "lr_t")
    fitted = artifacts.load("lr_t", "train/scikitlearn")
    assert fitted.score(x, y) >= 0.75

    ex.create_binary_execution("lr_p", "predict/scikitlearn", "lr_t", "predict",
                               {"X": "#[[0.0],[3.0]]"})
    wait_finished(ex.metadata, "lr_p")
    preds = artifacts.load("lr_p", "predict/scikitlearn")
    assert list(preds) == [0, 1]
    # lineage walks back to the model
    assert ex.metadata.walk_to_model("lr_p")["datasetName"] == "lr0"


def test_code_executor_function_python(db, artifacts):
    """function/python: exec + stdout capture + response artifact
    (code_execution.py:149-196)."""
    ex = Execution(db, artifacts)
    ex.create_code_execution("fn0", "function/python",
                             "print('hello')\nresponse = a + b", {"a": 2, "b": 3})
    wait_finished(ex.metadata, "fn0")
    doc = db["fn0"].find_one({"_id": 1})
    assert doc["functionMessage"] == "hello\n"
    assert artifacts.load("fn0", "function/python") == 5


def test_code_executor_exception_is_data(db, artifacts):
    ex = Execution(db, artifacts)
    ex.create_code_execution("fnerr", "function/python", "response = 1/0", {})
    meta = wait_finished(ex.metadata, "fnerr")
    assert "ZeroDivisionError" in meta["exception"]


def test_validation_errors(db, artifacts):
    ex = Execution(db, artifacts)
    with pytest.raises(ValidationError):
        ex.create_model("x", "model/scikitlearn", "sklearn.no_such_module", "Nope", {})
    with pytest.raises(ValidationError):
        ex.create_model("x", "model/scikitlearn", "sklearn.linear_model", "Nope", {})
    with pytest.raises(ValidationError):
        ex.create_binary_execution("y", "train/scikitlearn", "missing_parent", "fit", {})


def test_scheduler_device_serialization_and_concurrency():
    """GPU-tagged jobs serialize per device; CPU jobs run concurrently;
    failures are data (finished flag + exception), not crashes."""
    import threading
    import time as _t
    from learningorchestra_amd.executor.scheduler import JobScheduler
    from learningorchestra_amd.storage.docstore import DocumentStore
    from learningorchestra_amd.storage.metadata import Metadata

    md = Metadata(DocumentStore())
    sched = JobScheduler(md, max_workers=8)

    active = {"n": 0, "max": 0}
    lock = threading.Lock()

    def dev_job():
        with lock:
            active["n"] += 1
            active["max"] = max(active["max"], active["n"])
        _t.sleep(0.02)
        with lock:
            active["n"] -= 1

    jobs = [sched.submit(f"d{i}", dev_job, device="cuda:0") for i in range(6)]
    for j in jobs:
        j.wait(10)
    assert active["max"] == 1, "device jobs must serialize"

    # concurrent CPU jobs overlap
    active["max"] = 0
    jobs = [sched.submit(f"c{i}", dev_job) for i in range(6)]
    for j in jobs:
        j.wait(10)
    assert active["max"] >= 2, "cpu jobs should overlap"

    # a failing job records its exception into metadata
    md.create_file("boom", "train/torch")

    def bad():
        raise RuntimeError("kaput")

    j = sched.submit("boom", bad)
    try:
        j.wait(10)
    except RuntimeError:
        pass
    doc = md.get_metadata("boom")
    assert doc.get("finished") is True
    assert "kaput" in (doc.get("exception") or "")
