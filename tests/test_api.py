"""REST surface tests: the reference's URI scheme, field names, envelope,
status codes and poll contract, driven end-to-end (the Titanic plumbing
config from BASELINE.json runs entirely through HTTP here)."""
import json
import time

import pytest
from fastapi.testclient import TestClient

from learningorchestra_amd.api.app import PREFIX, Runtime, create_app
from learningorchestra_amd.data.synthetic import titanic_csv


@pytest.fixture()
def client(tmp_config, tmp_path):
    rt = Runtime(tmp_config)
    app = create_app(rt)
    with TestClient(app) as c:
        c.rt = rt
        yield c


def wait_finished(client, name, timeout=30.0):
    r = client.get(f"{PREFIX}/observe/{name}/wait",
                   params={"timeoutSeconds": timeout})
    assert r.status_code == 200
    doc = r.json()["result"]
    assert doc is not None and doc.get("finished"), doc
    return doc


def ingest_titanic(client, tmp_path, name="titanic"):
    p = tmp_path / "titanic.csv"
    p.write_text(titanic_csv(rows=200))
    r = client.post(f"{PREFIX}/dataset/csv",
                    json={"datasetName": name, "datasetURI": str(p)})
    assert r.status_code == 201
    assert r.json()["result"].startswith(f"{PREFIX}/dataset/csv/{name}")
    return wait_finished(client, name)


def test_dataset_csv_contract(client, tmp_path):
    meta = ingest_titanic(client, tmp_path)
    assert meta["type"] == "dataset/csv" and "PassengerId" in meta["fields"]
    # catalog
    r = client.get(f"{PREFIX}/dataset/csv")
    assert [d["datasetName"] for d in r.json()["result"]] == ["titanic"]
    # paged rows: first doc is the _id:0 metadata document (reference
    # database.read_file semantics)
    r = client.get(f"{PREFIX}/dataset/csv/titanic",
                   params={"query": "{}", "limit": 3, "skip": 0})
    rows = r.json()["result"]
    assert rows[0]["_id"] == 0 and rows[1]["_id"] == 1
    # filtered query
    r = client.get(f"{PREFIX}/dataset/csv/titanic",
                   params={"query": json.dumps({"Sex": "male"}), "limit": 5})
    assert all(x["Sex"] == "male" for x in r.json()["result"])
    # duplicate -> 409
    r = client.post(f"{PREFIX}/dataset/csv",
                    json={"datasetName": "titanic", "datasetURI": "x.csv"})
    assert r.status_code == 409
    # missing -> 404
    assert client.get(f"{PREFIX}/dataset/csv/nope").status_code == 404


def test_projection_and_datatype_and_histogram(client, tmp_path):
    ingest_titanic(client, tmp_path)
    r = client.post(f"{PREFIX}/transform/projection",
                    json={"inputDatasetName": "titanic",
                          "outputDatasetName": "titanic_p",
                          "names": ["Sex", "Age", "Survived"]})
    assert r.status_code == 201
    wait_finished(client, "titanic_p")
    r = client.get(f"{PREFIX}/transform/projection/titanic_p",
                   params={"limit": 2, "skip": 1})
    row = r.json()["result"][0]
    assert set(row) == {"_id", "Sex", "Age", "Survived"}
    # invalid field -> 406
    r = client.post(f"{PREFIX}/transform/projection",
                    json={"inputDatasetName": "titanic",
                          "outputDatasetName": "bad", "names": ["NoSuch"]})
    assert r.status_code == 406

    # dataType: stringify then re-numberify Age
    r = client.patch(f"{PREFIX}/transform/dataType",
                     json={"datasetName": "titanic_p",
                           "types": {"Age": "string"}})
    assert r.status_code == 200
    wait_finished(client, "titanic_p")
    row = client.get(f"{PREFIX}/dataset/csv/titanic_p",
                     params={"limit": 2, "skip": 1}).json()["result"][0]
    assert row["Age"] is None or isinstance(row["Age"], str)
    client.patch(f"{PREFIX}/transform/dataType",
                 json={"datasetName": "titanic_p", "types": {"Age": "number"}})
    wait_finished(client, "titanic_p")

    # histogram
    r = client.post(f"{PREFIX}/explore/histogram",
                    json={"inputDatasetName": "titanic",
                          "outputDatasetName": "hist_sex", "names": ["Sex"]})
    assert r.status_code == 201
    wait_finished(client, "hist_sex")
    rows = client.get(f"{PREFIX}/explore/histogram/hist_sex",
                      params={"limit": 10}).json()["result"]
    hist = next(x for x in rows if x.get("field") == "Sex")
    assert set(hist["values"]) == {"male", "female"}
    assert sum(hist["values"].values()) == 200


def test_model_train_predict_evaluate_sklearn(client, tmp_path):
    """The reference's canonical sklearn chain over REST: model -> train ->
    predict -> evaluate, with $dataset parameter resolution."""
    ingest_titanic(client, tmp_path)
    client.post(f"{PREFIX}/transform/projection",
                json={"inputDatasetName": "titanic", "outputDatasetName": "feat",
                      "names": ["Pclass", "SibSp", "Parch", "Fare", "Survived"]})
    wait_finished(client, "feat")

    r = client.post(f"{PREFIX}/model/scikitlearn",
                    json={"modelName": "lr_model",
                          "modulePath": "sklearn.linear_model",
                          "class": "LogisticRegression",
                          "classParameters": {"max_iter": 200}})
    assert r.status_code == 201
    wait_finished(client, "lr_model")

    r = client.post(f"{PREFIX}/train/scikitlearn",
                    json={"name": "lr_trained", "modelName": "lr_model",
                          "parentName": "lr_model", "method": "fit",
                          "methodParameters": {"X": "#[[0.],[1.],[2.],[3.]]",
                                               "y": "#[0,0,1,1]"}})
    assert r.status_code == 201
    doc = wait_finished(client, "lr_trained")
    assert doc.get("exception") in (None, "")

    r = client.post(f"{PREFIX}/predict/scikitlearn",
                    json={"name": "lr_pred", "modelName": "lr_model",
                          "parentName": "lr_trained", "method": "predict",
                          "methodParameters": {"X": "#[[0],[3]]"}})
    assert r.status_code == 201
    doc = wait_finished(client, "lr_pred")
    assert doc.get("exception") in (None, "")
    # evaluate verb (score) on the same chain
    r = client.post(f"{PREFIX}/evaluate/scikitlearn",
                    json={"name": "lr_eval", "modelName": "lr_model",
                          "parentName": "lr_trained", "method": "score",
                          "methodParameters": {"X": "#[[0],[3]]", "y": "#[0,1]"}})
    assert r.status_code == 201
    wait_finished(client, "lr_eval")
    # lineage: predict walks back to the model (binary_executor utils.py:257)
    assert client.rt.metadata.walk_to_model("lr_pred")["datasetName"] == "lr_model"


def test_function_python_verb(client):
    r = client.post(f"{PREFIX}/function/python",
                    json={"name": "fn1", "function": "print('hi')\nresponse = a * 2",
                          "functionParameters": {"a": 21}})
    assert r.status_code == 201
    wait_finished(client, "fn1")
    rows = client.get(f"{PREFIX}/function/python/fn1",
                      params={"limit": 10}).json()["result"]
    exec_doc = next(x for x in rows if x["_id"] == 1)
    assert exec_doc["functionMessage"] == "hi\n"
    assert client.rt.artifacts.load("fn1", "function/python") == 42


def test_builder_sparkml_verb(client, tmp_path):
    """builder/sparkml: modelingCode + native {lr,nb} classifiers end-to-end
    (dt/rf/gb covered in tree tests; two classifiers keep this test fast)."""
    ingest_titanic(client, tmp_path)
    code = (
        "import pandas as pd\n"
        "def prep(df):\n"
        "    out = df[['Pclass','SibSp','Parch','Fare']].copy()\n"
        "    out['sex_n'] = (df['Sex'] == 'female').astype(float)\n"
        "    out['label'] = df['Survived'].astype(float)\n"
        "    return out.fillna(0.0)\n"
        "features_training = prep(training_df)\n"
        "features_evaluation = prep(testing_df)\n"
        "features_testing = prep(testing_df).drop(columns=['label'])\n")
    r = client.post(f"{PREFIX}/builder/sparkml",
                    json={"trainDatasetName": "titanic",
                          "testDatasetName": "titanic",
                          "modelingCode": code,
                          "classifiersList": ["lr", "nb"]})
    assert r.status_code == 201
    uris = r.json()["result"]
    assert len(uris) == 2
    for c in ("lr", "nb"):
        doc = wait_finished(client, f"titanic{c}", timeout=120)
        assert doc.get("exception") in (None, ""), doc
        assert doc["accuracy"] > 0.5 and doc["fitTime"] > 0
        rows = client.get(f"{PREFIX}/builder/sparkml/titanic{c}",
                          params={"limit": 5, "skip": 1}).json()["result"]
        assert all("prediction" in x for x in rows)
    # invalid classifier -> 406
    r = client.post(f"{PREFIX}/builder/sparkml",
                    json={"trainDatasetName": "titanic",
                          "testDatasetName": "titanic",
                          "modelingCode": code, "classifiersList": ["svm"]})
    assert r.status_code == 406


def test_torch_tool_model_verb(client):
    """The native 'torch' tool: instantiate an engine model via the model
    verb (tensorflow module paths route here too)."""
    r = client.post(f"{PREFIX}/model/torch",
                    json={"modelName": "cnn0",
                          "modulePath": "learningorchestra_amd.models.zoo",
                          "class": "MnistCNN",
                          "classParameters": {"seed": 1}})
    assert r.status_code == 201
    doc = wait_finished(client, "cnn0")
    assert doc.get("exception") in (None, "")


def test_delete_and_404(client, tmp_path):
    ingest_titanic(client, tmp_path)
    assert client.delete(f"{PREFIX}/dataset/csv/titanic").status_code == 200
    assert client.get(f"{PREFIX}/dataset/csv/titanic").status_code == 404
    assert client.delete(f"{PREFIX}/dataset/csv/titanic").status_code == 404


def test_explore_plot_png(client, tmp_path):
    """explore/{tool} with the native Plot class renders a PNG served on GET
    (reference seaborn-scatterplot-to-PNG path)."""
    ingest_titanic(client, tmp_path)
    r = client.post(f"{PREFIX}/explore/torch",
                    json={"name": "plot1",
                          "modulePath": "learningorchestra_amd.models.explore",
                          "class": "Plot", "classParameters": {},
                          "method": "scatter",
                          "methodParameters": {"data": "$titanic",
                                               "x": "Age", "y": "Fare"}})
    assert r.status_code == 201
    wait_finished(client, "plot1")
    resp = client.get(f"{PREFIX}/explore/torch/plot1")
    assert resp.status_code == 200
    assert resp.headers["content-type"] == "image/png"
    assert resp.content[:8] == b"\x89PNG\r\n\x1a\n"


def test_concurrent_creates_unique_names(client):
    """The create race the reference had (TOCTOU): N threads
    POSTing the SAME name must yield exactly one 201 and the rest 409."""
    import threading
    csv_text = titanic_csv(40)
    codes = []
    lock = threading.Lock()

    def post():
        r = client.post(f"{PREFIX}/dataset/generic",
                        json={"datasetName": "race", "datasetURI":
                              "data:text/plain," + csv_text})
        with lock:
            codes.append(r.status_code)

    threads = [threading.Thread(target=post) for _ in range(8)]
    for t in threads:
        t.start()
    for t in threads:
        t.join()
    assert sorted(codes).count(201) == 1, codes
    assert all(c in (201, 409) for c in codes), codes


def test_concurrent_distinct_creates_all_succeed(client):
    import threading
    csv_text = titanic_csv(30)
    codes = {}

    def post(i):
        r = client.post(f"{PREFIX}/dataset/generic",
                        json={"datasetName": f"cc{i}", "datasetURI":
                              "data:text/plain," + csv_text})
        codes[i] = r.status_code

    threads = [threading.Thread(target=post, args=(i,)) for i in range(8)]
    for t in threads:
        t.start()
    for t in threads:
        t.join()
    assert all(v == 201 for v in codes.values()), codes
    for i in range(8):
        wait_finished(client, f"cc{i}")


def test_negative_paths_status_codes(client):
    """Reference status-code parity: 406 invalid tool/fields, 404 unknown."""
    # unknown tool = unmapped gateway path in the reference -> 404
    r = client.post(f"{PREFIX}/dataset/nosuchtool",
                    json={"datasetName": "x", "datasetURI": "file:///nope"})
    assert r.status_code == 404
    # missing required field
    r = client.post(f"{PREFIX}/dataset/csv", json={"datasetName": "x"})
    assert r.status_code == 406
    # model verb with an invalid executor tool (unmapped path)
    r = client.post(f"{PREFIX}/model/spark",
                    json={"modelName": "m", "modulePath": "m", "class": "C"})
    assert r.status_code == 404
    # binary verb on an unknown parent
    r = client.post(f"{PREFIX}/train/torch",
                    json={"name": "t1", "parentName": "ghost",
                          "method": "fit"})
    assert r.status_code in (404, 406)
    # observe unknown name
    assert client.get(f"{PREFIX}/observe/ghost").status_code == 404
    # metadata of unknown name
    assert client.get(f"{PREFIX}/train/torch/ghost/metadata").status_code == 404
    # builder with an invalid classifier
    r = client.post(f"{PREFIX}/builder/sparkml",
                    json={"trainDatasetName": "ghost", "testDatasetName":
                          "ghost", "modelingCode": "", "classifiersList":
                          ["nope"]})
    assert r.status_code == 406
    # invalid class parameters on a real module (constructor check)
    r = client.post(f"{PREFIX}/model/scikitlearn",
                    json={"modelName": "badparam",
                          "modulePath": "sklearn.linear_model",
                          "class": "LogisticRegression",
                          "classParameters": {"definitely_not_an_arg": 1}})
    assert r.status_code == 406


def test_observe_wait_timeout_flag(client, tmp_path):
    """observe/{name}/wait returns timedOut=true when the flag stays false."""
    client.rt.metadata.create_file("slowjob", "train/torch")
    r = client.get(f"{PREFIX}/observe/slowjob/wait",
                   params={"timeoutSeconds": 0.2})
    assert r.status_code == 200
    body = r.json()
    assert body.get("timedOut") is True
    assert body["result"]["finished"] is False


def test_metrics_endpoint(client, tmp_path):
    ingest_titanic(client, tmp_path)
    r = client.get(f"{PREFIX}/metrics")
    assert r.status_code == 200
    m = r.json()["result"]
    assert "artifactsByType" in m and "device" in m and "gpu" in m
    assert m["collections"] >= 1


def test_builder_multiclass_mnist_demo(client, tmp_path):
    """The MNIST builder demo shape (reference README.md:63): a 10-class
    dataset through builder/sparkml with dt + rf must return real multiclass
    accuracy, not silently-clamped binary garbage (r1 VERDICT missing #1)."""
    import io

    from learningorchestra_amd.data.synthetic import tabular_multiclass
    X, y = tabular_multiclass(1500, 12, n_classes=10, seed=7)
    buf = io.StringIO()
    cols = [f"px{i}" for i in range(12)]
    buf.write(",".join(cols) + ",digit\n")
    for i in range(X.shape[0]):
        buf.write(",".join(f"{v:.4f}" for v in X[i].tolist())
                  + f",{int(y[i])}\n")
    p = tmp_path / "mnist_small.csv"
    p.write_text(buf.getvalue())
    r = client.post(f"{PREFIX}/dataset/csv",
                    json={"datasetName": "mnist10", "datasetURI": str(p)})
    assert r.status_code == 201
    wait_finished(client, "mnist10")
    code = (
        "feat = training_df.drop(columns=['_id'], errors='ignore').copy()\n"
        "feat['label'] = feat.pop('digit').astype(float)\n"
        "features_training = feat\n"
        "features_evaluation = feat\n"
        "features_testing = feat.drop(columns=['label'])\n")
    r = client.post(f"{PREFIX}/builder/sparkml",
                    json={"trainDatasetName": "mnist10",
                          "testDatasetName": "mnist10",
                          "modelingCode": code,
                          "classifiersList": ["dt", "rf"]})
    assert r.status_code == 201
    for c in ("dt", "rf"):
        doc = wait_finished(client, f"mnist10{c}", timeout=120)
        assert doc.get("exception") in (None, ""), doc
        # 10-class majority is ~0.1; real multiclass trees clear 0.5 easily
        assert doc["accuracy"] > 0.5, (c, doc["accuracy"])
        rows = client.get(f"{PREFIX}/builder/sparkml/mnist10{c}",
                          params={"limit": 3, "skip": 1}).json()["result"]
        assert all(len(x["probability"]) == 10 for x in rows)
        assert any(x["prediction"] > 1 for x in rows)


def test_train_torch_multi_rank(client):
    """An API train/torch POST with "gpus": 2 spawns a 2-rank torchrun job
    (gloo on CPU hosts, RCCL on GPUs) through the scheduler's process-job
    path (r1 VERDICT missing #2: the API must reach multi-GPU training)."""
    r = client.post(f"{PREFIX}/model/torch",
                    json={"modelName": "ddpcnn",
                          "modulePath": "learningorchestra_amd.models.zoo",
                          "class": "MnistCNN",
                          "classParameters": {"channels": [8, 8],
                                              "fc_width": 32,
                                              "device": "cpu"}})
    assert r.status_code == 201
    wait_finished(client, "ddpcnn")
    r = client.post(f"{PREFIX}/train/torch",
                    json={"name": "ddptrain", "modelName": "ddpcnn",
                          "method": "fit",
                          "methodParameters": {
                              "gpus": 2,
                              "x": "#numpy.random.RandomState(0)"
                                   ".rand(64,784).astype('float32')",
                              "y": "#numpy.random.RandomState(1)"
                                   ".randint(0,10,64)",
                              "epochs": 1, "batch_size": 16}})
    assert r.status_code == 201
    doc = wait_finished(client, "ddptrain", timeout=240)
    assert doc.get("exception") in (None, ""), doc
    rows = client.get(f"{PREFIX}/train/torch/ddptrain",
                      params={"limit": 10}).json()["result"]
    exec_doc = next(x for x in rows if x["_id"] == 1)
    assert exec_doc["worldSize"] == 2
    assert exec_doc["executionParameters"]["gpus"] == 2
    # the fitted model artifact was persisted by rank 0
    assert client.rt.artifacts.exists("ddptrain", "train/torch")
    fitted = client.rt.artifacts.load("ddptrain", "train/torch",
                                      device="cpu")
    assert fitted.predict(
        __import__("numpy").random.rand(4, 784).astype("float32")).shape == (4,)


def test_gateway_response_cache(tmp_path, monkeypatch):
    """Reference-parity GET cache (krakend cache_ttl 300 s): repeated reads
    are served from cache within the TTL; any mutation invalidates."""
    from learningorchestra_amd.api.app import Runtime, create_app
    from learningorchestra_amd.config import Config, set_config
    cfg = Config(data_root=str(tmp_path), mongo_uri="", cache_ttl=300.0)
    set_config(cfg)
    try:
        rt = Runtime(cfg)
        client = TestClient(create_app(rt))
        client.rt = rt
        ingest_titanic(client, tmp_path, name="ct")
        r1 = client.get(f"{PREFIX}/dataset/csv/ct", params={"limit": 3})
        assert r1.status_code == 200
        # mutate the collection BEHIND the API: cached read doesn't see it
        rt.db["ct"].update_one({"_id": 1}, {"$set": {"Sex": "mutated"}})
        r2 = client.get(f"{PREFIX}/dataset/csv/ct", params={"limit": 3})
        assert r2.json() == r1.json()
        # an API mutation invalidates the cache
        r = client.post(f"{PREFIX}/transform/projection",
                        json={"inputDatasetName": "ct",
                              "outputDatasetName": "ct_p",
                              "names": ["Sex"]})
        assert r.status_code == 201
        r3 = client.get(f"{PREFIX}/dataset/csv/ct", params={"limit": 3})
        assert any(x.get("Sex") == "mutated" for x in r3.json()["result"])
    finally:
        set_config(None)


def test_observe_many_concurrent_waiters(client):
    """100 concurrent Observe waiters park on events (not worker threads) and
    all wake when the flag flips; normal traffic stays responsive."""
    import concurrent.futures
    import threading

    client.rt.metadata.create_file("slowjob", "train/torch")

    def waiter(_i):
        r = client.get(f"{PREFIX}/observe/slowjob/wait",
                       params={"timeoutSeconds": 30})
        return r.json()["result"].get("finished")

    with concurrent.futures.ThreadPoolExecutor(max_workers=104) as pool:
        futs = [pool.submit(waiter, i) for i in range(100)]
        # normal traffic while the waiters are parked
        time.sleep(0.3)
        assert client.get(f"{PREFIX}/metrics").status_code == 200
        assert not any(f.done() and f.result() for f in futs[:5])
        threading.Timer(
            0.2, lambda: client.rt.metadata.update_finished_flag(
                "slowjob", True)).start()
        results = [f.result(timeout=60) for f in futs]
    assert all(results), results.count(False)


def test_tune_torch_gridsearch_e2e(client):
    """The Tune verb end-to-end on the native tool: a GridSearch artifact
    created via model/torch, fitted via tune/torch, then queried for the
    winning candidate (reference path: tune/* through the binary executor,
    SURVEY §2.1; native scheduler in models/tuning.py)."""
    r = client.post(f"{PREFIX}/model/torch",
                    json={"modelName": "gs0",
                          "modulePath": "learningorchestra_amd.models.tuning",
                          "class": "GridSearch",
                          "classParameters": {
                              "modulePath":
                                  "learningorchestra_amd.models.tabular",
                              "className": "LogisticRegressionClassifier",
                              "paramGrid": {"lr": [0.05, 0.2]},
                              "fixedParameters": {"epochs": 40},
                              "validationSplit": 0.25}})
    assert r.status_code == 201
    wait_finished(client, "gs0")

    # separable 1-feature data: y = x > 1.5
    xs = "#[[float(i % 4)] for i in range(80)]"
    ys = "#[1 if (i % 4) > 1 else 0 for i in range(80)]"
    r = client.post(f"{PREFIX}/tune/torch",
                    json={"name": "gs_fit", "modelName": "gs0",
                          "parentName": "gs0", "method": "fit",
                          "methodParameters": {"x": xs, "y": ys}})
    assert r.status_code == 201
    doc = wait_finished(client, "gs_fit")
    assert doc.get("exception") in (None, ""), doc

    gs = client.rt.artifacts.load("gs_fit", "tune/torch")
    assert gs.best_params_ is not None and "lr" in gs.best_params_
    assert gs.best_score_ >= 0.5, (gs.best_score_, gs.results_)
    assert len(gs.results_) == 2


def test_patch_rerun_bumps_execution_documents(client):
    """PATCH on a train result re-executes with new parameters and appends
    a new execution document at the next _id (reference: binary_executor
    update_execution, server.py:74-118; docs appended at max+1)."""
    client.post(f"{PREFIX}/model/scikitlearn",
                json={"modelName": "m_p", "modulePath": "sklearn.linear_model",
                      "class": "LogisticRegression",
                      "classParameters": {"max_iter": 100}})
    wait_finished(client, "m_p")
    client.post(f"{PREFIX}/train/scikitlearn",
                json={"name": "t_p", "modelName": "m_p", "parentName": "m_p",
                      "method": "fit",
                      "methodParameters": {"X": "#[[0.],[1.],[2.],[3.]]",
                                           "y": "#[0,0,1,1]"}})
    wait_finished(client, "t_p")
    docs_before = client.rt.db["t_p"].count_documents({})

    r = client.patch(f"{PREFIX}/train/scikitlearn/t_p",
                     json={"methodParameters": {"X": "#[[0.],[1.],[2.],[5.]]",
                                                "y": "#[0,0,1,1]"},
                           "description": "re-fit with shifted data"})
    assert r.status_code == 200
    doc = wait_finished(client, "t_p")
    assert doc.get("exception") in (None, "")
    docs_after = client.rt.db["t_p"].count_documents({})
    assert docs_after == docs_before + 1
    # the newest execution doc carries the new description
    newest = max(client.rt.db["t_p"].find({"_id": {"$gt": 0}}),
                 key=lambda d: d["_id"])
    assert "re-fit" in str(newest.get("description", "")) or \
        "re-fit" in str(newest)


def test_train_torch_multi_rank_failure_records_exception(client):
    """A failing N-rank torchrun job must flip finished=true WITH the
    exception recorded (errors are data — reference parity), not hang the
    poll contract."""
    client.post(f"{PREFIX}/model/torch",
                json={"modelName": "ddpbad",
                      "modulePath": "learningorchestra_amd.models.zoo",
                      "class": "MnistCNN",
                      "classParameters": {"channels": [8, 8],
                                          "fc_width": 32, "device": "cpu"}})
    wait_finished(client, "ddpbad")
    r = client.post(f"{PREFIX}/train/torch",
                    json={"name": "ddpfail", "modelName": "ddpbad",
                          "method": "fit",
                          "methodParameters": {
                              "gpus": 2,
                              # wrong feature width: every rank raises
                              "x": "#numpy.random.RandomState(0)"
                                   ".rand(16,7).astype('float32')",
                              "y": "#numpy.random.RandomState(1)"
                                   ".randint(0,10,16)",
                              "epochs": 1, "batch_size": 8}})
    assert r.status_code == 201
    r = client.get(f"{PREFIX}/observe/ddpfail/wait",
                   params={"timeoutSeconds": 240})
    doc = r.json()["result"]
    assert doc is not None and doc.get("finished"), doc
    assert doc.get("exception"), doc


def test_cancel_running_multi_rank_train(client):
    """POST /cancel/{name} kills a RUNNING N-rank train's process group and
    records the cancellation (r2: real job control vs the reference's
    restart-the-service answer)."""
    client.post(f"{PREFIX}/model/torch",
                json={"modelName": "ddpslow",
                      "modulePath": "learningorchestra_amd.models.zoo",
                      "class": "MnistCNN",
                      "classParameters": {"channels": [8, 8],
                                          "fc_width": 32, "device": "cpu"}})
    wait_finished(client, "ddpslow")
    r = client.post(f"{PREFIX}/train/torch",
                    json={"name": "ddpcancel", "modelName": "ddpslow",
                          "method": "fit",
                          "methodParameters": {
                              "gpus": 2,
                              "x": "#numpy.random.RandomState(0)"
                                   ".rand(4096,784).astype('float32')",
                              "y": "#numpy.random.RandomState(1)"
                                   ".randint(0,10,4096)",
                              "epochs": 500, "batch_size": 64}})
    assert r.status_code == 201
    # let the torchrun tree actually start (generous: a loaded CI host can
    # take a while to fork + import torch in the children)
    deadline = time.time() + 120
    while time.time() < deadline:
        job = client.rt.scheduler.job("ddpcancel")
        if job is not None and job.proc is not None \
                and job.proc.poll() is None:
            break
        time.sleep(0.25)
    else:
        raise AssertionError("torchrun job never started")

    r = client.post(f"{PREFIX}/cancel/ddpcancel")
    assert r.status_code == 200
    assert "cancelled" in r.json()["result"]
    doc = wait_finished(client, "ddpcancel", timeout=60)
    assert doc.get("exception"), doc
    # the process tree is really gone
    deadline = time.time() + 30
    while time.time() < deadline and job.proc.poll() is None:
        time.sleep(0.25)
    assert job.proc.poll() is not None


def test_chaos_concurrent_trains_cancels_observers(client):
    """Stress the scheduler + metadata + docstore stack: 12 concurrent
    sklearn train jobs, observers long-polling each, random cancels racing
    completion, catalog/metrics reads throughout. Every job must terminate
    with finished=true (result or recorded exception) and the API must
    never 5xx."""
    import random
    import threading

    client.post(f"{PREFIX}/model/scikitlearn",
                json={"modelName": "chaos_m",
                      "modulePath": "sklearn.linear_model",
                      "class": "LogisticRegression",
                      "classParameters": {"max_iter": 50}})
    wait_finished(client, "chaos_m")

    names = [f"chaos_t{i}" for i in range(12)]
    errors = []

    def train(name):
        try:
            r = client.post(
                f"{PREFIX}/train/scikitlearn",
                json={"name": name, "modelName": "chaos_m",
                      "parentName": "chaos_m", "method": "fit",
                      "methodParameters": {
                          "X": "#[[float(i)] for i in range(200)]",
                          "y": "#[i % 2 for i in range(200)]"}})
            assert r.status_code in (201, 409), r.status_code
        except Exception as exc:  # noqa: BLE001
            errors.append(("train", name, repr(exc)))

    def observer(name):
        try:
            # observers launch concurrently with the train POSTs: a 404 just
            # means the metadata doc does not exist YET (correct API
            # behavior) — retry until the create lands
            deadline = time.time() + 60
            while True:
                r = client.get(f"{PREFIX}/observe/{name}/wait",
                               params={"timeoutSeconds": 30})
                if r.status_code == 200:
                    break
                assert r.status_code == 404, r.status_code
                assert time.time() < deadline, "doc never created"
                time.sleep(0.1)
        except Exception as exc:  # noqa: BLE001
            errors.append(("observe", name, repr(exc)))

    def chaos_reader(stop):
        rng = random.Random(7)
        while not stop.is_set():
            try:
                assert client.get(f"{PREFIX}/metrics").status_code == 200
                assert client.get(
                    f"{PREFIX}/train/scikitlearn").status_code == 200
                victim = rng.choice(names)
                r = client.post(f"{PREFIX}/cancel/{victim}")
                assert r.status_code in (200, 404), r.status_code
            except Exception as exc:  # noqa: BLE001
                errors.append(("reader", "-", repr(exc)))
                return

    threads = [threading.Thread(target=train, args=(n,)) for n in names]
    threads += [threading.Thread(target=observer, args=(n,)) for n in names]
    stop = threading.Event()
    reader = threading.Thread(target=chaos_reader, args=(stop,))
    for t in threads:
        t.start()
    reader.start()
    for t in threads:
        t.join(timeout=120)
    stop.set()
    reader.join(timeout=30)
    assert not errors, errors[:5]
    for n in names:
        doc = client.rt.metadata.get_metadata(n)
        assert doc is not None and doc.get("finished"), (n, doc)


def test_real_server_boot_and_request(tmp_config):
    """`python -m learningorchestra_amd` boots a real uvicorn server (the
    reference's ./run.sh role); one live HTTP request proves the launcher,
    app factory and config plumbing end to end."""
    import os
    import socket
    import subprocess
    import sys
    import urllib.request

    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    env = dict(os.environ,
               LO_DATA_ROOT=tmp_config.data_root,
               PYTHONPATH=repo)
    proc = subprocess.Popen(
        [sys.executable, "-m", "learningorchestra_amd",
         "--host", "127.0.0.1", "--port", str(port)],
        env=env, cwd=repo,
        stdout=subprocess.PIPE, stderr=subprocess.STDOUT, text=True)
    try:
        deadline = time.time() + 60
        last = None
        while time.time() < deadline:
            try:
                with urllib.request.urlopen(
                        f"http://127.0.0.1:{port}{PREFIX}/metrics",
                        timeout=2) as resp:
                    assert resp.status == 200
                    body = resp.read()
                    assert b"scheduler" in body or b"result" in body
                    break
            except Exception as exc:  # noqa: BLE001 - server still booting
                last = exc
                assert proc.poll() is None, proc.stdout.read()[-2000:]
                time.sleep(0.5)
        else:
            raise AssertionError(f"server never answered: {last!r}")
    finally:
        proc.terminate()
        try:
            proc.wait(timeout=15)
        except subprocess.TimeoutExpired:
            proc.kill()
