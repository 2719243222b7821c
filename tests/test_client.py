"""Python client (learning_orchestra_client) end-to-end in embedded mode —
the full Titanic pipeline through the client API (BASELINE config 1)."""
import pytest

from learning_orchestra_client import (BuilderSparkMl, Context, DatasetCsv,
                                       Evaluate, ExploreHistogram,
                                       FunctionPython, Model, Predict, Train,
                                       TransformDataType, TransformProjection)
from learningorchestra_amd.data.synthetic import titanic_csv


@pytest.fixture()
def ctx(tmp_config, tmp_path):
    from learningorchestra_amd.api.app import Runtime
    return Context.in_process(Runtime(tmp_config))


def test_titanic_pipeline_via_client(ctx, tmp_path):
    p = tmp_path / "titanic.csv"
    p.write_text(titanic_csv(rows=300))

    ds = DatasetCsv(ctx)
    meta = ds.insert_sync("titanic", str(p))
    assert meta["finished"] and len(meta["fields"]) == 12

    proj = TransformProjection(ctx)
    proj.create("titanic", "feat",
                ["Pclass", "SibSp", "Parch", "Fare", "Survived"])
    proj.wait("feat")
    rows = proj.search("feat", limit=3, skip=1)
    assert set(rows[0]) == {"_id", "Pclass", "SibSp", "Parch", "Fare", "Survived"}

    TransformDataType(ctx).convert("feat", {"Fare": "number"})
    TransformDataType(ctx).wait("feat")

    hist = ExploreHistogram(ctx)
    hist.create("titanic", "h1", ["Pclass"])
    hist.wait("h1")
    h = hist.search("h1", limit=5, skip=1)[0]
    assert sum(h["values"].values()) == 300

    model = Model(ctx, tool="scikitlearn")
    model.create("lr", "sklearn.linear_model", "LogisticRegression",
                 {"max_iter": 200})
    model.wait("lr")

    train = Train(ctx, tool="scikitlearn")
    train.create("lr_t", "lr", "fit",
                 {"X": "#[[0.],[1.],[2.],[3.]]", "y": "#[0,0,1,1]"})
    train.wait("lr_t")

    pred = Predict(ctx, tool="scikitlearn")
    pred.create("lr_p", "lr_t", "predict", {"X": "#[[0.],[3.]]"})
    pred.wait("lr_p")

    ev = Evaluate(ctx, tool="scikitlearn")
    ev.create("lr_e", "lr_t", "score", {"X": "#[[0.],[3.]]", "y": "#[0,1]"})
    doc = ev.wait("lr_e")
    assert doc["finished"]

    fn = FunctionPython(ctx)
    fn.run("f1", "response = 6 * 7", {})
    fn.wait("f1")

    builder = BuilderSparkMl(ctx)
    uris = builder.build("titanic", "titanic",
                         "import pandas as pd\n"
                         "def prep(df):\n"
                         "    out = df[['Pclass','Fare']].copy()\n"
                         "    out['label'] = df['Survived'].astype(float)\n"
                         "    return out.fillna(0.0)\n"
                         "features_training = prep(training_df)\n"
                         "features_evaluation = prep(testing_df)\n"
                         "features_testing = prep(testing_df).drop(columns=['label'])\n",
                         ["nb"])
    assert len(uris) == 1
    doc = builder.wait("titanicnb", timeout=120)
    assert doc["accuracy"] > 0.4


def test_client_error_mapping(ctx):
    from learning_orchestra_client.client import LearningOrchestraError
    ds = DatasetCsv(ctx)
    with pytest.raises(LearningOrchestraError) as e:
        ds.search("missing")
    assert e.value.status == 404
    fn = FunctionPython(ctx)
    fn.run("bad", "response = 1/0", {})
    with pytest.raises(LearningOrchestraError) as e:
        fn.wait("bad", timeout=30)
    assert "ZeroDivisionError" in str(e.value)


def test_titanic_example_runs(tmp_path):
    """examples/titanic_pipeline.py is the documented onboarding flow — it
    must stay runnable end-to-end."""
    import subprocess
    import sys
    from pathlib import Path
    script = Path(__file__).parent.parent / "examples" / "titanic_pipeline.py"
    out = subprocess.run([sys.executable, str(script)], capture_output=True,
                         text=True, timeout=300)
    assert out.returncode == 0, out.stderr[-2000:]
    assert "accuracy=" in out.stdout


def test_observe_watch_streams_result_rows(ctx, tmp_path):
    """Observe.watch yields result-collection rows as they appear and stops
    at the finished flag (reference pip client: Mongo change-stream watch on
    the result collection)."""
    from learning_orchestra_client import Observe

    p = tmp_path / "titanic.csv"
    p.write_text(titanic_csv(rows=50))
    DatasetCsv(ctx).insert_sync("t_watch", str(p))

    TransformProjection(ctx).create("t_watch", "w_feat", ["Pclass", "Fare"])
    rows = list(Observe(ctx).watch("w_feat", verb="transform",
                                   tool="projection", timeout=60.0))
    assert len(rows) == 50
    assert all(set(r) >= {"Pclass", "Fare"} for r in rows)
    ids = [r["_id"] for r in rows]
    assert ids == sorted(ids) and ids[0] >= 1
