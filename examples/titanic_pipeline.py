"""End-to-end Titanic pipeline — the reference's canonical usage flow
(README.md:92-103 style) against the embedded runtime: dataset ingest ->
projection -> dataType -> histogram -> builder (lr/rf/gb/nb/mlp) ->
read predictions.

Run:  python examples/titanic_pipeline.py           (CPU or GPU)
"""
import os
import sys
import tempfile

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from learning_orchestra_client import (BuilderSparkMl, Context, DatasetCsv,
                                       ExploreHistogram, Observe,
                                       TransformDataType, TransformProjection)
from learningorchestra_amd.config import Config, set_config
from learningorchestra_amd.data.synthetic import titanic_csv

# embedded single-process mode (point Context("host:port") at a server for
# the networked flavor)
set_config(Config(data_root=tempfile.mkdtemp(prefix="lo_demo_"), mongo_uri=""))
ctx = Context.in_process()

# 1. dataset ingest (file:// URI; reference used public dataset URLs)
path = os.path.join(tempfile.gettempdir(), "titanic_demo.csv")
with open(path, "w") as fh:
    fh.write(titanic_csv(400))
DatasetCsv(ctx).insert_sync("titanic", f"file://{path}")
print("rows:", len(DatasetCsv(ctx).search("titanic", limit=5)))

# 2. transform: project model features, cast types in place
TransformProjection(ctx).create(
    "titanic", "titanic_feat", ["Pclass", "Age", "Fare", "Survived"])
Observe(ctx).wait("titanic_feat")
TransformDataType(ctx).convert("titanic_feat",
                               {"Age": "number", "Fare": "number"})

# 3. explore: per-field value histogram
ExploreHistogram(ctx).create("titanic_feat", "titanic_hist", ["Pclass"])
Observe(ctx).wait("titanic_hist")

# 4. builder: user preprocessing code (the reference's modelingCode
# contract: build features_training/evaluation/testing with a 'label'
# column from training_df/testing_df) + one pipeline per classifier
modeling_code = '''
df = training_df.fillna(0).rename(columns={"Survived": "label"})
features_training = df
features_evaluation = df
features_testing = testing_df.fillna(0).rename(columns={"Survived": "label"})
'''
uris = BuilderSparkMl(ctx).build(
    "titanic_feat", "titanic_feat", modeling_code=modeling_code,
    classifiers=["lr", "nb", "mlp"])
names = [u.split("/")[-1].split("?")[0] for u in uris]   # poll URIs -> names
for n in names:
    doc = Observe(ctx).wait(n)
    print(f"{n}: accuracy={doc.get('accuracy'):.3f} "
          f"f1={doc.get('f1'):.3f} fitTime={doc.get('fitTime'):.2f}s")
print("predictions sample:", BuilderSparkMl(ctx).search(names[0], limit=2))
