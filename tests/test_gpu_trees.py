"""GPU tree tests: tree_hist HIP kernel vs the torch reference, and GBT
end-to-end on MI355X."""
import pytest
import torch

pytestmark = pytest.mark.gpu


def test_tree_hist_kernel_vs_reference():
    from learningorchestra_amd.models.trees import build_histograms
    torch.manual_seed(0)
    N, F, nodes = 200_000, 12, 8
    binned = torch.randint(0, 255, (N, F), dtype=torch.uint8)
    node_of = torch.randint(-1, nodes, (N,), dtype=torch.int32)
    grad = torch.randn(N)
    hess = torch.rand(N)
    ref = build_histograms(binned, node_of, grad, hess, nodes)
    got = build_histograms(binned.cuda(), node_of.cuda(), grad.cuda(),
                           hess.cuda(), nodes).cpu()
    rel = ((got - ref).norm() / (ref.norm() + 1e-8)).item()
    assert rel < 1e-4, rel


def test_tree_hist_many_nodes_global_path():
    from learningorchestra_amd.models.trees import build_histograms
    torch.manual_seed(1)
    N, F, nodes = 50_000, 4, 256   # too many nodes for LDS -> global atomics
    binned = torch.randint(0, 255, (N, F), dtype=torch.uint8)
    node_of = torch.randint(0, nodes, (N,), dtype=torch.int32)
    grad = torch.randn(N)
    hess = torch.rand(N)
    ref = build_histograms(binned, node_of, grad, hess, nodes)
    got = build_histograms(binned.cuda(), node_of.cuda(), grad.cuda(),
                           hess.cuda(), nodes).cpu()
    rel = ((got - ref).norm() / (ref.norm() + 1e-8)).item()
    assert rel < 1e-4, rel


def test_gbt_gpu_end_to_end():
    from learningorchestra_amd.data.synthetic import tabular
    from learningorchestra_amd.models.trees import GBTClassifier
    X, y = tabular(200_000, 16, seed=3)
    clf = GBTClassifier(n_trees=20, max_depth=5, device="cuda")
    clf.fit(X.numpy(), y.numpy())
    acc = (clf.predict(X[:20000].numpy()).astype(int)
           == y[:20000].numpy().astype(int)).mean()
    assert acc > 0.8, acc


def test_builder_native_classifiers_gpu(tmp_path):
    """builder/sparkml with the native GPU tabular engines end-to-end via the
    in-process client (LR on the engine + GBT histogram trees on cuda)."""
    from learning_orchestra_client import BuilderSparkMl, Context, DatasetCsv
    from learningorchestra_amd.api.app import Runtime
    from learningorchestra_amd.config import Config
    from learningorchestra_amd.data.synthetic import titanic_csv
    cfg = Config(data_root=str(tmp_path), mongo_uri="")
    ctx = Context.in_process(Runtime(cfg))
    p = tmp_path / "t.csv"
    p.write_text(titanic_csv(rows=400))
    DatasetCsv(ctx).insert_sync("titanic", str(p))
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
    b = BuilderSparkMl(ctx)
    b.build("titanic", "titanic", code, ["lr", "gb"])
    for c in ("lr", "gb"):
        doc = b.wait(f"titanic{c}", timeout=240)
        assert doc.get("exception") in (None, ""), doc
        assert doc["accuracy"] > 0.55, (c, doc["accuracy"])


@pytest.mark.gpu
def test_class_histograms_gpu_vs_cpu():
    """Multiclass per-class count histograms: GPU tree_hist packing (2
    one-hot classes per kernel call) vs the CPU index_add reference."""
    import torch

    from learningorchestra_amd.models.trees import build_class_histograms
    torch.manual_seed(0)
    N, F, K = 20000, 6, 5
    binned = torch.randint(0, 256, (N, F), dtype=torch.uint8)
    node_of = torch.randint(-1, 4, (N,), dtype=torch.int32)
    onehot = torch.nn.functional.one_hot(
        torch.randint(0, K, (N,)), K).float()
    ref = build_class_histograms(binned, node_of, onehot, 4)
    got = build_class_histograms(binned.cuda(), node_of.cuda(),
                                 onehot.cuda(), 4).cpu()
    assert torch.allclose(ref, got, atol=1e-3), (ref - got).abs().max()


@pytest.mark.gpu
def test_multiclass_rf_gpu():
    from learningorchestra_amd.data.synthetic import tabular_multiclass
    from learningorchestra_amd.models.trees import RandomForestClassifier
    X, y = tabular_multiclass(8000, 16, n_classes=10, seed=3)
    clf = RandomForestClassifier(n_trees=8, max_depth=7,
                                 device="cuda").fit(X.numpy(), y.numpy())
    acc = (clf.predict(X.numpy()).astype(int) == y.numpy()).mean()
    assert clf.n_classes == 10 and acc > 0.8, acc
