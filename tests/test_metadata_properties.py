"""Property-based tests for the metadata/lineage layer (storage/metadata.py):
execution documents get monotonically increasing _ids starting after the
metadata doc, parentName chains resolve, and lineage walks terminate even on
adversarial (cyclic) parent graphs."""
from hypothesis import given, settings, strategies as st

from learningorchestra_amd.storage.docstore import DocumentStore
from learningorchestra_amd.storage.metadata import Metadata

names = st.text(alphabet=st.characters(codec="ascii", min_codepoint=97,
                                       max_codepoint=122),
                min_size=1, max_size=8)


@settings(max_examples=100, deadline=None)
@given(st.integers(1, 8))
def test_execution_documents_monotonic_ids(n_exec):
    md = Metadata(DocumentStore())
    md.create_file("a", "train/torch")
    ids = []
    for i in range(n_exec):
        ids.append(md.create_execution_document("a", f"run {i}", {"i": i}))
    assert ids == sorted(ids)
    assert len(set(ids)) == len(ids)
    assert min(ids) >= 1  # metadata doc owns _id 0


@settings(max_examples=100, deadline=None)
@given(st.lists(names, min_size=2, max_size=6, unique=True))
def test_lineage_chain_resolves(chain):
    md = Metadata(DocumentStore())
    md.create_file(chain[0], "model/torch")
    for parent, child in zip(chain, chain[1:]):
        md.create_file(child, "train/torch", parentName=parent)
    lin = md.lineage(chain[-1])
    got = [d.get("datasetName") or d.get("name") for d in lin]
    # the walk must visit every ancestor exactly once, child-to-root or
    # root-to-child (either order is fine as long as the set matches)
    assert set(filter(None, got)) == set(chain)


def test_lineage_terminates_on_cycle():
    md = Metadata(DocumentStore())
    md.create_file("x", "train/torch", parentName="y")
    md.create_file("y", "train/torch", parentName="x")
    lin = md.lineage("x")           # must not loop forever
    assert len(lin) <= 64


def test_walk_to_model_finds_root_module():
    md = Metadata(DocumentStore())
    md.create_file("m", "model/torch", modulePath="mod.path", className="K")
    md.create_file("t", "train/torch", parentName="m")
    md.create_file("p", "predict/torch", parentName="t")
    root = md.walk_to_model("p")
    assert root is not None and root.get("modulePath") == "mod.path"


@settings(max_examples=50, deadline=None)
@given(st.dictionaries(names, st.one_of(st.integers(), st.text(max_size=6),
                                        st.lists(st.floats(-5, 5),
                                                 max_size=4)),
                       max_size=5))
def test_artifact_store_roundtrip(payload):
    """ArtifactStore must round-trip arbitrary picklable objects under the
    reference's /binaries/{type}/{name} layout."""
    import tempfile
    from learningorchestra_amd.storage.artifacts import ArtifactStore
    store = ArtifactStore(tempfile.mkdtemp(prefix="lo_art_"))
    store.save(payload, "obj", "train/torch")
    assert store.load("obj", "train/torch") == payload
    assert store.exists("obj", "train/torch")
    store.delete("obj", "train/torch")
    assert not store.exists("obj", "train/torch")
