"""Hypothesis property suite for the artifact store (VERDICT r1 next-round
item 10: extend the property suites to the artifact store).

Mirrors the reference's binary-store contract (/root/reference/microservices/
binary_executor_image/utils.py:195-233 — keras-then-dill save, dill-then-keras
read, /binaries/{service_type}/{filename} path scheme) with the added r2
containment guarantees: names that could escape the data root are rejected
everywhere, and whatever round-trips does so losslessly.
"""
from __future__ import annotations

import os
import string

import pytest
from hypothesis import given, settings
from hypothesis import strategies as st

from learningorchestra_amd.storage.artifacts import ArtifactStore, check_name

VALID_FIRST = string.ascii_letters + string.digits
VALID_REST = VALID_FIRST + "_. -"

valid_names = st.builds(
    lambda first, rest: first + rest,
    st.sampled_from(VALID_FIRST),
    st.text(alphabet=VALID_REST, max_size=40),
).filter(lambda s: ".." not in s)

service_types = st.sampled_from(
    ["train/torch", "model/torch", "predict/sklearn", "function/python"])

payloads = st.recursive(
    st.none() | st.booleans() | st.integers(-2**31, 2**31)
    | st.floats(allow_nan=False, allow_infinity=False, width=32)
    | st.text(max_size=30),
    lambda inner: st.lists(inner, max_size=4)
    | st.dictionaries(st.text(alphabet=string.ascii_lowercase, min_size=1,
                              max_size=8), inner, max_size=4),
    max_leaves=12)


@settings(max_examples=60, deadline=None)
@given(name=valid_names, stype=service_types, obj=payloads)
def test_roundtrip_arbitrary_objects(tmp_path_factory, name, stype, obj):
    store = ArtifactStore(str(tmp_path_factory.mktemp("art")))
    store.save(obj, name, stype)
    assert store.exists(name, stype)
    assert store.load(name, stype) == obj
    store.delete(name, stype)
    assert not store.exists(name, stype)


@settings(max_examples=80, deadline=None)
@given(name=st.text(max_size=40))
def test_invalid_names_never_escape(tmp_path_factory, name):
    """Either the name is accepted by check_name and the final path stays
    under the store root, or every store entry point raises ValueError."""
    root = str(tmp_path_factory.mktemp("art"))
    store = ArtifactStore(root)
    try:
        check_name(name)
        ok = True
    except ValueError:
        ok = False
    if ok:
        p = store.path(name, "train/torch")
        assert os.path.realpath(p).startswith(os.path.realpath(root))
    else:
        with pytest.raises(ValueError):
            store.path(name, "train/torch")
        with pytest.raises(ValueError):
            store.save({"x": 1}, name, "train/torch")
        with pytest.raises(ValueError):
            store.load(name, "train/torch")


@pytest.mark.parametrize("evil", [
    "../escape", "..", "a/../../b", "/etc/passwd", "a/b", ".hidden",
    "", "name\x00null", "a" * 201,
])
def test_known_traversal_names_rejected(tmp_path, evil):
    store = ArtifactStore(str(tmp_path))
    with pytest.raises(ValueError):
        store.path(evil, "train/torch")


@settings(max_examples=30, deadline=None)
@given(name=valid_names, obj=payloads)
def test_overwrite_is_last_writer_wins(tmp_path_factory, name, obj):
    store = ArtifactStore(str(tmp_path_factory.mktemp("art")))
    store.save({"first": True}, name, "model/torch")
    store.save(obj, name, "model/torch")
    assert store.load(name, "model/torch") == obj


def test_spec_model_roundtrips_cpu(tmp_path):
    """Engine models with lo_spec + state_dict go through the portable
    spec.json + state.pt path and rebuild identically on CPU."""
    import torch

    from learningorchestra_amd.models.zoo import MnistCNN

    store = ArtifactStore(str(tmp_path))
    m = MnistCNN(device="cpu", seed=3)
    store.save(m, "mnist-prop", "train/torch")
    m2 = store.load("mnist-prop", "train/torch")
    sd1, sd2 = m.state_dict(), m2.state_dict()
    assert sd1.keys() == sd2.keys()
    for k in sd1:
        assert torch.equal(sd1[k], sd2[k]), k
