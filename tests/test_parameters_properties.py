"""Property-based tests for the $/# parameter-resolution protocol
(executor/parameters.py — the reference's model.py:32-64 substitution
surface): arbitrary nested JSON-ish structures must round-trip with ONLY
marked strings substituted, in place, preserving shape."""
from hypothesis import given, settings, strategies as st

from learningorchestra_amd.executor.parameters import (Parameters,
                                                       UserCodeDisabled)


class FakeData:
    """Stands in for the Data facade: name -> rows list."""

    def __init__(self, objects):
        self._objects = objects

    def get_object(self, name):
        if name not in self._objects:
            raise KeyError(name)
        return self._objects[name]


plain_strings = st.text(
    alphabet=st.characters(blacklist_characters="$#", codec="ascii"),
    max_size=12)
scalars = st.one_of(st.integers(-100, 100), st.booleans(), st.none(),
                    plain_strings)
json_vals = st.recursive(
    scalars,
    lambda ch: st.one_of(st.lists(ch, max_size=4),
                         st.dictionaries(plain_strings, ch, max_size=4)),
    max_leaves=20)


@settings(max_examples=200, deadline=None)
@given(st.dictionaries(plain_strings, json_vals, max_size=5))
def test_unmarked_structures_pass_through_unchanged(params):
    p = Parameters(FakeData({}))
    assert p.treat(params) == params


@settings(max_examples=100, deadline=None)
@given(json_vals, st.integers(-1000, 1000))
def test_marked_strings_substitute_at_any_depth(container, payload):
    p = Parameters(FakeData({"ds": payload}))

    def inject(v, depth=0):
        # replace the first plain string we find with a marker
        if isinstance(v, list):
            return [inject(e) for e in v]
        if isinstance(v, dict):
            return {k: inject(e) for k, e in v.items()}
        return v

    treated = p.treat({"a": "$ds", "rest": inject(container)})
    assert treated["a"] == payload
    assert treated["rest"] == container


@settings(max_examples=50, deadline=None)
@given(st.integers(-50, 50), st.integers(-50, 50))
def test_code_marker_evaluates(a, b):
    p = Parameters(FakeData({}), allow_user_code=True)
    assert p.treat_value(f"#{a} + {b}") == a + b


def test_code_marker_disabled_raises():
    p = Parameters(FakeData({}), allow_user_code=False)
    try:
        p.treat_value("#1+1")
        assert False, "should have raised"
    except UserCodeDisabled:
        pass


def test_dataset_attr_access():
    p = Parameters(FakeData({"obj": {"col": [1, 2, 3]}}))
    assert p.treat_value("$obj.col") == [1, 2, 3]
