"""Property-based tests (hypothesis) for the embedded document store.

The store is the framework's Mongo-compatible substrate; these
tests check its query/update semantics against an independent pure-Python
model over randomly generated documents and filters, far beyond the
hand-written cases in test_docstore.py.
"""
from hypothesis import given, settings, strategies as st

from learningorchestra_amd.storage.docstore import DocumentStore, match

# -- document / filter generators -------------------------------------------

field_names = st.sampled_from(["a", "b", "c", "nested"])
scalars = st.one_of(
    st.integers(min_value=-50, max_value=50),
    st.floats(min_value=-10, max_value=10, allow_nan=False, width=32),
    st.sampled_from(["x", "y", "z", ""]),
    st.booleans(),
    st.none(),
)


@st.composite
def documents(draw):
    doc = {k: draw(scalars) for k in draw(st.sets(field_names, max_size=3))}
    if "nested" in doc:
        doc["nested"] = {"inner": draw(scalars)}
    return doc


@st.composite
def filters(draw):
    field = draw(st.sampled_from(["a", "b", "c", "nested.inner"]))
    kind = draw(st.sampled_from(["eq", "cmp", "in", "exists", "or"]))
    if kind == "eq":
        return {field: draw(scalars)}
    if kind == "cmp":
        op = draw(st.sampled_from(["$gt", "$gte", "$lt", "$lte", "$ne"]))
        return {field: {op: draw(st.one_of(st.integers(-50, 50),
                                           st.sampled_from(["x", "m"])))}}
    if kind == "in":
        op = draw(st.sampled_from(["$in", "$nin"]))
        return {field: {op: draw(st.lists(scalars, max_size=3))}}
    if kind == "exists":
        return {field: {"$exists": draw(st.booleans())}}
    return {"$or": [{field: draw(scalars)},
                    {draw(st.sampled_from(["a", "b"])): draw(scalars)}]}


# -- independent model of the matcher ---------------------------------------

def _get(doc, path):
    cur, ok = doc, True
    for part in path.split("."):
        if isinstance(cur, dict) and part in cur:
            cur = cur[part]
        else:
            return None, False
    return cur, ok


def _cmp_model(a, b, op):
    try:
        if op == "$gt":
            return a > b
        if op == "$gte":
            return a >= b
        if op == "$lt":
            return a < b
        if op == "$lte":
            return a <= b
    except TypeError:
        return False
    return False


def model_match(doc, flt):
    """Mongo semantics: a missing field behaves as null for every value
    operator ($ne/$in/comparisons/equality); only $exists sees presence."""
    for key, cond in flt.items():
        if key == "$or":
            if not any(model_match(doc, c) for c in cond):
                return False
            continue
        val, present = _get(doc, key)
        if isinstance(cond, dict) and any(k.startswith("$") for k in cond):
            for op, ref in cond.items():
                if op == "$exists":
                    if present != bool(ref):
                        return False
                elif op == "$ne":
                    if val == ref:
                        return False
                elif op == "$in":
                    if val not in ref:
                        return False
                elif op == "$nin":
                    if val in ref:
                        return False
                else:
                    if val is None or not _cmp_model(val, ref, op):
                        return False
        else:
            if val != cond:
                return False
    return True


# -- properties --------------------------------------------------------------

@settings(max_examples=300, deadline=None)
@given(st.lists(documents(), max_size=12), filters())
def test_find_matches_model(docs, flt):
    store = DocumentStore()
    col = store["col"]
    for d in docs:
        col.insert_one(dict(d))
    got = sorted(d["_id"] for d in col.find(flt))
    want = sorted(d["_id"] for d in col.find()
                  if model_match({k: v for k, v in d.items() if k != "_id"}, flt))
    assert got == want


@settings(max_examples=200, deadline=None)
@given(st.lists(documents(), max_size=10), filters())
def test_count_delete_consistency(docs, flt):
    store = DocumentStore()
    col = store["col"]
    for d in docs:
        col.insert_one(dict(d))
    n_match = col.count_documents(flt)
    assert n_match == len(list(col.find(flt)))
    res = col.delete_many(flt)
    assert res.deleted_count == n_match
    assert col.count_documents(flt) == 0
    assert col.count_documents({}) == len(docs) - n_match


@settings(max_examples=200, deadline=None)
@given(st.lists(documents(), min_size=1, max_size=10))
def test_sort_skip_limit(docs):
    store = DocumentStore()
    col = store["col"]
    for d in docs:
        col.insert_one(dict(d))
    ids = [d["_id"] for d in col.find().sort("_id", -1)]
    assert ids == sorted(ids, reverse=True)
    page = [d["_id"] for d in col.find().sort("_id", 1).skip(1).limit(2)]
    assert page == sorted(ids)[1:3]


@settings(max_examples=150, deadline=None)
@given(st.lists(documents(), min_size=1, max_size=8),
       st.integers(-20, 20))
def test_update_set_inc_roundtrip(docs, delta):
    store = DocumentStore()
    col = store["col"]
    for d in docs:
        col.insert_one(dict(d))
    col.update_many({}, {"$set": {"tag": "t"}, "$inc": {"n": delta}})
    for d in col.find():
        assert d["tag"] == "t"
        assert d["n"] == delta
    col.update_many({}, {"$inc": {"n": delta}})
    for d in col.find():
        assert d["n"] == 2 * delta


@settings(max_examples=150, deadline=None)
@given(st.lists(documents(), max_size=8), filters())
def test_match_function_agrees_with_model(docs, flt):
    for d in docs:
        assert match(d, flt) == model_match(d, flt), (d, flt)


@settings(max_examples=150, deadline=None)
@given(st.lists(st.tuples(st.sampled_from(["a", "b", "c"]),
                          st.integers(-20, 20)), max_size=20))
def test_group_aggregation_matches_model(pairs):
    """$group {_id: "$k", count: {$sum: 1}, total: {$sum: "$v"}} — the
    histogram verb's aggregation — vs a dict model."""
    store = DocumentStore()
    col = store["col"]
    for k, v in pairs:
        col.insert_one({"k": k, "v": v})
    got = {d["_id"]: d for d in col.aggregate([
        {"$group": {"_id": "$k", "count": {"$sum": 1},
                    "total": {"$sum": "$v"}}}])}
    want = {}
    for k, v in pairs:
        e = want.setdefault(k, {"count": 0, "total": 0})
        e["count"] += 1
        e["total"] += v
    assert set(got) == set(want)
    for k, e in want.items():
        assert got[k]["count"] == e["count"]
        assert got[k]["total"] == e["total"]
