"""Embedded Mongo-compatible document store tests (the storage core that every
verb's metadata/lineage/polling contract sits on)."""
import pytest

from learningorchestra_amd.storage.docstore import DocumentStore, DuplicateKeyError


def test_insert_find_roundtrip(db):
    col = db["ds"]
    col.insert_one({"_id": 0, "type": "dataset/csv", "finished": False})
    col.insert_many([{"_id": i, "x": i * 2} for i in range(1, 5)])
    assert col.find_one({"_id": 0})["type"] == "dataset/csv"
    rows = list(col.find({"_id": {"$ne": 0}}).sort("_id", 1))
    assert [r["x"] for r in rows] == [2, 4, 6, 8]


def test_query_operators(db):
    col = db["c"]
    col.insert_many([{"a": i, "tag": "even" if i % 2 == 0 else "odd"} for i in range(10)])
    assert col.count_documents({"a": {"$gte": 5}}) == 5
    assert col.count_documents({"a": {"$in": [1, 3, 99]}}) == 2
    assert col.count_documents({"$or": [{"a": 0}, {"tag": "odd"}]}) == 6
    assert col.count_documents({"missing": {"$exists": False}}) == 10
    assert col.count_documents({"a": {"$not": {"$lt": 8}}}) == 2


def test_sort_skip_limit_projection(db):
    col = db["c"]
    col.insert_many([{"v": 10 - i} for i in range(10)])
    got = list(col.find({}, {"v": 1, "_id": 0}).sort("v", 1).skip(2).limit(3))
    assert got == [{"v": 3}, {"v": 4}, {"v": 5}]


def test_update_and_delete(db):
    col = db["c"]
    col.insert_one({"_id": 0, "finished": False})
    col.update_one({"_id": 0}, {"$set": {"finished": True}})
    assert col.find_one({"_id": 0})["finished"] is True
    col.insert_many([{"x": 1}, {"x": 1}, {"x": 2}])
    assert col.delete_many({"x": 1}).deleted_count == 2


def test_duplicate_id_raises(db):
    col = db["c"]
    col.insert_one({"_id": 7})
    with pytest.raises(DuplicateKeyError):
        col.insert_one({"_id": 7})


def test_aggregate_group_histogram(db):
    # the histogram verb's pipeline: $group {_id:"$field", count:{$sum:1}}
    # (reference histogram.py:31-32)
    col = db["ds"]
    col.insert_many([{"Sex": "male"}] * 3 + [{"Sex": "female"}] * 2)
    out = col.aggregate([{"$group": {"_id": "$Sex", "count": {"$sum": 1}}},
                         {"$sort": {"count": -1}}])
    assert out == [{"_id": "male", "count": 3}, {"_id": "female", "count": 2}]


def test_aggregate_match_avg(db):
    col = db["c"]
    col.insert_many([{"g": "a", "v": 1}, {"g": "a", "v": 3}, {"g": "b", "v": 10}])
    out = col.aggregate([{"$match": {"v": {"$lt": 5}}},
                         {"$group": {"_id": "$g", "avg": {"$avg": "$v"}}}])
    assert out == [{"_id": "a", "avg": 2.0}]


def test_persistence_roundtrip(tmp_path):
    root = str(tmp_path / "dbroot")
    store = DocumentStore(root)
    store["ds"].insert_many([{"_id": i, "x": i} for i in range(3)])
    store.flush()
    store2 = DocumentStore(root)
    assert store2["ds"].count_documents({}) == 3
    assert store2["ds"].find_one({"_id": 2})["x"] == 2


def test_list_collections_and_drop(db):
    db["a"].insert_one({"x": 1})
    db["b"].insert_one({"x": 1})
    assert set(db.list_collection_names()) == {"a", "b"}
    db.drop_collection("a")
    assert db.list_collection_names() == ["b"]
