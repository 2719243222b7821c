"""Dataset verb: batched CSV ingest keeping the reference's outward contract
(row-documents, _id from 1, sanitized headers, finished flag)."""
import time

from learningorchestra_amd.data.csv_ingest import CsvIngest
from learningorchestra_amd.data.synthetic import titanic_csv
from learningorchestra_amd.executor.scheduler import JobScheduler
from learningorchestra_amd.storage import Data, Metadata


def test_ingest_titanic_shape(db, artifacts):
    md = Metadata(db)
    md.create_file("titanic", "dataset/csv", url="synthetic", fields=[])
    ing = CsvIngest(db)
    n = ing.ingest_text("titanic", titanic_csv(rows=100))
    md.update_finished_flag("titanic", True)
    assert n == 100
    meta = md.get_metadata("titanic")
    # headers sanitized via re.sub('\W+','') like database.py:118
    assert "PassengerId" in meta["fields"] and len(meta["fields"]) == 12
    first = db["titanic"].find_one({"_id": 1})
    assert first["Pclass"] in (1, 2, 3)          # numeric inference
    assert isinstance(first["Name"], str)
    assert first["Cabin"] is None                 # empty string -> None
    df = Data(db, artifacts).get_dataset_content("titanic")
    assert df.shape == (100, 12)


def test_ingest_async_pipeline(db, tmp_path):
    path = tmp_path / "t.csv"
    path.write_text("a,b!b\n1,2\n3,\n")
    md = Metadata(db)
    sched = JobScheduler(md)
    ing = CsvIngest(db)
    ing.run_async("t", str(path), sched)
    deadline = time.time() + 10
    while not md.is_finished("t") and time.time() < deadline:
        time.sleep(0.01)
    meta = md.get_metadata("t")
    assert meta["finished"] is True
    assert meta["fields"] == ["a", "bb"]
    rows = list(db["t"].find({"_id": {"$ne": 0}}).sort("_id", 1))
    assert rows[0]["a"] == 1 and rows[1]["bb"] is None
