"""Dataset verb: CSV ingest.

The reference streams a CSV from a URL through a 3-stage thread pipeline
(download -> treat -> save) with bounded queues and an ``insert_one`` PER ROW
(/root/reference/microservices/database_api_image/database.py:99-151) — its
biggest ingest bottleneck. The rebuild keeps the same outward
contract (collection of row-documents, ``_id`` = row number starting at 1,
header names sanitized with ``re.sub('\\W+','')``, metadata doc updated with
``fields`` then ``finished``) but ingests in batches with ``insert_many``,
with numeric type inference (the reference stores raw strings and needs the
dataType transform afterwards; we store parsed numbers AND keep the dataType
verb for parity).
"""
from __future__ import annotations

import csv
import io
import re
from typing import Iterable, List, Optional

from ..storage.metadata import Metadata

BATCH = 4096


def _sanitize_header(field: str) -> str:
    # database.py:118 does re.sub('\W+','') on each header
    return re.sub(r"\W+", "", field)


def _parse_value(v: str):
    if v == "":
        return None
    try:
        f = float(v)
        if f.is_integer() and "." not in v and "e" not in v.lower():
            return int(v)
        return f
    except ValueError:
        return v


class CsvIngest:
    def __init__(self, database):
        self._db = database
        self._metadata = Metadata(database)

    def ingest_rows(self, name: str, lines: Iterable[str],
                    parse_numbers: bool = True) -> int:
        """Core ingest from an iterable of CSV lines. Returns row count."""
        reader = csv.reader(lines)
        headers: Optional[List[str]] = None
        batch = []
        rowcount = 0
        col = self._db[name]
        for row in reader:
            if headers is None:
                headers = [_sanitize_header(h) for h in row]
                # a user column literally named "_id" would collide with the
                # row-document key (fuzz-found: DuplicateKeyError mid-batch)
                headers = [h if h != "_id" else "_id_" for h in headers]
                self._metadata.update_file_headers(name, headers)
                continue
            rowcount += 1
            doc = {"_id": rowcount}
            for h, v in zip(headers, row):
                doc[h] = _parse_value(v) if parse_numbers else (v if v != "" else None)
            batch.append(doc)
            if len(batch) >= BATCH:
                col.insert_many(batch)
                batch = []
        if batch:
            col.insert_many(batch)
        return rowcount

    def ingest_url(self, name: str, url: str) -> int:
        import requests
        with requests.get(url, stream=True, timeout=60) as resp:
            resp.raise_for_status()
            lines = (ln.decode("utf-8", errors="replace")
                     for ln in resp.iter_lines() if ln is not None)
            return self.ingest_rows(name, lines)

    def ingest_path(self, name: str, path: str) -> int:
        with open(path, "r", encoding="utf-8", newline="") as fh:
            return self.ingest_rows(name, fh)

    def ingest_text(self, name: str, text: str) -> int:
        return self.ingest_rows(name, io.StringIO(text))

    def run_async(self, name: str, source: str, scheduler,
                  service_type: str = "dataset/csv") -> None:
        """The POST /dataset/csv pipeline: metadata first (finished:false),
        async ingest, flip finished (database.py:99-105 + utils.py:72-77)."""
        self._metadata.create_file(name, service_type, url=source, fields=[])

        def pipeline():
            if re.match(r"^https?://", source):
                self.ingest_url(name, source)
            else:
                # accept file:// URIs and plain paths alike
                self.ingest_path(name, re.sub(r"^file://", "", source))
            self._metadata.update_finished_flag(name, True)

        scheduler.submit(name, pipeline)
