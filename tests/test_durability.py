"""Crash-durability (WAL) + scheduler-hardening tests (r1 VERDICT missing
#3/#4: the reference's Mongo replica set + Swarm restart_policy roles)."""
import json
import os
import signal
import subprocess
import sys
import textwrap
import time

import pytest

from learningorchestra_amd.storage.docstore import DocumentStore


def test_wal_replay_after_kill9(tmp_path):
    """Writes made after the last flush survive a SIGKILL via WAL replay."""
    root = str(tmp_path / "db")
    prog = textwrap.dedent(f"""
        import os, signal
        from learningorchestra_amd.storage.docstore import DocumentStore
        db = DocumentStore({root!r})
        db["ds"].insert_one({{"_id": 0, "finished": False, "type": "dataset/csv"}})
        db["ds"].insert_many([{{"_id": i, "v": i * i}} for i in range(1, 51)])
        db.flush()
        # post-flush mutations: only the WAL has these
        db["ds"].update_one({{"_id": 0}}, {{"$set": {{"finished": True}}}})
        db["ds"].insert_one({{"_id": 99, "v": "tail"}})
        db["other"].insert_one({{"_id": 1, "x": 1}})
        print("READY", flush=True)
        os.kill(os.getpid(), signal.SIGKILL)  # no atexit, no flush
    """)
    proc = subprocess.run([sys.executable, "-c", prog], capture_output=True,
                          text=True, timeout=120)
    assert "READY" in proc.stdout
    assert proc.returncode == -signal.SIGKILL
    db = DocumentStore(root)
    assert db["ds"].find_one({"_id": 0})["finished"] is True
    assert db["ds"].find_one({"_id": 99})["v"] == "tail"
    assert db["ds"].count_documents({}) == 52
    assert db["other"].find_one({"_id": 1})["x"] == 1
    # replayed state flushes cleanly and the WAL resets
    db.flush()
    assert not os.path.exists(os.path.join(root, "wal.jsonl"))
    db2 = DocumentStore(root)
    assert db2["ds"].count_documents({}) == 52


def test_wal_torn_tail(tmp_path):
    """A partially-written last WAL line stops replay without corruption."""
    root = str(tmp_path / "db")
    db = DocumentStore(root)
    db["c"].insert_one({"_id": 1, "v": "a"})
    db["c"].insert_one({"_id": 2, "v": "b"})
    del db  # keep the WAL (no flush): simulate crash
    wal = os.path.join(root, "wal.jsonl")
    with open(wal, "a") as fh:
        fh.write('{"c": "c", "op": "insert_one", "a": [{"_id": 3')  # torn
    import atexit
    db2 = DocumentStore(root)
    atexit.unregister(db2.flush)
    assert db2["c"].count_documents({}) == 2


def test_wal_drop_and_update_ops(tmp_path):
    root = str(tmp_path / "db")
    db = DocumentStore(root)
    db["a"].insert_one({"_id": 1, "n": 0})
    db["a"].update_many({}, {"$inc": {"n": 5}})
    db["b"].insert_one({"_id": 1})
    db["b"].drop()
    db["a"].delete_one({"_id": 1})
    db["a"].insert_one({"_id": 2, "n": 7})
    import atexit
    atexit.unregister(db.flush)
    db._wal_fh.close()
    db._wal_fh = None  # simulate crash (no flush)
    db2 = DocumentStore(root)
    atexit.unregister(db2.flush)
    assert db2["a"].find_one({"_id": 2})["n"] == 7
    assert db2["a"].find_one({"_id": 1}) is None
    assert db2["b"].estimated_document_count() == 0


# ---------------------------------------------------------------- scheduler --
def test_scheduler_timeout_cancels_job():
    from learningorchestra_amd.executor.scheduler import JobScheduler
    from learningorchestra_amd.storage.docstore import DocumentStore
    from learningorchestra_amd.storage.metadata import Metadata
    db = DocumentStore()
    md = Metadata(db)
    md.create_file("slow", "train/torch")
    sched = JobScheduler(md, max_workers=2)

    def slow(cancel_event=None):
        for _ in range(200):
            if cancel_event.is_set():
                raise RuntimeError("cancelled cooperatively")
            time.sleep(0.05)

    job = sched.submit("slow", slow, timeout=0.5)
    with pytest.raises(Exception):
        job.wait(timeout=30)
    doc = md.get_metadata("slow")
    assert doc["finished"] and "timeout" in str(doc.get("exception"))
    assert job.state in ("timeout after 0.5s", "cancelled", "error")


def test_scheduler_cancel_running_and_queued():
    from learningorchestra_amd.executor.scheduler import JobScheduler
    sched = JobScheduler(max_workers=1)
    started = []

    def body(cancel_event=None):
        started.append(1)
        while not cancel_event.is_set():
            time.sleep(0.02)
        raise RuntimeError("stopped")

    j1 = sched.submit("j1", body)
    for _ in range(100):
        if started:
            break
        time.sleep(0.02)
    j2 = sched.submit("j2", body)  # queued behind j1 (1 worker)
    assert sched.cancel("j2")
    assert sched.cancel("j1")
    with pytest.raises(Exception):
        j1.wait(timeout=30)
    with pytest.raises(Exception):
        j2.wait(timeout=30)
    assert j2.state == "cancelled"
    stats = sched.stats()
    assert stats["jobs"].get("cancelled", 0) + stats["jobs"].get("error", 0) >= 2


def test_scheduler_process_job_kill():
    from learningorchestra_amd.executor.scheduler import JobScheduler
    sched = JobScheduler(max_workers=2)
    job = sched.submit_process(
        "sleeper", [sys.executable, "-c", "import time; time.sleep(600)"],
        timeout=1.0)
    t0 = time.time()
    with pytest.raises(Exception):
        job.wait(timeout=60)
    assert time.time() - t0 < 30
    assert job.proc.poll() is not None  # really dead


def test_scheduler_device_pool_spreads_jobs():
    from learningorchestra_amd.executor.scheduler import JobScheduler
    sched = JobScheduler(max_workers=8,
                         devices=[f"cuda:{i}" for i in range(8)])
    seen = []
    lock = __import__("threading").Lock()

    def trial(dev):
        with lock:
            seen.append(dev)
        time.sleep(0.3)

    jobs = []
    for i in range(8):
        dev = sched.pick_device()
        jobs.append(sched.submit(f"trial{i}", trial, dev, device=dev))
    for j in jobs:
        j.wait(timeout=30)
    # 8 concurrent trials across 8 simulated devices: every device used once
    assert sorted(seen) == [f"cuda:{i}" for i in range(8)]


def test_scheduler_process_job_success_output():
    from learningorchestra_amd.executor.scheduler import JobScheduler
    sched = JobScheduler(max_workers=2)
    outs = []
    job = sched.submit_process(
        "hello", [sys.executable, "-c", "print('out42')"],
        on_done=lambda rc, tail: outs.append((rc, tail)))
    job.wait(timeout=60)
    assert outs and outs[0][0] == 0 and "out42" in outs[0][1]


def test_scheduler_multi_device_job_excludes_single_device_jobs():
    """An N-rank train claims EVERY device it spans (scheduler `devices=`):
    a single-device job on any spanned GPU must wait until the multi-device
    job finishes — two processes time-slicing one GPU corrupts long-running
    kernels under mid-kernel preemption (parallel/ddp.py device_step_lock)."""
    import sys
    import threading

    from learningorchestra_amd.executor.scheduler import JobScheduler
    sched = JobScheduler(max_workers=4,
                         devices=["cuda:0", "cuda:1"])
    order = []
    olock = threading.Lock()

    big = sched.submit_process(
        "ddp-train", [sys.executable, "-c", "import time; time.sleep(1.0)"],
        devices=["cuda:0", "cuda:1"],
        on_done=lambda rc, tail: order.append("big-done"))
    time.sleep(0.3)   # let it start and take both slots

    def small():
        with olock:
            order.append("small-ran")

    j = sched.submit("single", small, device="cuda:1")
    time.sleep(0.2)
    # the single-device job must still be blocked on the claimed slot
    assert "small-ran" not in order
    big.wait(timeout=30)
    j.wait(timeout=30)
    assert order.index("big-done") < order.index("small-ran")


def test_wal_autoflush_bounds_wal(tmp_path, monkeypatch):
    """A long-running server must checkpoint periodically: after the
    LO_WAL_AUTOFLUSH-th logged mutation the store snapshots and truncates
    the WAL, and the data survives a reopen."""
    import os

    monkeypatch.setenv("LO_WAL_AUTOFLUSH", "50")
    from learningorchestra_amd.storage.docstore import DocumentStore

    store = DocumentStore(str(tmp_path))
    col = store["events"]
    for i in range(120):
        col.insert_one({"i": i})
    wal = os.path.join(str(tmp_path), "wal.jsonl")
    # 120 ops with threshold 50: two auto-flushes happened; the WAL holds
    # only the tail since the last checkpoint (< threshold lines)
    n_lines = sum(1 for _ in open(wal)) if os.path.exists(wal) else 0
    assert n_lines < 50, n_lines
    store.flush()
    reopened = DocumentStore(str(tmp_path))
    assert reopened["events"].count_documents({}) == 120


def test_docstore_concurrent_writers_and_flushes(tmp_path):
    """8 threads hammer inserts/updates across 4 collections while another
    thread flushes continuously: no exceptions, no lost writes, and a
    reopen (snapshot + WAL replay) sees every document."""
    import threading

    from learningorchestra_amd.storage.docstore import DocumentStore

    store = DocumentStore(str(tmp_path))
    errors = []
    N_PER = 200

    def writer(t):
        try:
            col = store[f"c{t % 4}"]
            for i in range(N_PER):
                col.insert_one({"_id": f"w{t}_{i}", "v": i})
                if i % 50 == 0:
                    col.update_one({"_id": f"w{t}_{i}"},
                                   {"$set": {"touched": True}})
        except Exception as exc:  # noqa: BLE001
            errors.append(repr(exc))

    stop = threading.Event()

    def flusher():
        while not stop.is_set():
            try:
                store.flush()
            except Exception as exc:  # noqa: BLE001
                errors.append(repr(exc))
                return

    threads = [threading.Thread(target=writer, args=(t,)) for t in range(8)]
    fl = threading.Thread(target=flusher)
    fl.start()
    for t in threads:
        t.start()
    for t in threads:
        t.join(timeout=120)
    stop.set()
    fl.join(timeout=30)
    assert not errors, errors[:3]
    store.flush()
    reopened = DocumentStore(str(tmp_path))
    total = sum(reopened[f"c{k}"].count_documents({}) for k in range(4))
    assert total == 8 * N_PER, total
