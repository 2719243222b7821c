"""Hypothesis property suite for the hardened JobScheduler (r2): random
interleavings of submit / cancel / timeout must always terminate with every
job in a terminal state, consistent stats, and no device slot left owned."""
import threading
import time

import pytest
from hypothesis import HealthCheck, given, settings
from hypothesis import strategies as st

from learningorchestra_amd.executor.scheduler import JobScheduler

ACTIONS = st.lists(
    st.tuples(
        st.sampled_from(["fast", "slow", "fail", "cancel_prev", "timed"]),
        st.sampled_from([None, "cuda:0", "cuda:1", "gpu"]),
    ),
    min_size=1, max_size=12)


@settings(max_examples=15, deadline=None,
          suppress_health_check=[HealthCheck.too_slow])
@given(ACTIONS)
def test_scheduler_always_terminates(actions):
    sched = JobScheduler(max_workers=4,
                         devices=["cuda:0", "cuda:1"])
    jobs = []
    for i, (kind, device) in enumerate(actions):
        name = f"j{i}"
        if kind == "fast":
            jobs.append(sched.submit(name, lambda: 42, device=device))
        elif kind == "slow":
            def slow(cancel_event=None):
                for _ in range(40):
                    if cancel_event is not None and cancel_event.is_set():
                        raise RuntimeError("cancelled")
                    time.sleep(0.005)
                return "slow-done"
            jobs.append(sched.submit(name, slow, device=device))
        elif kind == "fail":
            def boom():
                raise ValueError("boom")
            jobs.append(sched.submit(name, boom, device=device))
        elif kind == "timed":
            def napper(cancel_event=None):
                while not cancel_event.is_set():
                    time.sleep(0.005)
                raise RuntimeError("timed out cooperatively")
            jobs.append(sched.submit(name, napper, device=device,
                                     timeout=0.15))
        elif kind == "cancel_prev" and jobs:
            sched.cancel(jobs[-1].name)
    # every job reaches a terminal state
    deadline = time.time() + 30
    for j in jobs:
        try:
            j.future.result(max(0.1, deadline - time.time()))
        except Exception:
            pass  # errors/cancels are data
    for j in jobs:
        assert j.done(), j.name
        assert j.state in ("ok", "cancelled", "error") \
            or j.state.startswith("timeout"), (j.name, j.state)
    # stats are consistent and no device slot is left owned
    stats = sched.stats()
    assert sum(stats["jobs"].values()) == len(jobs)
    assert stats["running"] == [] and stats["queued"] == []
    for dev, owner in stats["deviceOwners"].items():
        assert owner is None, (dev, owner)
    sched.shutdown()


@settings(max_examples=10, deadline=None)
@given(st.integers(min_value=1, max_value=16))
def test_device_pool_balances(n_jobs):
    sched = JobScheduler(max_workers=8,
                         devices=[f"cuda:{i}" for i in range(4)])
    picked = [sched.pick_device() for _ in range(n_jobs)]
    jobs = [sched.submit(f"p{i}", lambda d=d: time.sleep(0.02), device=d)
            for i, d in enumerate(picked)]
    for j in jobs:
        j.wait(timeout=20)
    # no device gets more than its fair share (+1 rounding)
    from collections import Counter
    counts = Counter(picked)
    assert max(counts.values()) <= (n_jobs + 3) // 4 + 1, counts
    sched.shutdown()
