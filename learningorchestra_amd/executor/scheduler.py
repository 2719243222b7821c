"""Async job scheduler — the finished-flag contract.

The reference's pattern (every ``create()``: write metadata with
``finished:false``, ``ThreadPoolExecutor().submit`` the pipeline, return 201
with a poll URI — e.g. /root/reference/microservices/binary_executor_image/
binary_execution.py:118-134) is kept, but as ONE real job queue instead of a
throwaway ThreadPool per request: a single process owns the GPUs, so jobs that
need a device are serialized per device while CPU jobs run concurrently
(SURVEY §7 hard-part 5).

Exceptions are data, not crashes: a failing job records its traceback into the
artifact's metadata/execution document (binary_execution.py:163-170) and flips
``finished`` with the exception set.
"""
from __future__ import annotations

import threading
import time
import traceback
from concurrent.futures import Future, ThreadPoolExecutor
from typing import Any, Callable, Dict, Optional


class Job:
    def __init__(self, name: str, future: Future):
        self.name = name
        self.future = future
        self.submitted_at = time.time()

    def done(self) -> bool:
        return self.future.done()

    def wait(self, timeout: Optional[float] = None) -> Any:
        return self.future.result(timeout)


class JobScheduler:
    """Submit pipelines; GPU-tagged jobs serialize on a per-device lock."""

    def __init__(self, metadata=None, max_workers: int = 8):
        self._metadata = metadata
        self._pool = ThreadPoolExecutor(max_workers=max_workers,
                                        thread_name_prefix="lo-job")
        self._device_locks: Dict[str, threading.Lock] = {}
        self._jobs: Dict[str, Job] = {}
        self._lock = threading.Lock()

    def _device_lock(self, device: str) -> threading.Lock:
        with self._lock:
            if device not in self._device_locks:
                self._device_locks[device] = threading.Lock()
            return self._device_locks[device]

    def submit(self, name: str, fn: Callable[..., Any], *args,
               device: Optional[str] = None,
               on_error: Optional[Callable[[BaseException], None]] = None,
               **kwargs) -> Job:
        """Run ``fn`` asynchronously. If ``device`` is given ('cuda:0', ...),
        the job holds that device's lock for its duration. On exception the
        metadata finished-flag is set with the exception recorded."""

        def run():
            try:
                if device is not None:
                    with self._device_lock(device):
                        return fn(*args, **kwargs)
                return fn(*args, **kwargs)
            except BaseException as exc:  # noqa: BLE001 - errors are data
                tb = traceback.format_exc()
                if self._metadata is not None:
                    try:
                        self._metadata.update_finished_flag(
                            name, finished=True, exception=f"{exc!r}")
                        self._metadata.update_fields(name, traceback=tb)
                    except Exception:
                        pass
                if on_error is not None:
                    on_error(exc)
                raise

        job = Job(name, self._pool.submit(run))
        with self._lock:
            self._jobs[name] = job
        return job

    def get(self, name: str) -> Optional[Job]:
        with self._lock:
            return self._jobs.get(name)

    def wait_all(self, timeout: Optional[float] = None) -> None:
        with self._lock:
            jobs = list(self._jobs.values())
        deadline = None if timeout is None else time.time() + timeout
        for j in jobs:
            remaining = None if deadline is None else max(0.0, deadline - time.time())
            try:
                j.future.result(remaining)
            except Exception:
                pass  # recorded in metadata by run()

    def shutdown(self) -> None:
        self._pool.shutdown(wait=True)
