"""Async job scheduler — the finished-flag contract, hardened.

The reference's pattern (every ``create()``: write metadata with
``finished:false``, ``ThreadPoolExecutor().submit`` the pipeline, return 201
with a poll URI — e.g. /root/reference/microservices/binary_executor_image/
binary_execution.py:118-134) is kept, but as ONE real job queue instead of a
throwaway ThreadPool per request: a single process owns the GPUs, so jobs that
need a device are serialized per device while CPU jobs run concurrently
(SURVEY §7 hard-part 5).

Hardening beyond the reference (its Swarm ``restart_policy: on-failure``,
docker-compose.yml:3-6, restarted a whole container; r1 VERDICT items 4/7):

* per-job **timeout** — a watchdog marks the job failed in metadata and
  cancels it (cooperatively for thread jobs; by killing the process group for
  process jobs);
* **cancel(name)** — not-yet-started jobs are dropped, running thread jobs get
  their ``cancel_event`` set (pipelines may poll it), running process jobs are
  killed for real;
* **process jobs** (``submit_process``) — multi-rank GPU training runs as a
  torchrun process tree in its own session, so cancellation and timeout are
  real (SIGTERM → SIGKILL on the process group), matching how the reference's
  Spark fan-out could be killed by Swarm;
* **device pool** — ``device="gpu"`` placement round-robins over the
  least-loaded visible device (GridSearch trials spread across all 8 GPUs);
* **introspection** — ``stats()`` feeds the ``/metrics`` endpoint.

Exceptions are data, not crashes: a failing job records its traceback into the
artifact's metadata/execution document (binary_execution.py:163-170) and flips
``finished`` with the exception set.
"""
from __future__ import annotations

import inspect
import os
import signal
import subprocess
import threading
import time
import traceback
from concurrent.futures import Future, ThreadPoolExecutor
from typing import Any, Callable, Dict, List, Optional


class _DeviceSlot:
    """Revocable device ownership (a plain Lock can never be stolen from a
    wedged thread; process jobs that get killed must free their device)."""

    def __init__(self) -> None:
        self.cond = threading.Condition()
        self.owner: Optional[str] = None

    def acquire(self, name: str, cancel_event: threading.Event) -> bool:
        with self.cond:
            while self.owner is not None:
                if cancel_event.is_set():
                    return False
                self.cond.wait(0.1)
            self.owner = name
            return True

    def release(self, name: str) -> None:
        with self.cond:
            if self.owner == name:
                self.owner = None
                self.cond.notify_all()

    def revoke(self) -> None:
        with self.cond:
            self.owner = None
            self.cond.notify_all()


class Job:
    def __init__(self, name: str, device: Optional[str] = None,
                 timeout: Optional[float] = None,
                 devices: Optional[List[str]] = None):
        self.name = name
        self.device = device
        self.devices = devices  # multi-GPU jobs own every device they span
        self.timeout = timeout
        self.future: Optional[Future] = None
        self.cancel_event = threading.Event()
        self.proc: Optional[subprocess.Popen] = None
        self.submitted_at = time.time()
        self.started_at: Optional[float] = None
        self.finished_at: Optional[float] = None
        self.outcome: Optional[str] = None  # ok | error | cancelled | timeout

    def done(self) -> bool:
        return self.future is not None and self.future.done()

    @property
    def state(self) -> str:
        if self.outcome is not None and self.done():
            return self.outcome
        if self.started_at is None:
            return "queued"
        if not self.done():
            return "running"
        return self.outcome or "ok"

    def wait(self, timeout: Optional[float] = None) -> Any:
        return self.future.result(timeout)


class JobScheduler:
    """Submit pipelines; GPU-tagged jobs serialize on a per-device slot."""

    def __init__(self, metadata=None, max_workers: int = 8,
                 devices: Optional[List[str]] = None):
        self._metadata = metadata
        self._pool = ThreadPoolExecutor(max_workers=max_workers,
                                        thread_name_prefix="lo-job")
        self._device_slots: Dict[str, _DeviceSlot] = {}
        self._jobs: Dict[str, Job] = {}
        self._lock = threading.Lock()
        self._devices = devices
        self._rr = 0
        self._watchdog: Optional[threading.Thread] = None

    # -- device pool ---------------------------------------------------------
    def gpu_devices(self) -> List[str]:
        if self._devices is None:
            try:
                import torch
                n = torch.cuda.device_count() if torch.cuda.is_available() else 0
            except Exception:
                n = 0
            self._devices = [f"cuda:{i}" for i in range(n)]
        return self._devices

    def pick_device(self) -> Optional[str]:
        """Least-loaded visible GPU (round-robin on ties) — GridSearch trials
        and independent train jobs spread across the node's 8 GPUs."""
        devs = self.gpu_devices()
        if not devs:
            return None
        with self._lock:
            loads = {d: 0 for d in devs}
            for j in self._jobs.values():
                if j.device in loads and not j.done():
                    loads[j.device] += 1
            self._rr += 1
            rr = self._rr
            return min(devs, key=lambda d: (loads[d],
                                            (devs.index(d) + rr) % len(devs)))

    def job(self, name: str) -> Optional[Job]:
        """Look up a submitted job by name (None if unknown/pruned)."""
        with self._lock:
            return self._jobs.get(name)

    def _device_slot(self, device: str) -> _DeviceSlot:
        with self._lock:
            if device not in self._device_slots:
                self._device_slots[device] = _DeviceSlot()
            return self._device_slots[device]

    # -- submission ----------------------------------------------------------
    def submit(self, name: str, fn: Callable[..., Any], *args,
               device: Optional[str] = None,
               timeout: Optional[float] = None,
               on_error: Optional[Callable[[BaseException], None]] = None,
               **kwargs) -> Job:
        """Run ``fn`` asynchronously. ``device``: 'cuda:N' serializes on that
        device's slot; 'gpu' picks the least-loaded device (pool). If ``fn``
        accepts a ``cancel_event`` kwarg it receives the job's cancel token.
        On exception the metadata finished-flag is set with the exception
        recorded."""
        if device == "gpu":
            device = self.pick_device()
        job = Job(name, device=device, timeout=timeout)
        return self._submit(job, fn, args, kwargs, on_error)

    def _submit(self, job: Job, fn: Callable[..., Any], args, kwargs,
                on_error: Optional[Callable[[BaseException], None]] = None
                ) -> Job:
        name, device = job.name, job.device
        try:
            if "cancel_event" in inspect.signature(fn).parameters:
                kwargs = dict(kwargs, cancel_event=job.cancel_event)
        except (TypeError, ValueError):
            pass

        def run():
            job.started_at = time.time()
            # a job may claim several devices (an N-rank torchrun train owns
            # every GPU it spans): acquire in sorted order so two multi-
            # device jobs can never deadlock. Cross-process GPU sharing is
            # not allowed — two processes time-slicing one device corrupts
            # long-running kernels under mid-kernel preemption (measured,
            # parallel/ddp.py device_step_lock).
            devices = sorted(getattr(job, "devices", None)
                             or ([device] if device is not None else []))
            slots = [self._device_slot(d) for d in devices]
            held: List[_DeviceSlot] = []
            try:
                if job.cancel_event.is_set():
                    job.outcome = job.outcome or "cancelled"
                    raise JobCancelled(name)
                for slot in slots:
                    if not slot.acquire(name, job.cancel_event):
                        job.outcome = job.outcome or "cancelled"
                        raise JobCancelled(name)
                    held.append(slot)
                result = fn(*args, **kwargs)
                job.outcome = job.outcome or "ok"
                return result
            except BaseException as exc:  # noqa: BLE001 - errors are data
                if job.outcome is None:
                    job.outcome = "error"
                tb = traceback.format_exc()
                # a cancel/timeout reason set by cancel() is authoritative —
                # the cooperative exception the pipeline raised in response
                # must not overwrite it
                reason = (job.outcome if job.outcome not in ("error", "ok")
                          else f"{exc!r}")
                if self._metadata is not None:
                    try:
                        self._metadata.update_finished_flag(
                            name, finished=True, exception=reason)
                        self._metadata.update_fields(name, traceback=tb)
                    except Exception:
                        pass
                if on_error is not None:
                    on_error(exc)
                raise
            finally:
                job.finished_at = time.time()
                for slot in held:
                    slot.release(name)

        with self._lock:
            self._jobs[name] = job
            # long-running servers: prune the oldest finished jobs so the
            # registry cannot grow without bound
            if len(self._jobs) > 4096:
                done = [n for n, j in self._jobs.items() if j.done()]
                for n in done[: len(done) // 2]:
                    self._jobs.pop(n, None)
        job.future = self._pool.submit(run)
        if job.timeout is not None:
            self._ensure_watchdog()
        return job

    def submit_process(self, name: str, cmd: List[str],
                       env: Optional[Dict[str, str]] = None,
                       timeout: Optional[float] = None,
                       device: Optional[str] = None,
                       devices: Optional[List[str]] = None,
                       cwd: Optional[str] = None,
                       on_done: Optional[Callable[[int, str], None]] = None
                       ) -> Job:
        """Run ``cmd`` as a subprocess in its own session (killable process
        group) — the multi-rank torchrun jobs go through here. ``on_done``
        gets (returncode, combined-output-tail); non-zero exit raises so the
        standard exception-to-metadata path records it."""
        job = Job(name, device=device, timeout=timeout, devices=devices)

        def run_proc():
            full_env = dict(os.environ)
            if env:
                full_env.update(env)
            proc = subprocess.Popen(
                cmd, stdout=subprocess.PIPE, stderr=subprocess.STDOUT,
                start_new_session=True, env=full_env, cwd=cwd, text=True)
            job.proc = proc
            out_chunks: List[str] = []

            def pump():
                for line in proc.stdout:
                    out_chunks.append(line)
                    if len(out_chunks) > 400:
                        del out_chunks[:200]
            t = threading.Thread(target=pump, daemon=True)
            t.start()
            while proc.poll() is None:
                if job.cancel_event.is_set():
                    _kill_group(proc)
                time.sleep(0.2)
            t.join(timeout=5)
            tail = "".join(out_chunks[-200:])
            rc = proc.returncode
            if job.cancel_event.is_set() and rc != 0:
                job.outcome = job.outcome or "cancelled"
                raise JobCancelled(f"{name} (exit {rc})")
            if on_done is not None:
                on_done(rc, tail)
            if rc != 0:
                raise RuntimeError(
                    f"process job '{name}' exited {rc}; output tail:\n{tail}")
            return tail

        # the SAME Job object carries proc + cancel_event (cancel()/watchdog
        # must see the live proc to kill the group)
        return self._submit(job, run_proc, (), {})

    # -- control -------------------------------------------------------------
    def cancel(self, name: str, reason: str = "cancelled") -> bool:
        """Cancel a job: queued jobs never start; running process jobs are
        killed; running thread jobs get their cancel_event (cooperative)."""
        with self._lock:
            job = self._jobs.get(name)
        if job is None or job.done():
            return False
        job.outcome = "cancelled" if reason == "cancelled" else reason
        job.cancel_event.set()
        if job.proc is not None and job.proc.poll() is None:
            _kill_group(job.proc)
            for d in (job.devices or
                      ([job.device] if job.device is not None else [])):
                self._device_slot(d).revoke()
        if self._metadata is not None:
            try:
                self._metadata.update_finished_flag(
                    name, finished=True, exception=reason)
            except Exception:
                pass
        return True

    def _ensure_watchdog(self) -> None:
        with self._lock:
            if self._watchdog is not None and self._watchdog.is_alive():
                return
            self._watchdog = threading.Thread(target=self._watch, daemon=True,
                                              name="lo-watchdog")
            self._watchdog.start()

    def _watch(self) -> None:
        while True:
            time.sleep(0.25)
            now = time.time()
            with self._lock:
                jobs = [j for j in self._jobs.values()
                        if j.timeout is not None and not j.done()]
            if not jobs:
                # no timed jobs left: let the watchdog exit (restarted on the
                # next timed submit)
                with self._lock:
                    active = any(j.timeout is not None and not j.done()
                                 for j in self._jobs.values())
                    if not active:
                        self._watchdog = None
                        return
                continue
            for j in jobs:
                start = j.started_at or j.submitted_at
                if now - start > j.timeout:
                    self.cancel(j.name, reason=f"timeout after {j.timeout}s")

    # -- introspection --------------------------------------------------------
    def get(self, name: str) -> Optional[Job]:
        with self._lock:
            return self._jobs.get(name)

    def stats(self) -> Dict[str, Any]:
        with self._lock:
            jobs = list(self._jobs.values())
            owners = {d: s.owner for d, s in self._device_slots.items()}
        by_state: Dict[str, int] = {}
        for j in jobs:
            by_state[j.state] = by_state.get(j.state, 0) + 1
        return {
            "jobs": by_state,
            "running": [j.name for j in jobs if j.state == "running"],
            "queued": [j.name for j in jobs if j.state == "queued"],
            "deviceOwners": owners,
            "gpuDevices": self.gpu_devices(),
        }

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
        with self._lock:
            jobs = list(self._jobs.values())
        for j in jobs:
            if j.proc is not None and j.proc.poll() is None:
                _kill_group(j.proc)
        self._pool.shutdown(wait=True)


class JobCancelled(RuntimeError):
    pass


def _kill_group(proc: subprocess.Popen, grace: float = 3.0) -> None:
    """SIGTERM the process group, escalate to SIGKILL after ``grace``.
    Targets the exact group created by start_new_session — never a pattern."""
    try:
        pgid = os.getpgid(proc.pid)
    except ProcessLookupError:
        return
    try:
        os.killpg(pgid, signal.SIGTERM)
    except ProcessLookupError:
        return
    deadline = time.time() + grace
    while time.time() < deadline:
        if proc.poll() is not None:
            return
        time.sleep(0.1)
    try:
        os.killpg(pgid, signal.SIGKILL)
    except ProcessLookupError:
        pass
