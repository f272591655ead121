"""Ray-free local GPU scheduler (replaces the reference's Ray/Redis cluster,
reference search.py:60-67, 164, 230-245).

Worker processes are spawned once and pinned to GPUs via
HIP_VISIBLE_DEVICES (set in the child before the HIP runtime initializes);
tasks are (fn, args, kwargs) tuples dispatched over queues; results come
back as futures. On a CPU-only machine workers simply run unpinned, which is
how the multi-process CPU tests exercise the scheduler.
"""
from __future__ import annotations

import multiprocessing as mp
import os
import queue
import traceback
from typing import Any, Callable, Dict, List, Optional


def _set_pdeathsig():
    """Kill this worker if the parent dies (even by SIGKILL) — the role the
    reference's safe_shell_exec parent-death pipe plays
    (reference safe_shell_exec.py:29-60)."""
    try:
        import ctypes
        import signal
        libc = ctypes.CDLL("libc.so.6", use_errno=True)
        PR_SET_PDEATHSIG = 1
        libc.prctl(PR_SET_PDEATHSIG, signal.SIGTERM)
    except Exception:
        pass  # non-Linux / restricted: daemon=True still covers clean exits


def _worker_main(worker_id: int, gpu_id: Optional[int], task_q, result_q):
    _set_pdeathsig()
    if gpu_id is not None:
        os.environ["HIP_VISIBLE_DEVICES"] = str(gpu_id)
        os.environ["CUDA_VISIBLE_DEVICES"] = str(gpu_id)
    while True:
        item = task_q.get()
        if item is None:
            break
        task_id, fn, args, kwargs = item
        try:
            result = fn(*args, **kwargs)
            result_q.put((task_id, True, result))
        except Exception:
            result_q.put((task_id, False, traceback.format_exc()))


class TaskError(RuntimeError):
    pass


class Future:
    def __init__(self, scheduler: "LocalGpuScheduler", task_id: int):
        self._sched = scheduler
        self._id = task_id

    def result(self, timeout: Optional[float] = None) -> Any:
        return self._sched._wait_for(self._id, timeout)

    def done(self) -> bool:
        self._sched._drain(block=False)
        return self._id in self._sched._results


class LocalGpuScheduler:
    """N long-lived workers, one per GPU (or CPU workers when no GPU)."""

    def __init__(self, n_workers: Optional[int] = None, use_gpus: bool = True):
        import torch
        n_gpus = torch.cuda.device_count() if use_gpus else 0
        if n_workers is None:
            n_workers = n_gpus if n_gpus > 0 else 2
        ctx = mp.get_context("spawn")
        self._task_q = ctx.Queue()
        self._result_q = ctx.Queue()
        self._results: Dict[int, Any] = {}
        self._next_id = 0
        self._procs: List[mp.Process] = []
        for w in range(n_workers):
            gpu = w % n_gpus if n_gpus > 0 else None
            p = ctx.Process(target=_worker_main, args=(w, gpu, self._task_q, self._result_q),
                            daemon=True)
            p.start()
            self._procs.append(p)
        self.n_workers = n_workers

    def submit(self, fn: Callable, *args, **kwargs) -> Future:
        task_id = self._next_id
        self._next_id += 1
        self._task_q.put((task_id, fn, args, kwargs))
        return Future(self, task_id)

    def map(self, fn: Callable, arg_list) -> List[Any]:
        futs = [self.submit(fn, *a) if isinstance(a, tuple) else self.submit(fn, a)
                for a in arg_list]
        return [f.result() for f in futs]

    def _drain(self, block: bool, timeout: Optional[float] = None):
        while True:
            try:
                tid, ok, payload = self._result_q.get(block=block, timeout=timeout)
            except queue.Empty:
                return
            self._results[tid] = (ok, payload)
            block = False

    def _wait_for(self, task_id: int, timeout: Optional[float]):
        while task_id not in self._results:
            self._drain(block=True, timeout=timeout)
        ok, payload = self._results.pop(task_id)
        if not ok:
            raise TaskError(payload)
        return payload

    def shutdown(self):
        for _ in self._procs:
            self._task_q.put(None)
        for p in self._procs:
            p.join(timeout=30)
            if p.is_alive():
                p.terminate()
        self._procs.clear()

    def __enter__(self):
        return self

    def __exit__(self, *exc):
        self.shutdown()
