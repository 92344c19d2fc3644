"""Fault-tolerant parallel task pool for the mining pipeline.

Capability parity with the corpus' failure-detection story (SURVEY.md §5:
ray heartbeats node_manager.cc:393 + task re-execution
reconstruction_policy.h:40, at mining scale): tasks run in worker
subprocesses; a crashed or hung worker is detected and its task re-executed
elsewhere, up to `retries` times.  Deterministic output order.
"""
from __future__ import annotations

import multiprocessing as mp
import queue
import time
import traceback
from typing import Any, Callable, List, Sequence, Tuple

Task = Tuple[Callable, tuple]


def _worker(task_q, result_q):
    while True:
        try:
            item = task_q.get(timeout=1.0)
        except queue.Empty:
            return
        if item is None:
            return
        idx, fn, args = item
        try:
            result_q.put((idx, True, fn(*args)))
        except BaseException:
            result_q.put((idx, False, traceback.format_exc()))


class TaskFailed(RuntimeError):
    pass


def run_tasks(tasks: Sequence[Task], workers: int = 4, retries: int = 2,
              task_timeout: float = 600.0) -> List[Any]:
    """Run tasks across worker processes; retry on crash; ordered results."""
    ctx = mp.get_context("spawn")
    task_q = ctx.Queue()
    result_q = ctx.Queue()
    attempts = {i: 0 for i in range(len(tasks))}
    for i, (fn, args) in enumerate(tasks):
        task_q.put((i, fn, args))
        attempts[i] += 1
    procs = [ctx.Process(target=_worker, args=(task_q, result_q), daemon=True)
             for _ in range(min(workers, max(len(tasks), 1)))]
    for p in procs:
        p.start()

    results: dict = {}
    deadline = time.monotonic() + task_timeout * (1 + len(tasks) / max(workers, 1))
    last_err = ""
    while len(results) < len(tasks):
        try:
            idx, ok, payload = result_q.get(timeout=2.0)
            if ok:
                results[idx] = payload
            else:
                last_err = payload
                if attempts[idx] <= retries:
                    task_q.put((idx, *tasks[idx]))
                    attempts[idx] += 1
                else:
                    _shutdown(procs, task_q)
                    raise TaskFailed(
                        f"task {idx} failed after {attempts[idx]} attempts:\n"
                        f"{payload}")
            continue
        except queue.Empty:
            pass
        # failure detection: a dead worker with work outstanding -> respawn
        alive = [p for p in procs if p.is_alive()]
        if len(alive) < len(procs):
            dead = len(procs) - len(alive)
            procs = alive
            missing = [i for i in range(len(tasks)) if i not in results]
            # re-enqueue anything that could have died with the worker
            for i in missing:
                if attempts[i] <= retries:
                    task_q.put((i, *tasks[i]))
                    attempts[i] += 1
            for _ in range(dead):
                p = ctx.Process(target=_worker, args=(task_q, result_q),
                                daemon=True)
                p.start()
                procs.append(p)
        if time.monotonic() > deadline:
            _shutdown(procs, task_q)
            raise TaskFailed(
                f"pool timed out with {len(tasks) - len(results)} tasks "
                f"outstanding; last error:\n{last_err}")
    _shutdown(procs, task_q)
    return [results[i] for i in range(len(tasks))]


def _shutdown(procs, task_q):
    for _ in procs:
        try:
            task_q.put(None)
        except Exception:
            pass
    for p in procs:
        p.join(timeout=5)
        if p.is_alive():
            p.terminate()
