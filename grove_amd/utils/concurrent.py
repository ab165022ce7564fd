"""Concurrency helpers.

Parity source: operator/internal/utils/concurrent.go:190 — RunConcurrently /
RunConcurrentlyWithSlowStart: named tasks with panic recovery; slow start doubles the
batch size (1, 2, 4, …) so a systematic failure (bad template, quota) is discovered
after one cheap attempt instead of hammering the apiserver with N failures.
"""
from __future__ import annotations

import logging
from concurrent.futures import ThreadPoolExecutor
from typing import Callable, List, Tuple

log = logging.getLogger("grove.concurrent")

Task = Tuple[str, Callable[[], None]]


def run_concurrently(tasks: List[Task], max_workers: int = 8) -> List[Exception]:
    errors: List[Exception] = []
    if not tasks:
        return errors
    if len(tasks) == 1:
        try:
            tasks[0][1]()
        except Exception as e:
            errors.append(e)
        return errors
    with ThreadPoolExecutor(max_workers=min(max_workers, len(tasks))) as pool:
        futs = {pool.submit(fn): name for name, fn in tasks}
        for fut, name in futs.items():
            try:
                fut.result()
            except Exception as e:
                log.debug("task %s failed: %s", name, e)
                errors.append(e)
    return errors


def run_concurrently_with_slow_start(tasks: List[Task], initial_batch: int = 1,
                                     max_workers: int = 8) -> List[Exception]:
    """Execute in doubling batches; abort remaining work when a whole batch fails."""
    errors: List[Exception] = []
    i = 0
    batch = max(1, initial_batch)
    while i < len(tasks):
        chunk = tasks[i:i + batch]
        errs = run_concurrently(chunk, max_workers)
        errors.extend(errs)
        if len(errs) == len(chunk):
            break  # systematic failure — stop hammering
        i += len(chunk)
        batch *= 2
    return errors
