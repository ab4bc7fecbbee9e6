"""Concurrent base-learner fits on HIP streams.

The reference's ``HasParallelism`` dispatches independent base-learner
fits on a driver-side thread pool (Spark Futures + ThreadUtils,
BaggingRegressor.scala:145-166, StackingRegressor.scala:141-153).  The
MI355X analog: one Python thread per in-flight fit, each running its
kernels on its OWN HIP stream (``torch.cuda.Stream`` is thread-local
current), so one fit's host-side bookkeeping (per-level split fetch)
overlaps another fit's device work.  Wins are largest for many SMALL
fits (OOF folds, small bagged learners) where a single fit cannot fill
256 CUs and the host gaps dominate.

Scope rules:
  * world_size == 1 only — the per-level histogram all-reduce inside
    tree fits is ordered; concurrent collectives from threads would
    deadlock RCCL.  Distributed runs keep the sequential order (every
    rank identical).
  * CUDA only; on CPU threads just contend for the GIL.
  * Caller pre-warms any shared caches (e.g. the binned matrix) before
    spawning: the tasks must not race on first-build.
"""

from __future__ import annotations

from typing import Callable, List, Sequence

import torch


def parallel_fits(tasks: Sequence[Callable], parallelism: int,
                  warm_first: bool = False) -> List:
    """Run independent fit closures, up to ``parallelism`` concurrently
    on per-thread HIP streams; returns results in task order.  Falls back
    to sequential execution whenever concurrency is unsafe (distributed)
    or pointless (CPU, single task, parallelism 1).

    ``warm_first``: run tasks[0] alone before the rest — for task sets
    whose first execution populates a shared cache (binned features) the
    others must not race to build."""
    from . import get_comm

    n = len(tasks)
    if (
        parallelism <= 1
        or n <= 1
        or not torch.cuda.is_available()
        or get_comm().is_distributed
    ):
        return [t() for t in tasks]

    from concurrent.futures import ThreadPoolExecutor

    results: List = [None] * n
    first = 0
    if warm_first:
        results[0] = tasks[0]()
        first = 1
        if n == 1:
            return results
    # everything produced so far (shared binned matrix, weights) must be
    # visible to the new streams
    torch.cuda.synchronize()

    def run(i: int):
        stream = torch.cuda.Stream()
        with torch.cuda.stream(stream):
            results[i] = tasks[i]()
        stream.synchronize()

    with ThreadPoolExecutor(max_workers=int(parallelism)) as ex:
        list(ex.map(run, range(first, n)))
    # downstream consumers run on the default stream
    torch.cuda.synchronize()
    return results
