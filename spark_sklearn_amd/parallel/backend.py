"""Execution backends for the (candidate x fold) task graph.

The reference's execution layer is Apache Spark: cloudpickled
closures over driver->executor RPC, TorrentBroadcast of X/y, ``collect()`` of
results.  Here the same task graph runs on:

* :class:`SequentialBackend` -- in-process loop (debugging, tiny grids).
* :class:`ThreadBackend`     -- CPU thread pool; sklearn releases the GIL in
  its fit hot loops, so threads give real parallelism without pickling.
  This is the moral equivalent of Spark local mode (BASELINE config 1).
* :class:`DistributedBackend` (parallel/distributed.py) -- one process per
  GPU over torch.distributed (RCCL on ROCm / gloo on CPU), static
  round-robin task partition, score all-reduce, rank-0 refit.

Independent of the backend, *how a single task's fit executes* is routed by
:func:`execute_local_tasks`: estimators from :mod:`spark_sklearn_amd.models`
implement the batched-device protocol and run through the CDNA4 HIP engine
(many candidates folded into single kernel launches); anything else runs the
host path (clone -> fit -> score with real sklearn).
"""

from __future__ import annotations

import os
from concurrent.futures import ThreadPoolExecutor
from typing import List, Optional

from ..base_search import Task, TaskResult, fit_and_score_task

__all__ = ["Backend", "SequentialBackend", "ThreadBackend",
           "execute_local_tasks"]


def _is_hip_batched(estimator) -> bool:
    """True if the estimator family implements the batched-device protocol."""
    return getattr(type(estimator), "_hip_batched", False)


def _device_available() -> bool:
    try:
        import torch

        return torch.cuda.is_available()
    except Exception:  # pragma: no cover - torch always present in this env
        return False


def execute_local_tasks(estimator, X, y, tasks: List[Task], *, scorer,
                        fit_params, return_train_score, error_score, verbose,
                        n_threads: Optional[int] = None,
                        device=None, progress=None,
                        column_batch: Optional[int] = None
                        ) -> List[TaskResult]:
    """Execute this process's share of tasks, on device or host.

    Device path: candidates batched per fold into single HIP kernel launches
    (the key GPU-efficiency idea, SURVEY.md section 7.1 -- a lone
    256-feature fit cannot feed 256 CUs; B candidates per launch can).
    Host path: thread pool over per-task sklearn fits.
    """
    if not tasks:
        return []
    if _is_hip_batched(estimator) and (device is not None or _device_available()):
        from ..engine import run_batched_tasks

        kw = {}
        if column_batch:
            kw["column_batch"] = int(column_batch)
        return run_batched_tasks(
            estimator, X, y, tasks, scorer=scorer, fit_params=fit_params,
            return_train_score=return_train_score, error_score=error_score,
            verbose=verbose, device=device, progress=progress, **kw)

    n_threads = n_threads or min(len(tasks), os.cpu_count() or 4)
    if n_threads <= 1 or len(tasks) == 1:
        out = []
        for i, t in enumerate(tasks):
            out.append(fit_and_score_task(
                estimator, X, y, scorer, t, fit_params=fit_params,
                return_train_score=return_train_score,
                error_score=error_score))
            _report(progress, i + 1, len(tasks))
        return out
    with ThreadPoolExecutor(max_workers=n_threads) as pool:
        futs = [pool.submit(fit_and_score_task, estimator, X, y, scorer, t,
                            fit_params=fit_params,
                            return_train_score=return_train_score,
                            error_score=error_score)
                for t in tasks]
        out = []
        for i, f in enumerate(futs):
            out.append(f.result())
            _report(progress, i + 1, len(futs))
        return out


def _report(progress, done, total):
    """Invoke the context's progress callback; callbacks must never kill
    a search."""
    if progress is None:
        return
    try:
        progress(done, total)
    except Exception:  # noqa: BLE001 - observability must not break work
        pass


class Backend:
    """Abstract execution backend for a search."""

    def run_search(self, estimator, X, y, tasks, *, scorer, fit_params,
                   return_train_score, error_score, verbose):
        raise NotImplementedError

    # Optional: backends may override refit (e.g. rank-0 refit + broadcast).
    refit = None


class SequentialBackend(Backend):
    """Single-threaded in-process execution (debugging / tiny grids)."""

    def run_search(self, estimator, X, y, tasks, *, scorer, fit_params,
                   return_train_score, error_score, verbose):
        return execute_local_tasks(
            estimator, X, y, tasks, scorer=scorer, fit_params=fit_params,
            return_train_score=return_train_score, error_score=error_score,
            verbose=verbose, n_threads=1,
            progress=getattr(self, "progress", None),
            column_batch=getattr(self, "candidate_batch", None))


class ThreadBackend(Backend):
    """CPU thread-pool backend -- the Spark-local-mode equivalent."""

    def __init__(self, n_threads: Optional[int] = None):
        self.n_threads = n_threads

    def run_search(self, estimator, X, y, tasks, *, scorer, fit_params,
                   return_train_score, error_score, verbose):
        return execute_local_tasks(
            estimator, X, y, tasks, scorer=scorer, fit_params=fit_params,
            return_train_score=return_train_score, error_score=error_score,
            verbose=verbose, n_threads=self.n_threads,
            progress=getattr(self, "progress", None),
            column_batch=getattr(self, "candidate_batch", None))
