"""Shared search engine: task-graph construction, dispatch, cv_results_ assembly.

MI355X-native re-implementation of the behavior of the reference's shared
``_fit`` (spark-sklearn ``grid_search.py::_fit`` ~L110-300 [R], see
SURVEY.md section 3.1).  The reference builds a (candidate x fold) task list,
ships it to Spark executors via ``sc.parallelize(...).map(fun).collect()``
with the training matrix broadcast once per executor, then assembles the
sklearn ``cv_results_`` dict on the driver.

Here the same task graph is scheduled onto a pluggable :class:`Backend`:

* ``LocalBackend``       -- CPU thread pool (the moral equivalent of Spark
                            local mode); real sklearn fits inside tasks.
* ``DistributedBackend`` -- one process per GPU over ``torch.distributed``
                            (RCCL on ROCm; gloo for CPU tests), the dataset
                            broadcast once into each GPU's HBM, fold scores
                            all-gathered, refit on rank 0.
* ``HipEngineBackend``   -- batched candidate fits through the hand-written
                            CDNA4 HIP kernels (spark_sklearn_amd.ops).

The ``cv_results_`` layout is the conformance contract -- identical key set,
shapes, masked ``param_*`` arrays, tie-ranking and ``best_index_`` semantics
as ``sklearn.model_selection.GridSearchCV`` (oracle: installed sklearn),
which the reference deliberately mirrors.
"""

from __future__ import annotations

import numbers
import time
import warnings
from collections import defaultdict
from typing import Any, Dict, List

import numpy as np
from numpy.ma import MaskedArray
from scipy.stats import rankdata

from sklearn.base import BaseEstimator, MetaEstimatorMixin, clone, is_classifier
from sklearn.utils.metaestimators import available_if
from sklearn.metrics import check_scoring
from sklearn.model_selection import check_cv
from sklearn.utils.validation import indexable

__all__ = ["Task", "TaskResult", "fit_and_score_task", "BaseSearchCV"]


def _search_estimator_has(attr):
    """available_if check: delegate attribute availability to the
    (fitted) estimator; refit=False raises the sklearn message chain."""

    def check(self):
        self._check_refit_for(attr)
        if hasattr(self, "best_estimator_"):
            getattr(self.best_estimator_, attr)
            return True
        getattr(self.estimator, attr)
        return True

    return check


class Task:
    """One (candidate, fold) unit of work.

    Plain-old-data by design: the reference ships ``(params, train_idx,
    test_idx)`` tuples tagged with a submission index so results can be
    re-ordered deterministically after an unordered ``collect()``
    (reference ``grid_search.py`` ~L120 [R]).  We keep the same tagging so
    every backend (threadpool, torch.distributed, HIP engine) is free to
    return results in any order.
    """

    __slots__ = ("index", "cand_index", "fold_index", "params", "train", "test")

    def __init__(self, index, cand_index, fold_index, params, train, test):
        self.index = index
        self.cand_index = cand_index
        self.fold_index = fold_index
        self.params = params
        self.train = train
        self.test = test


class TaskResult:
    """Fixed-width per-task result record.

    Mirrors the tuple returned by the closure the reference ships to
    executors (``_fit_and_score`` output: train_score?, test_score,
    n_test_samples, fit_time, score_time -- SURVEY.md section 3.1).  This is
    also the RCCL all-gather wire format (6 scalars / task, see
    parallel/distributed.py).
    """

    __slots__ = ("index", "test_score", "train_score", "n_test",
                 "fit_time", "score_time", "error")

    def __init__(self, index, test_score, train_score, n_test, fit_time,
                 score_time, error=None):
        self.index = index
        self.test_score = test_score
        self.train_score = train_score
        self.n_test = n_test
        self.fit_time = fit_time
        self.score_time = score_time
        self.error = error     # formatted exception text for failed fits


class _MultiScorer:
    """Dispatchable wrapper for multi-metric scoring: calls every scorer
    of the dict and returns ``{name: float}``.  Backends treat it exactly
    like a single scorer; the dict-valued scores flow through TaskResult
    into the per-metric ``cv_results_`` assembly (sklearn multimetric
    contract -- a capability the reference era predates)."""

    def __init__(self, scorers):
        self.scorers = dict(scorers)

    def keys(self):
        return self.scorers.keys()

    def __call__(self, est, X, y=None):
        if y is None:
            return {k: float(sc(est, X)) for k, sc in self.scorers.items()}
        return {k: float(sc(est, X, y)) for k, sc in self.scorers.items()}


def _coerce_score(v):
    """float for single-metric, {name: float} for multimetric scorers."""
    if isinstance(v, dict):
        for k, val in v.items():
            if not isinstance(val, numbers.Number):
                raise ValueError(
                    f"scoring must return a number, got {val!r} "
                    f"(key {k!r}) instead.")
        return {k: float(val) for k, val in v.items()}
    return float(v)


def return_train_score_effective(search, result) -> bool:
    """Whether a per-fit verbose line should show the train score."""
    if not search.return_train_score:
        return False
    ts = result.train_score
    return not (isinstance(ts, float) and np.isnan(ts))


def fit_and_score_task(base_estimator, X, y, scorer, task: Task,
                       fit_params=None, return_train_score=True,
                       error_score="raise") -> TaskResult:
    """Clone-fit-score one task on the host (CPU tier).

    Behavioral equivalent of the worker closure in the reference
    (``grid_search.py`` ~L155 [R]): ``clone`` -> ``set_params`` ->
    ``fit(X[train], y[train])`` -> score on the test fold, with
    ``error_score`` swallow-or-raise semantics for robustness on large
    grids (one bad candidate must not kill a 10k-task job).
    """
    est = clone(base_estimator)
    if task.params:
        # clone parameter values that are themselves estimators, so grid
        # entries are never fitted in place (sklearn contract)
        est.set_params(**{k: clone(v, safe=False)
                          for k, v in task.params.items()})
    fit_params = fit_params or {}
    if fit_params:
        # legacy sklearn semantics: per-sample fit params (sample_weight
        # and friends -- any array of length n_samples) are sliced to the
        # train fold
        n_samples = X.shape[0] if hasattr(X, "shape") else len(X)
        fit_params = {
            k: (_safe_index(np.asarray(v), task.train)
                if hasattr(v, "__len__") and not np.isscalar(v) and
                len(v) == n_samples else v)
            for k, v in fit_params.items()}

    if _is_pairwise(est):
        # precomputed kernels/distances: slice BOTH axes (train columns)
        import scipy.sparse as _sp

        if not (hasattr(X, "shape") or _sp.issparse(X)):
            raise ValueError(
                "Precomputed kernels or affinity matrices have to be "
                "passed as arrays or sparse matrices.")
        X = np.asarray(X)
        if X.ndim != 2 or X.shape[0] != X.shape[1]:
            raise ValueError(
                "X should be a square kernel matrix")
        X_train = X[np.ix_(task.train, task.train)]
        X_test = X[np.ix_(task.test, task.train)]
    else:
        X_train, X_test = _safe_index(X, task.train), _safe_index(X, task.test)
    y_train, y_test = _safe_index(y, task.train), _safe_index(y, task.test)

    t0 = time.perf_counter()
    try:
        if y_train is None:
            est.fit(X_train, **fit_params)
        else:
            est.fit(X_train, y_train, **fit_params)
        fit_time = time.perf_counter() - t0
    except Exception as exc:  # noqa: BLE001 - error_score contract
        fit_time = time.perf_counter() - t0
        if error_score == "raise":
            raise
        if not isinstance(error_score, numbers.Number):
            raise ValueError(
                "error_score must be the string 'raise' or a numeric value"
            ) from exc
        import traceback

        err = traceback.format_exc()
        n_test = len(task.test)
        return TaskResult(task.index, float(error_score),
                          float(error_score) if return_train_score else np.nan,
                          n_test, fit_time, 0.0, error=err)

    # scoring errors are NOT fit failures (sklearn contract): they get
    # error_score + their own warning, without tripping FitFailedWarning
    t1 = time.perf_counter()
    try:
        if y_test is None:
            test_score = scorer(est, X_test)
            score_time = time.perf_counter() - t1
            train_score = scorer(est, X_train) if return_train_score \
                else np.nan
        else:
            test_score = scorer(est, X_test, y_test)
            score_time = time.perf_counter() - t1
            train_score = scorer(est, X_train, y_train) \
                if return_train_score else np.nan
    except Exception as exc:  # noqa: BLE001 - error_score contract
        score_time = time.perf_counter() - t1
        if error_score == "raise":
            raise
        if not isinstance(error_score, numbers.Number):
            raise ValueError(
                "error_score must be the string 'raise' or a numeric value"
            ) from exc
        warnings.warn(
            "Scoring failed. The score on this train-test partition for "
            f"these parameters will be set to {error_score}. "
            f"Details: \n{exc!r}", UserWarning)
        test_score = float(error_score)
        train_score = float(error_score) if return_train_score else np.nan

    n_test = len(task.test)
    return TaskResult(task.index, _coerce_score(test_score),
                      _coerce_score(train_score)
                      if isinstance(train_score, dict)
                      else float(train_score),
                      n_test, fit_time, score_time)


def _is_pairwise(estimator) -> bool:
    try:
        from sklearn.utils import get_tags

        return bool(get_tags(estimator).input_tags.pairwise)
    except Exception:
        return False


def _safe_index(data, indices):
    if data is None:
        return None
    if hasattr(data, "iloc"):
        return data.iloc[indices]
    if isinstance(data, np.ndarray):
        return data[indices]
    if hasattr(data, "__getitem__") and hasattr(data, "shape"):  # scipy sparse
        return data[indices]
    return [data[i] for i in indices]


class BaseSearchCV(MetaEstimatorMixin, BaseEstimator):
    """Common machinery for :class:`GridSearchCV` / :class:`RandomizedSearchCV`.

    Constructor surface follows the reference era (spark-sklearn 0.3.0
    tracking sklearn 0.18/0.19 [R]) with a leading context argument in the
    subclasses; ``n_jobs``/``pre_dispatch`` are accepted-but-ignored for
    drop-in compatibility, exactly as the reference ignores them
    (SURVEY.md C2).

    Divergences from the 0.18-era surface (documented per SURVEY.md 3.1):

    * ``iid`` defaults to ``False`` (modern sklearn semantics -- unweighted
      mean across folds) so that ``cv_results_`` matches the installed
      sklearn oracle bit-for-bit; pass ``iid=True`` for the reference's
      n_test-weighted means.
    * ``return_train_score`` keeps the reference default ``True``.
    """

    def __init__(self, estimator, scoring=None, fit_params=None, n_jobs=1,
                 iid=False, refit=True, cv=None, verbose=0,
                 pre_dispatch="2*n_jobs", error_score=np.nan,
                 return_train_score=True):
        self.estimator = estimator
        self.scoring = scoring
        self.fit_params = fit_params
        self.n_jobs = n_jobs          # accepted-but-ignored (reference parity)
        self.iid = iid
        self.refit = refit
        self.cv = cv
        self.verbose = verbose
        self.pre_dispatch = pre_dispatch  # accepted-but-ignored
        self.error_score = error_score
        self.return_train_score = return_train_score

    # -- subclass hooks ----------------------------------------------------
    def _get_param_iterator(self):
        raise NotImplementedError

    def _maybe_broadcast_input(self, backend, X, y):
        if getattr(backend, "world_size", 1) <= 1 or \
                not hasattr(backend, "broadcast_data"):
            if X is None:
                raise ValueError(
                    "X is None: only ranks of a distributed backend may "
                    "omit X (rank 0 must always provide it)")
            return X, y
        dist = backend._dist
        flags = [None] * backend.world_size
        dist.all_gather_object(flags, X is not None)
        if all(flags):
            return X, y
        X_t, y_t = backend.broadcast_data(X, y)
        dev = getattr(backend, "device", None)
        if dev is None or getattr(dev, "type", "cpu") != "cuda":
            X_t = X_t.numpy()
            y_t = y_t.numpy() if y_t is not None else None
        return X_t, y_t

    def _get_backend(self):
        """Resolve the execution backend from the context handle."""
        ctx = getattr(self, "ctx", None)
        from .parallel import resolve_backend

        return resolve_backend(ctx)

    # -- the engine --------------------------------------------------------
    def fit(self, X, y=None, groups=None, **fit_params):
        """Run the full search: build tasks, dispatch, aggregate, refit."""
        estimator = self.estimator
        backend = self._get_backend()
        # RC1: the reference's _fit broadcasts X/y itself (SURVEY.md
        # section 3.1); under a distributed backend, non-root ranks may
        # pass X=None and receive rank 0's arrays (resident in HBM on the
        # RCCL tier) before any splitting happens.
        X, y = self._maybe_broadcast_input(backend, X, y)
        cv = check_cv(self.cv, y, classifier=is_classifier(estimator))
        if isinstance(self.scoring, (dict, list, tuple, set)):
            from sklearn.metrics._scorer import _check_multimetric_scoring

            scoring = self.scoring
            if isinstance(scoring, (tuple, set)):
                scoring = list(scoring)
            scorers_dict = _check_multimetric_scoring(estimator, scoring)
            self._validate_multimetric_refit(scorers_dict.keys())
            self.scorer_ = scorers_dict
            dispatch_scorer = _MultiScorer(scorers_dict)
        else:
            self.scorer_ = check_scoring(estimator, scoring=self.scoring)
            dispatch_scorer = self.scorer_

        X, y, groups = indexable(X, y, groups)
        merged_fit_params = dict(self.fit_params or {})
        merged_fit_params.update(fit_params)
        if "sample_weight" in merged_fit_params:
            self._warn_scorers_without_sample_weight()

        candidates = list(self._get_param_iterator())
        n_candidates = len(candidates)
        splits = list(cv.split(X, y, groups))
        n_splits = len(splits)
        expected_splits = cv.get_n_splits(X, y, groups)
        if n_splits * n_candidates == 0:
            raise ValueError(
                "No fits were performed. Was the CV iterator empty? "
                "Were there no candidates?")
        if expected_splits != n_splits:
            raise ValueError(
                "cv.split and cv.get_n_splits returned inconsistent "
                f"results. Expected {expected_splits} splits, got "
                f"{n_splits}")

        if self.verbose > 0:
            print(f"Fitting {n_splits} folds for each of {n_candidates} "
                  f"candidates, totalling {n_candidates * n_splits} fits")

        tasks = []
        idx = 0
        for cand_i, params in enumerate(candidates):
            for fold_i, (train, test) in enumerate(splits):
                tasks.append(Task(idx, cand_i, fold_i, params,
                                  np.asarray(train), np.asarray(test)))
                idx += 1

        # resume: replay completed (candidate, fold) fits from a prior
        # run's score log; only the remainder executes
        resume_path = getattr(backend, "resume_from_log", None)
        restored: List[TaskResult] = []
        todo = tasks
        log_header = self._score_log_header(X, y, merged_fit_params)
        if resume_path:
            done = self._load_resume_records(resume_path, n_splits,
                                             log_header)
            if done:
                todo = []
                for t in tasks:
                    rec = done.get((self._param_key(t.params),
                                    t.fold_index))
                    if rec is None:
                        todo.append(t)
                    else:
                        tr = rec.get("train_score")
                        restored.append(TaskResult(
                            t.index, rec["test_score"],
                            np.nan if tr is None else tr,
                            rec["n_test"], rec["fit_time"],
                            rec["score_time"]))
                if self.verbose > 0:
                    print(f"[search] resume: {len(restored)} of "
                          f"{len(tasks)} fits restored from "
                          f"{resume_path}")
        self.n_resumed_ = len(restored)

        t_run0 = time.perf_counter()
        results = backend.run_search(
            estimator, X, y, todo,
            scorer=dispatch_scorer,
            fit_params=merged_fit_params,
            return_train_score=self.return_train_score,
            error_score=self.error_score,
            verbose=self.verbose,
        ) if todo else []
        results = list(results) + restored
        run_elapsed = time.perf_counter() - t_run0
        # observability counters: candidate-fits/sec
        # is the north-star metric; exposed on the fitted object
        self.n_fits_ = len(todo)
        self.fit_duration_ = run_elapsed
        self.fits_per_sec_ = len(todo) / max(run_elapsed, 1e-9)
        if self.verbose > 0:
            print(f"[search] {self.n_fits_} fits in {run_elapsed:.2f}s "
                  f"({self.fits_per_sec_:.1f} candidate-fits/s)")
        from .util import record_search_metrics

        record_search_metrics(self.n_fits_, self.fits_per_sec_)
        # restore deterministic submit order regardless of backend ordering
        by_index = {r.index: r for r in results}
        missing = [t.index for t in tasks if t.index not in by_index]
        if missing:
            raise RuntimeError(
                f"backend returned {len(results)} results for {len(tasks)} "
                f"tasks; missing indices {missing[:5]}..."
            )
        ordered = [by_index[i] for i in range(len(tasks))]
        self._handle_fit_failures(tasks, ordered)

        if self.verbose > 2 and getattr(backend, "rank", 0) == 0:
            # sklearn's per-fit verbose lines (single-metric form)
            for t, r in zip(tasks, ordered):
                if isinstance(r.test_score, dict):
                    continue
                pstr = ", ".join(f"{k}={v}" for k, v in
                                 sorted(t.params.items()))
                if return_train_score_effective(self, r):
                    msg = (f"score=(train={r.train_score:.3f}, "
                           f"test={r.test_score:.3f})")
                else:
                    msg = f"score={r.test_score:.3f}"
                print(f"[CV {t.fold_index + 1}/{n_splits}] END {pstr};, "
                      f"{msg} total time={r.fit_time + r.score_time:.3f}s")

        score_log = getattr(backend, "score_log", None)
        if score_log and getattr(backend, "rank", 0) == 0:
            executed = {t.index for t in todo}
            self._append_score_log(
                score_log,
                [t for t in tasks if t.index in executed],
                [r for r in ordered if r.index in executed],
                header=log_header)

        # multimetric-ness is declared (dict/list scoring) or discovered
        # (a plain callable that returned a dict of scores)
        first_ok = next((r for r in ordered if r.error is None), None)
        metric_names = None
        if first_ok is not None and isinstance(first_ok.test_score, dict):
            metric_names = list(first_ok.test_score.keys())
            self._validate_multimetric_refit(metric_names)

        self.cv_results_ = self._assemble_cv_results(
            candidates, ordered, n_candidates, n_splits,
            metric_names=metric_names)
        self.n_splits_ = n_splits
        self.multimetric_ = metric_names is not None
        refit_metric = self.refit if (self.multimetric_ and
                                      isinstance(self.refit, str))             else "score"
        if callable(self.refit):
            # sklearn contract: refit may be a callable selecting
            # best_index_ from cv_results_; best_score_ is then undefined
            best = self.refit(self.cv_results_)
            if not isinstance(best, (int, np.integer)):
                raise TypeError("best_index_ returned is not an integer")
            if not 0 <= best < n_candidates:
                raise IndexError("best_index_ index out of range")
            self.best_index_ = int(best)
            self.best_params_ = self.cv_results_["params"][self.best_index_]
        elif self.refit or not self.multimetric_:
            self.best_index_ = int(np.flatnonzero(
                self.cv_results_[f"rank_test_{refit_metric}"] == 1)[0])
            self.best_score_ = float(
                self.cv_results_[f"mean_test_{refit_metric}"]
                [self.best_index_])
            self.best_params_ = self.cv_results_["params"][self.best_index_]

        if self.refit:
            # param values that are estimators must be cloned so grid
            # entries are never fitted in place (sklearn contract)
            best_estimator = clone(estimator).set_params(
                **{k: clone(v, safe=False)
                   for k, v in self.best_params_.items()})
            t0 = time.perf_counter()
            refit_fn = getattr(backend, "refit", None)
            if refit_fn is not None:
                best_estimator = refit_fn(best_estimator, X, y,
                                          merged_fit_params)
            elif y is not None:
                best_estimator.fit(X, y, **merged_fit_params)
            else:
                best_estimator.fit(X, **merged_fit_params)
            self.refit_time_ = time.perf_counter() - t0
            self.best_estimator_ = best_estimator
        return self

    def _warn_scorers_without_sample_weight(self):
        """sklearn parity: fitting with sample_weight warns per scorer
        that cannot route it (statistically incorrect results)."""
        from inspect import signature

        def accepts(sc):
            try:
                if hasattr(sc, "_accept_sample_weight"):
                    return sc._accept_sample_weight()
                return "sample_weight" in signature(sc).parameters
            except (TypeError, ValueError):
                return True

        if isinstance(self.scoring, dict):
            from sklearn.metrics import get_scorer

            for name, sc in self.scoring.items():
                sc = get_scorer(sc) if isinstance(sc, str) else sc
                if not accepts(sc):
                    warnings.warn(
                        f"The scoring {name}={sc} does not support "
                        "sample_weight, which may lead to statistically "
                        f"incorrect results when fitting {self} with "
                        "sample_weight. ")
        elif self.scoring is not None and not isinstance(self.scoring,
                                                         (list, tuple,
                                                          set)):
            sc = self.scoring
            if isinstance(sc, str):
                from sklearn.metrics import get_scorer

                sc = get_scorer(sc)
            if not accepts(sc):
                warnings.warn(
                    f"The scoring {sc} does not support sample_weight, "
                    "which may lead to statistically incorrect results "
                    f"when fitting {self} with sample_weight. ")

    def _validate_multimetric_refit(self, keys):
        """refit must be False, a callable, or a key of the scorer dict
        (sklearn multimetric contract, same message)."""
        if self.refit is False or self.refit is None or                 callable(self.refit):
            return
        if isinstance(self.refit, str) and self.refit in set(keys):
            return
        raise ValueError(
            "For multi-metric scoring, the parameter refit must be set to "
            "a scorer key or a callable to refit an estimator with the "
            "best parameter setting on the whole data and make the best_* "
            "attributes available for that metric. If this is not needed, "
            "refit should be set to False explicitly.")

    def _handle_fit_failures(self, tasks, ordered):
        """sklearn failure contract: one aggregated FitFailedWarning when
        some fits failed under a numeric error_score; ValueError when ALL
        fits failed."""
        failed = [r for r in ordered if r.error is not None]
        if not failed:
            return
        from collections import Counter

        from sklearn.exceptions import FitFailedWarning

        n, nf = len(ordered), len(failed)
        by_msg = Counter(r.error for r in failed)
        details = "\n".join(
            f"--------------------------------------------------------\n"
            f"{cnt} fits failed with the following error:\n{msg}"
            for msg, cnt in by_msg.items())
        if nf == n:
            raise ValueError(
                f"\nAll the {n} fits failed.\n"
                "It is very likely that your model is misconfigured.\n"
                "You can try to debug the error by setting "
                "error_score='raise'.\n\n"
                f"Below are more details about the failures:\n{details}")
        warnings.warn(
            f"\n{nf} fits failed out of a total of {n}.\n"
            "The score on these train-test partitions for these parameters"
            f" will be set to {self.error_score}.\n"
            "If these failures are not expected, you can try to debug them "
            "by setting error_score='raise'.\n\n"
            f"Below are more details about the failures:\n{details}",
            FitFailedWarning)

    @staticmethod
    def _param_key(params) -> str:
        """Stable text key for a candidate's params (resume matching)."""
        return repr(sorted((k, repr(v)) for k, v in params.items()))

    def _score_log_header(self, X, y, fit_params=None):
        """Identity of this search for score-log resume validation: a
        cheap strided fingerprint of (X, y) AND of any fit-time
        sample_weight, plus the splitter, the estimator class WITH its
        constructor params (a log written with class_weight='balanced'
        or momentum=0.5 must not replay into a differently-configured
        search) and the scoring spec."""
        import hashlib

        h = hashlib.sha1()

        def upd(a):
            if a is None:
                h.update(b"none")
                return
            flat = None
            try:
                if hasattr(a, "detach"):          # torch tensor (any device)
                    t = a.detach().reshape(-1)
                    step = max(1, t.shape[0] // 1024)
                    flat = t[::step][:1024].cpu().numpy()
                    h.update(str(tuple(a.shape)).encode())
                    h.update(str(a.dtype).encode())
                else:
                    arr = np.asarray(a)
                    h.update(str(arr.shape).encode())
                    h.update(str(arr.dtype).encode())
                    if arr.size:
                        f2 = arr.reshape(-1)
                        step = max(1, f2.shape[0] // 1024)
                        flat = f2[::step][:1024]
                if flat is not None:
                    h.update(np.ascontiguousarray(flat).tobytes())
            except Exception:
                h.update(str(type(a)).encode())

        upd(X)
        upd(y)
        upd((fit_params or {}).get("sample_weight"))
        try:
            est_params = repr(sorted(
                (k, repr(v))
                for k, v in self.estimator.get_params(deep=False).items()))
        except Exception:  # noqa: BLE001 - header must never break fit
            est_params = ""
        return {
            "header": 1,
            "data": h.hexdigest(),
            "cv": repr(self.cv),
            "estimator": type(self.estimator).__name__,
            "estimator_params": est_params,
            "fit_param_keys": sorted((fit_params or {}).keys()),
            "scoring": repr(self.scoring),
        }

    def _append_score_log(self, path, tasks, ordered, header=None):
        """Append-only JSONL score log: one record
        per (candidate, fold) fit so a killed 100k-candidate sweep leaves
        its completed work on disk; ``resume_from_log`` replays it."""
        import json
        import os as _os

        if header is not None and (not _os.path.exists(path) or
                                   _os.path.getsize(path) == 0):
            with open(path, "a") as f:
                f.write(json.dumps(header) + "\n")
        with open(path, "a") as f:
            for t, r in zip(tasks, ordered):
                if r.error is not None:
                    continue      # failed fits re-run on resume
                f.write(json.dumps({
                    "cand": t.cand_index, "fold": t.fold_index,
                    "key": self._param_key(t.params),
                    "params": {k: repr(v) for k, v in t.params.items()},
                    "test_score": r.test_score,
                    "train_score": (None if (isinstance(r.train_score,
                                                        float) and
                                             np.isnan(r.train_score))
                                    else r.train_score),
                    "n_test": r.n_test,
                    "fit_time": r.fit_time,
                    "score_time": r.score_time}) + "\n")

    def _load_resume_records(self, path, n_splits, expected_header=None):
        """Completed (param-key, fold) -> TaskResult fields from a prior
        run's score log.  Records missing the resume fields (old-format
        logs) or train scores when this run needs them are skipped, so
        those tasks simply re-run.  A header record written by
        ``_append_score_log`` identifies the search (data fingerprint,
        cv, estimator class, scoring); a mismatching log is refused
        rather than silently replayed, and a header-less (old-format)
        log only warns."""
        import json
        import os as _os
        import warnings as _warnings

        done = {}
        if not path or not _os.path.exists(path):
            return done
        saw_header = False
        with open(path) as f:
            for line in f:
                line = line.strip()
                if not line:
                    continue
                try:
                    rec = json.loads(line)
                except ValueError:
                    continue      # partial last line of a killed run
                if rec.get("header") == 1:
                    saw_header = True
                    if expected_header is not None:
                        for field in expected_header:
                            if field == "header":
                                continue
                            if rec.get(field) != expected_header[field]:
                                raise ValueError(
                                    "resume_from_log refused: the score "
                                    f"log at {path!r} was written for a "
                                    f"different search ({field} differs: "
                                    f"log {rec.get(field)!r} vs current "
                                    f"{expected_header[field]!r}). Delete "
                                    "the log or point resume_from_log at "
                                    "the matching one.")
                    continue
                if "key" not in rec or rec.get("fold") is None:
                    continue
                if rec["fold"] >= n_splits:
                    continue      # different cv: ignore the whole record
                if self.return_train_score and rec.get("train_score") \
                        is None:
                    continue
                done[(rec["key"], rec["fold"])] = rec
        if done and not saw_header:
            _warnings.warn(
                "resume_from_log: score log has no header record (written "
                "by an older version); resuming without validating that "
                "it matches this search's data/cv/estimator/scoring.",
                UserWarning)
        return done

    # -- cv_results_ assembly ---------------------------------------------
    def _assemble_cv_results(self, candidates, ordered_results,
                             n_candidates, n_splits,
                             metric_names=None) -> Dict[str, Any]:
        """Build the sklearn-layout ``cv_results_`` dict.

        Exact key set: ``mean_fit_time``,
        ``std_fit_time``, ``mean_score_time``, ``std_score_time``,
        ``split{k}_test_score``, ``mean_test_score``, ``std_test_score``,
        ``rank_test_score``, train-score variants iff
        ``return_train_score``, masked ``param_<p>`` arrays, ``params``.
        Multimetric (``metric_names``): ``*_test_score`` keys become
        ``*_test_<name>`` per metric, exactly as sklearn lays them out.
        """

        def _metric_matrix(get, name):
            # dict scores from a multimetric scorer; failed fits carry the
            # scalar error_score, broadcast to every metric
            vals = [get(r) for r in ordered_results]
            return np.array(
                [v[name] if isinstance(v, dict) else v for v in vals],
                dtype=np.float64).reshape(n_candidates, n_splits)

        fit_times = np.array(
            [r.fit_time for r in ordered_results], dtype=np.float64
        ).reshape(n_candidates, n_splits)
        score_times = np.array(
            [r.score_time for r in ordered_results], dtype=np.float64
        ).reshape(n_candidates, n_splits)
        n_test = np.array(
            [r.n_test for r in ordered_results], dtype=np.float64
        ).reshape(n_candidates, n_splits)

        results: Dict[str, Any] = {}

        def _store(key, array, splits=False, rank=False, weights=None):
            if splits:
                for k in range(n_splits):
                    results[f"split{k}_{key}"] = array[:, k]
            means = np.average(array, axis=1, weights=weights)
            results[f"mean_{key}"] = means
            # weighted std around the (possibly weighted) mean, as sklearn does
            results[f"std_{key}"] = np.sqrt(
                np.average((array - means[:, np.newaxis]) ** 2,
                           axis=1, weights=weights))
            if key.startswith(("train_", "test_")) and \
                    np.any(~np.isfinite(means)):
                warnings.warn(
                    f"One or more of the {key.split('_')[0]} scores "
                    f"are non-finite: {means}", UserWarning)
            if rank:
                if np.isnan(means).all():
                    results[f"rank_{key}"] = np.ones_like(
                        means, dtype=np.int32)
                else:
                    filled = np.nan_to_num(means,
                                           nan=np.nanmin(means) - 1)
                    results[f"rank_{key}"] = np.asarray(
                        rankdata(-filled, method="min"), dtype=np.int32)

        score_weights = n_test if self.iid else None
        _store("fit_time", fit_times)
        _store("score_time", score_times)
        for mname in (metric_names or ["score"]):
            _store(f"test_{mname}",
                   _metric_matrix(lambda r: r.test_score, mname),
                   splits=True, rank=True, weights=score_weights)
            if self.return_train_score:
                _store(f"train_{mname}",
                       _metric_matrix(lambda r: r.train_score, mname),
                       splits=True)

        # masked param_* arrays: mask=True where a candidate lacks that
        # param (heterogeneous grids) -- reference grid_search.py ~L270 [R].
        # dtype semantics follow the sklearn contract: inferred from the
        # present values; strings and ragged sequences fall back to object.
        param_values: Dict[str, dict] = defaultdict(dict)
        for cand_i, params in enumerate(candidates):
            for name, value in params.items():
                param_values[f"param_{name}"][cand_i] = value
        for key, present in param_values.items():
            plist = list(present.values())
            try:
                arr = np.array(plist)
                dtype = (arr.dtype if arr.dtype.kind != "U" and
                         arr.ndim == 1 else object)
            except ValueError:
                dtype = object
            ma = MaskedArray(np.empty(n_candidates, dtype=dtype), mask=True)
            for i, v in present.items():
                ma[i] = v
            results[key] = ma
        results["params"] = list(candidates)
        return results

    # -- post-fit surface delegated to best_estimator_ ---------------------
    # available_if: the method exists on the search object only when it
    # exists on the (fitted or prototype) estimator, and refit=False makes
    # access raise AttributeError with sklearn's exact message chain.
    def _check_refit_for(self, attr):
        if not self.refit:
            raise AttributeError(
                f"This {type(self).__name__} instance was initialized with "
                f"`refit=False`. {attr} is available only after refitting "
                "on the best parameters. You can refit an estimator "
                "manually using the `best_params_` attribute")

    @available_if(_search_estimator_has("predict"))
    def predict(self, X):
        return self.best_estimator_.predict(X)

    @available_if(_search_estimator_has("predict_proba"))
    def predict_proba(self, X):
        return self.best_estimator_.predict_proba(X)

    @available_if(_search_estimator_has("predict_log_proba"))
    def predict_log_proba(self, X):
        return self.best_estimator_.predict_log_proba(X)

    @available_if(_search_estimator_has("decision_function"))
    def decision_function(self, X):
        return self.best_estimator_.decision_function(X)

    @available_if(_search_estimator_has("transform"))
    def transform(self, X):
        return self.best_estimator_.transform(X)

    @available_if(_search_estimator_has("inverse_transform"))
    def inverse_transform(self, X):
        return self.best_estimator_.inverse_transform(X)

    def score(self, X, y=None):
        if hasattr(self, "scorer_") and self.scorer_ is not None and \
                hasattr(self, "best_estimator_"):
            if getattr(self, "multimetric_", False) and \
                    isinstance(self.refit, str):
                if isinstance(self.scorer_, dict):
                    return self.scorer_[self.refit](
                        self.best_estimator_, X, y)
                # plain callable that returns a dict of scores
                return self.scorer_(self.best_estimator_, X, y)[self.refit]
            return self.scorer_(self.best_estimator_, X, y)
        self._check_refit_for("score")
        return self.best_estimator_.score(X, y)

    @property
    def classes_(self):
        self._check_refit_for("classes_")
        return self.best_estimator_.classes_

    @property
    def n_features_in_(self):
        if not hasattr(self, "best_estimator_"):
            raise AttributeError(
                f"{type(self).__name__} object has no n_features_in_ "
                "attribute (call fit first)")
        return self.best_estimator_.n_features_in_

    def _sk_visual_block_(self):
        """sklearn HTML-diagram block: show best_estimator_ when fitted
        with refit, else the prototype estimator (sklearn's exact block)."""
        from sklearn.utils._estimator_html_repr import _VisualBlock

        if hasattr(self, "best_estimator_"):
            key, estimator = "best_estimator_", self.best_estimator_
        else:
            key, estimator = "estimator", self.estimator
        return _VisualBlock(
            "parallel", [estimator],
            names=[f"{key}: {estimator.__class__.__name__}"],
            name_details=[str(estimator)])

    def __sklearn_tags__(self):
        tags = super().__sklearn_tags__()
        try:
            from sklearn.utils import get_tags

            sub = get_tags(self.estimator)
            tags.input_tags.pairwise = sub.input_tags.pairwise
            tags.estimator_type = sub.estimator_type
        except Exception:
            pass
        return tags
