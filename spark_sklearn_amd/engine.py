"""Batched-fit engine: candidate batching onto the CDNA4 kernels.

The key idiomatic-GPU decision with no reference analogue (SURVEY.md
section 7.1): the reference runs 1 fit per CPU task; a lone 256-feature fit
cannot feed 256 CUs.  Here B candidates (x OVR classes / centroid blocks)
fold into ONE batched kernel launch, turning a 10k-candidate grid into
~10k/B large MFMA launches per GPU.

Execution tiers:

* GPU (``torch.cuda.is_available()``): the ``_hip_ops`` extension
  (ops/csrc/*.hip, gfx950 MFMA kernels).  The extension is REQUIRED on a
  GPU box -- a missing .so raises instead of silently falling back, so GPU
  test runs genuinely exercise the native path.
* CPU: the numpy reference solvers (models/_solver_ref.py, _kmeans_ref.py),
  which define the numerics the kernels must match.

Estimator families plug in through a small adapter protocol (``FAMILIES``):
encode targets, build a batch plan, create host/device executors, map
sklearn scorers to device-computable metrics, and install fitted parameters
back onto estimator objects.  ``run_batched_tasks`` itself is family-
agnostic: fold grouping, candidate chunking, timing attribution and the
error_score contract are shared.
"""

from __future__ import annotations

import os
import time
import warnings
from typing import Dict, List, Optional

import numpy as np

from sklearn.base import clone, is_classifier

from .base_search import Task, TaskResult

__all__ = ["run_batched_tasks", "fit_single", "FAMILIES"]

DEFAULT_COLUMN_BATCH = 256   # columns per kernel launch


def _device_or_none(device):
    """Resolve execution device; native extension mandatory on GPU."""
    try:
        import torch

        if device is not None and getattr(device, "type", None) == "cuda":
            return device
        if device is None and torch.cuda.is_available():
            return torch.device("cuda", torch.cuda.current_device())
    except ImportError:
        pass
    return None


_exec_cache: Dict = {}


class LinearFamily:
    """Adapter for the linear estimator family (logistic/hinge/squared)."""

    name = "linear"

    @staticmethod
    def encode_targets(estimator, y):
        y = np.asarray(y)
        if is_classifier(estimator):
            classes, y_enc = np.unique(y, return_inverse=True)
            return classes, y_enc.astype(np.int32)
        return None, np.asarray(y, dtype=np.float32)

    @staticmethod
    def cols_per_cand(n_classes):
        return n_classes if n_classes > 2 else 1

    @staticmethod
    def cand_chunk(estimator, params_list, column_batch):
        # resolved against the label count at run time by the caller
        return None

    @staticmethod
    def make_executor(X, y_enc, classes, device):
        if device is not None:
            # executors hold device-resident state (y upload, msq, fold
            # index caches): re-creating one per run_batched_tasks call
            # costs milliseconds -- cache per (X, y_enc, device).  The
            # entry HOLDS both arrays (id() is unsound after GC).
            key = (id(X), id(y_enc), str(device))
            hit = _exec_cache.get(key)
            if hit is not None and hit[0] is X and hit[1] is y_enc:
                return hit[2]
            ex = LinearFamily._new_device_executor(X, y_enc, classes,
                                                   device)
            if len(_exec_cache) > 4:
                _exec_cache.clear()
            _exec_cache[key] = (X, y_enc, ex)
            return ex
        return _HostLinearExecutor(X, y_enc, classes)

    @staticmethod
    def _new_device_executor(X, y_enc, classes, device):
        import scipy.sparse as sp

        if sp.issparse(X):
            from .ops import sparse_executor as se

            density = X.nnz / max(1, X.shape[0] * X.shape[1])
            # dense X + transpose, bf16
            dense_bytes = 2.0 * X.shape[0] * X.shape[1] * 2
            if density <= se.CSR_DENSITY_LIMIT and \
                    dense_bytes > se.CSR_MIN_DENSE_BYTES:
                # CSR-resident K5 kernels: never densified.  Chosen only
                # when the dense form would strain HBM -- the measured
                # crossover (profiles/r02_csr_crossover.md) shows dense-
                # ingest is faster whenever it fits.
                return se.SparseLinearExecutor(X, y_enc, classes, device)
        from .ops import DeviceLinearExecutor

        return DeviceLinearExecutor(X, y_enc, classes, device)

    @staticmethod
    def build_plan(estimator, params_list, n_samples, executor, n_classes,
                   error_score, msq_scale=1.0):
        from .models._solver_ref import BatchPlan

        cols_per_cand = LinearFamily.cols_per_cand(n_classes)
        specs, bad = [], {}
        # one prototype + dict merge per candidate (sklearn clone costs
        # ~30us each: seconds of host overhead at 10k candidates)
        proto = clone(estimator)
        base_params = proto.get_params()
        # msq_scale > 1 under sample_weight: the auto-lr curvature bound
        # becomes c * max(w) * E[||x||^2] (w is mean-1 normalized), a safe
        # upper bound computable identically on host and device tiers
        eff_msq = executor.msq * msq_scale
        for slot, params in enumerate(params_list):
            try:
                unknown = set(params) - set(base_params)
                if unknown:
                    raise ValueError(
                        f"invalid parameters {sorted(unknown)} for "
                        f"{type(proto).__name__}")
                proto.__dict__.update(base_params)
                proto.__dict__.update(params)
                spec = proto._column_spec(n_samples, eff_msq)
            except Exception as exc:  # noqa: BLE001 - error_score contract
                if error_score == "raise":
                    raise
                bad[slot] = exc
                spec = dict(loss=0, lr=0.0, l1=0.0, l2=0.0, momentum=0.0,
                            tol=0.0, max_iter=0, fit_intercept=0)
            specs.append(spec)

        B = len(specs) * cols_per_cand
        plan = BatchPlan(
            loss=np.empty(B, np.int32), lr=np.empty(B, np.float32),
            l1=np.empty(B, np.float32), l2=np.empty(B, np.float32),
            momentum=np.empty(B, np.float32), tol=np.empty(B, np.float32),
            max_iter=np.empty(B, np.int32),
            target_class=np.empty(B, np.int32),
            col_to_cand=np.empty(B, np.int32), n_classes=cols_per_cand,
            fit_intercept=np.empty(B, np.uint8))
        for slot, spec in enumerate(specs):
            for c in range(cols_per_cand):
                j = slot * cols_per_cand + c
                plan.loss[j] = spec["loss"]
                plan.lr[j] = spec["lr"]
                plan.l1[j] = spec["l1"]
                plan.l2[j] = spec["l2"]
                plan.momentum[j] = spec["momentum"]
                plan.tol[j] = spec["tol"]
                plan.max_iter[j] = spec["max_iter"]
                plan.fit_intercept[j] = spec["fit_intercept"]
                if n_classes == 0:
                    plan.target_class[j] = -1
                elif cols_per_cand == 1:
                    plan.target_class[j] = 1
                else:
                    plan.target_class[j] = c
                plan.col_to_cand[j] = slot
        return plan, bad

    @staticmethod
    def metric_for(scorer, estimator) -> Optional[str]:
        if scorer is None or type(scorer).__name__ == "_PassthroughScorer":
            return "accuracy" if is_classifier(estimator) else "r2"
        fn = getattr(scorer, "_score_func", None)
        name = getattr(fn, "__name__", "")
        sign = getattr(scorer, "_sign", 1)
        if name == "accuracy_score":
            return "accuracy"
        if name == "r2_score":
            return "r2"
        if name == "log_loss" and sign == -1:
            return "neg_log_loss"
        return None

    @staticmethod
    def install_fitted(est, fitted, slot, plan, classes):
        W, b = fitted
        C = plan.n_classes
        cols = slice(slot * C, (slot + 1) * C)
        est._finalize_fit(np.ascontiguousarray(W[:, cols]),
                          np.ascontiguousarray(b[cols]), None, classes)
        return est


class _HostLinearExecutor:
    """Numpy/scipy execution of a linear candidate batch (reference
    semantics); accepts dense arrays or scipy CSR matrices."""

    def __init__(self, X, y_enc, classes):
        import scipy.sparse as sp

        if sp.issparse(X):
            self.X = sp.csr_matrix(X, dtype=np.float32)
            sq = self.X.multiply(self.X).sum(axis=1)
            self.msq = float(np.mean(np.asarray(sq)))
        else:
            self.X = np.ascontiguousarray(np.asarray(X), dtype=np.float32)
            self.msq = float(np.mean(np.sum(self.X.astype(np.float64) ** 2,
                                            axis=1)))
        self.y = y_enc
        self.classes = classes

    supports_sample_weight = True

    def fit(self, train_idx, plan, sample_weight=None):
        from .models._solver_ref import linear_fit_ref

        sw = (np.asarray(sample_weight, np.float32)[train_idx]
              if sample_weight is not None else None)
        W, b, n_iter, loss = linear_fit_ref(
            self.X[train_idx], self.y[train_idx], plan, sample_weight=sw)
        return (W, b), n_iter, loss

    def scores(self, idx, fitted, plan, metric):
        from .models._solver_ref import linear_scores_ref

        W, b = fitted
        # labels already encoded -> scoring in encoded space
        return linear_scores_ref(self.X[idx], self.y[idx], W, b, plan,
                                 metric=metric, classes=None)


def get_family(estimator):
    fam = getattr(type(estimator), "_engine_family", None)
    if fam is None:
        raise TypeError(
            f"{type(estimator).__name__} is not a batched-engine family")
    return FAMILIES[fam]


# --------------------------------------------------------------------------
# the generic batched task runner
# --------------------------------------------------------------------------

_encode_cache: Dict = {}


def _encode_cached(family, estimator, y):
    """Label encoding is O(n log n) (np.unique sort) on the full target
    vector -- cache it per (family, y, classifier-ness) so repeated
    run_batched_tasks calls on the same data (every bench step, every
    halving rung) don't re-sort 1M labels.  Entries HOLD y: id() alone is
    unsound after GC recycling."""
    key = (family.name, id(y), bool(is_classifier(estimator)))
    hit = _encode_cache.get(key)
    if hit is not None and hit[0] is y:
        return hit[1]
    out = family.encode_targets(estimator, y)
    if len(_encode_cache) > 8:
        _encode_cache.clear()
    _encode_cache[key] = (y, out)
    return out


def _route_fit_params(fit_params, executor, X):
    """Split fit_params into (normalized sample_weight, msq_scale,
    ignored dict).  sample_weight is honored when the executor supports
    it (linear family, dense device + host tiers): validated, normalized
    to MEAN 1 over the full dataset (so the (1/n)-scaled objective keeps
    its magnitude and the auto-lr bound only needs max(w)), float32."""
    if not fit_params:
        return None, 1.0, {}
    ignored = {k: v for k, v in fit_params.items() if k != "sample_weight"}
    sw = fit_params.get("sample_weight")
    if sw is None:
        return None, 1.0, ignored
    if not getattr(executor, "supports_sample_weight", False):
        ignored["sample_weight"] = sw
        return None, 1.0, ignored
    n_rows = X.shape[0] if hasattr(X, "shape") else len(X)
    sw = np.asarray(sw, dtype=np.float64).reshape(-1)
    if sw.shape[0] != n_rows:
        raise ValueError(
            f"sample_weight has {sw.shape[0]} entries for {n_rows} rows")
    if not np.isfinite(sw).all() or (sw < 0).any():
        raise ValueError("sample_weight must be finite and non-negative")
    tot = float(sw.sum())
    if tot <= 0:
        raise ValueError("sample_weight sums to zero")
    w = (sw * (n_rows / tot)).astype(np.float32)
    return w, float(w.max()), ignored


def _class_weight_rows(cw, classes, y_enc, train_idx):
    """Per-row multiplier from class_weight.  'balanced' uses the TRAIN
    FOLD's class frequencies (n_fold / (K * count_c) -- sklearn computes
    it on the data each fit sees); a dict maps ORIGINAL class labels,
    missing classes default to 1.0."""
    K = len(classes)
    yi = np.asarray(y_enc).astype(np.int64)
    if isinstance(cw, str):
        if cw != "balanced":
            raise ValueError(
                f"class_weight must be None, 'balanced' or a dict, "
                f"got {cw!r}")
        cnt = np.bincount(yi[train_idx], minlength=K)
        nt = len(train_idx)
        wmap = np.where(cnt > 0, nt / (K * np.maximum(cnt, 1)), 1.0)
    elif isinstance(cw, dict):
        wmap = np.ones(K)
        for k, v in cw.items():
            hits = np.flatnonzero(classes == k)
            if len(hits) == 0:
                raise ValueError(
                    f"class_weight key {k!r} is not a class label")
            wmap[hits[0]] = float(v)
    else:
        raise ValueError(
            f"class_weight must be None, 'balanced' or a dict, got "
            f"{type(cw).__name__}")
    return wmap[yi]


def _cw_key(v):
    """Hashable identity for a class_weight value (dicts normalized)."""
    return repr(sorted(v.items())) if isinstance(v, dict) else repr(v)


def _uniform_class_weight(estimator, tasks, classes, executor):
    """Resolve one candidate batch's class_weight (constructor value,
    optionally overridden by the candidates -- run_batched_tasks groups
    tasks so each batch is uniform: one weight vector per kernel
    batch)."""
    base = getattr(estimator, "class_weight", None)
    seen = {_cw_key(t.params.get("class_weight", base)) for t in tasks}
    if len(seen) > 1:
        # callers group tasks by class_weight before reaching here
        raise ValueError(
            "internal: mixed class_weight values in one candidate batch")
    cw = (tasks[0].params.get("class_weight", base) if tasks else base)
    if cw is None:
        return None
    if classes is None:
        # regressor grids naming class_weight fall through to the
        # build_plan unknown-parameter path (error_score contract)
        return None
    if not getattr(executor, "supports_sample_weight", False):
        warnings.warn(
            "class_weight is IGNORED by this executor tier (fits run "
            "unweighted)", stacklevel=3)
        return None
    return cw


def _fold_weights(sample_weight, msq_scale, cw, classes, y_enc, train):
    """Effective per-row weights for ONE fold: user sample_weight (mean-1
    normalized) times the class_weight row multiplier, renormalized to
    mean 1 over the fold's train rows."""
    if cw is None:
        return sample_weight, msq_scale
    wrow = _class_weight_rows(cw, classes, y_enc, train)
    eff = wrow if sample_weight is None else sample_weight * wrow
    tr = eff[train]
    tot = float(tr.sum())
    if tot <= 0:
        raise ValueError("class_weight zeroes out every training row")
    eff = (eff * (len(train) / tot)).astype(np.float32)
    return eff, float(eff[train].max())


def run_batched_tasks(estimator, X, y, tasks: List[Task], *, scorer,
                      fit_params, return_train_score, error_score, verbose,
                      device=None,
                      column_batch: int = DEFAULT_COLUMN_BATCH,
                      progress=None) -> List[TaskResult]:
    """Execute (candidate, fold) tasks through batched kernel launches.

    Tasks are grouped by fold (all candidates of a fold share the same
    train/test row sets, hence the same resident data view); candidates
    within a fold are chunked to ``column_batch`` columns per launch.
    """
    family = get_family(estimator)
    classes, y_enc = _encode_cached(family, estimator, y)
    n_classes = len(classes) if classes is not None else 0
    device = _device_or_none(device)
    executor = family.make_executor(X, y_enc, classes, device)
    # per-candidate class_weight grids: one weight vector feeds a kernel
    # batch, so tasks are GROUPED by class_weight value and each group
    # runs as its own batch sequence (executors/binning are cached, so
    # the split costs only smaller batches)
    base_cw = getattr(estimator, "class_weight", None)
    if classes is not None and tasks:
        keys = [_cw_key(t.params.get("class_weight", base_cw))
                for t in tasks]
        if len(set(keys)) > 1:
            groups: Dict[str, List[Task]] = {}
            for k, t in zip(keys, tasks):
                groups.setdefault(k, []).append(t)
            out: List[TaskResult] = []
            done_base = 0
            for g in groups.values():
                def _prog(done, total, _b=done_base):
                    if progress is not None:
                        progress(_b + done, len(tasks))
                out.extend(run_batched_tasks(
                    estimator, X, y, g, scorer=scorer,
                    fit_params=fit_params,
                    return_train_score=return_train_score,
                    error_score=error_score, verbose=verbose,
                    device=device, column_batch=column_batch,
                    progress=_prog if progress is not None else None))
                done_base += len(g)
            return out
    sample_weight, msq_scale, ignored_fp = _route_fit_params(
        fit_params, executor, X)
    class_weight = _uniform_class_weight(estimator, tasks, classes,
                                         executor)
    if ignored_fp:
        warnings.warn(
            "fit_params "
            f"({', '.join(sorted(ignored_fp))}) are IGNORED by the batched "
            "kernel path: fits run unweighted/unmodified, so results "
            "differ from an estimator that honors them (sample_weight is "
            "honored by every batched family on the dense tiers; other "
            "fit_params are a roadmap item -- see ROADMAP.md)",
            stacklevel=2)
    metric = family.metric_for(scorer, estimator)
    # multimetric fast path: when EVERY metric of a _MultiScorer maps to a
    # device metric, score each on-device instead of the host fallback
    multi_metrics = None
    if metric is None:
        from .base_search import _MultiScorer

        if isinstance(scorer, _MultiScorer):
            mm = {name: family.metric_for(sc, estimator)
                  for name, sc in scorer.scorers.items()}
            if all(v is not None for v in mm.values()):
                multi_metrics = mm

    by_fold: Dict[int, List[Task]] = {}
    for t in tasks:
        by_fold.setdefault(t.fold_index, []).append(t)

    from .ops.device_executor import DeviceLinearExecutor as _DLE

    if family is LinearFamily and device is not None and \
            isinstance(executor, _DLE) and \
            metric is not None and not fit_params and \
            class_weight is None:
        fast = _linear_device_search(
            estimator, executor, by_fold, n_classes, metric,
            return_train_score, error_score, verbose)
        if fast is not None:
            if progress is not None:
                try:
                    progress(len(fast), len(tasks))
                except Exception:  # noqa: BLE001 - observability only
                    pass
            return fast

    cols_per_cand = family.cols_per_cand(n_classes)
    cand_chunk = family.cand_chunk(estimator, [t.params for t in tasks],
                                   column_batch)
    if cand_chunk is None:
        cand_chunk = max(1, column_batch // max(cols_per_cand, 1))
    results: List[TaskResult] = []

    for fold_index in sorted(by_fold):
        fold_tasks = by_fold[fold_index]
        train, test = fold_tasks[0].train, fold_tasks[0].test
        fold_sw, fold_msq = _fold_weights(sample_weight, msq_scale,
                                          class_weight, classes, y_enc,
                                          train)
        for start in range(0, len(fold_tasks), cand_chunk):
            chunk = fold_tasks[start:start + cand_chunk]
            params_list = [t.params for t in chunk]
            t0 = time.perf_counter()
            if fold_msq != 1.0 and family is LinearFamily:
                plan, bad = family.build_plan(
                    estimator, params_list, len(train), executor,
                    n_classes, error_score, msq_scale=fold_msq)
            else:
                plan, bad = family.build_plan(
                    estimator, params_list, len(train), executor,
                    n_classes, error_score)
            fit_kw = ({"sample_weight": fold_sw}
                      if fold_sw is not None else {})
            try:
                fitted, n_iter, _ = executor.fit(train, plan, **fit_kw)
            except RuntimeError as exc:
                # per-chunk retry on a transient HIP error (SURVEY.md
                # section 5.3a); a second failure aborts the job
                if "HIP" not in str(exc) and "hip" not in str(exc):
                    raise
                warnings.warn(
                    f"retrying candidate chunk after HIP error: {exc}",
                    RuntimeWarning, stacklevel=2)
                fitted, n_iter, _ = executor.fit(train, plan, **fit_kw)
            fit_time = time.perf_counter() - t0

            t1 = time.perf_counter()
            if metric is not None:
                test_scores = executor.scores(test, fitted, plan, metric)
                train_scores = (executor.scores(train, fitted, plan, metric)
                                if return_train_score else
                                np.full(len(chunk), np.nan))
            elif multi_metrics is not None:
                per_t = {name: executor.scores(test, fitted, plan, m)
                         for name, m in multi_metrics.items()}
                test_scores = [
                    {name: float(per_t[name][i]) for name in per_t}
                    for i in range(len(chunk))]
                if return_train_score:
                    per_r = {name: executor.scores(train, fitted, plan, m)
                             for name, m in multi_metrics.items()}
                    train_scores = [
                        {name: float(per_r[name][i]) for name in per_r}
                        for i in range(len(chunk))]
                else:
                    train_scores = [float("nan")] * len(chunk)
            else:
                test_scores, train_scores = _host_score_fallback(
                    family, estimator, X, y, fitted, plan, chunk, scorer,
                    classes, train, test, return_train_score)
            score_time = time.perf_counter() - t1

            per_fit = fit_time / len(chunk)
            per_score = score_time / len(chunk)
            if progress is not None:
                try:
                    progress(len(results) + len(chunk), len(tasks))
                except Exception:  # noqa: BLE001 - observability only
                    pass
            for slot, task in enumerate(chunk):
                err = None
                if slot in bad:
                    if not isinstance(error_score, (int, float)):
                        raise bad[slot]
                    ts = tr = float(error_score)
                    err = (f"{type(bad[slot]).__name__}: {bad[slot]}\n"
                           f"(candidate {task.params!r} failed plan "
                           f"validation)")
                else:
                    from .base_search import _coerce_score

                    ts = _coerce_score(test_scores[slot])
                    tr = _coerce_score(train_scores[slot]) \
                        if isinstance(train_scores[slot], dict) \
                        else float(train_scores[slot])
                results.append(TaskResult(task.index, ts, tr,
                                          len(task.test), per_fit,
                                          per_score, error=err))
        if verbose > 1:
            print(f"[engine] fold {fold_index}: "
                  f"{len(fold_tasks)} candidates done")
    return results


def _linear_device_search(estimator, executor, by_fold, n_classes, metric,
                          return_train_score, error_score, verbose):
    """Single native call for the whole (fold x chunk) loop: the C++
    engine core (ops/csrc linear_search_run).  Returns None when the
    search shape doesn't fit the fast path (odd class counts, mismatched
    fold candidate lists), in which case the generic per-chunk loop runs.
    """
    import torch

    from .models._solver_ref import BatchPlan

    C = LinearFamily.cols_per_cand(n_classes)
    if C > 256 or 256 % C != 0:
        return None
    if executor.Fp % 128 != 0:
        return None      # the native runner needs wide feature tiles
    folds = sorted(by_fold)
    first = by_fold[folds[0]]
    n_cand = len(first)
    for f in folds[1:]:
        if len(by_fold[f]) != n_cand or any(
                a.cand_index != b.cand_index
                for a, b in zip(first, by_fold[f])):
            return None
    # the whole-search plan encodes per-sample scales (1/(C*n_train)) from
    # fold 0; for KFold-style splits fold sizes differ by at most one row
    # (O(1/n) scale deviation, documented divergence), but splitters with
    # materially unequal train folds fall back to the per-fold generic
    # path, which rebuilds the plan per fold.
    sizes = [len(by_fold[f][0].train) for f in folds]
    if max(sizes) > min(sizes) * 1.001:
        return None

    dbg = os.environ.get("SKA_TIME") == "1"
    if dbg:
        torch.cuda.synchronize()
        _t0 = time.perf_counter()
    params_list = [t.params for t in first]
    plan, bad = LinearFamily.build_plan(
        estimator, params_list, len(first[0].train), executor,
        n_classes, error_score)
    if dbg:
        _t1 = time.perf_counter()

    cand_chunk = 256 // C
    n_chunks = (n_cand + cand_chunk - 1) // cand_chunk
    Bt = n_chunks * 256

    def pad_host(src, fill, dtype):
        out = np.full(Bt, fill, dtype=dtype)
        for ci in range(n_chunks):
            lo = ci * cand_chunk
            hi = min(lo + cand_chunk, n_cand)
            seg = src[lo * C:hi * C]
            out[ci * 256:ci * 256 + len(seg)] = seg
        return out

    # two H2D copies instead of nine: float plan rows and int plan rows
    fhost = np.stack([pad_host(plan.lr, 0.0, np.float32),
                      pad_host(plan.l1, 0.0, np.float32),
                      pad_host(plan.l2, 0.0, np.float32),
                      pad_host(plan.momentum, 0.0, np.float32),
                      pad_host(plan.tol, 0.0, np.float32)])
    ihost = np.stack([pad_host(plan.loss, 0, np.int32),
                      pad_host(plan.max_iter, 0, np.int32),
                      pad_host(plan.target_class, 0, np.int32),
                      pad_host(plan.fit_intercept, 0, np.int32)])
    f_d = torch.as_tensor(fhost).to(executor.device)
    i_d = torch.as_tensor(ihost).to(executor.device)
    lr_t, l1_t, l2_t, mom_t, tol_t = f_d[0], f_d[1], f_d[2], f_d[3], f_d[4]
    loss_t, mi_t, tgt_t = i_d[0], i_d[1], i_d[2]
    ic_t = i_d[3].to(torch.uint8)

    train_ts, nts, test_ts = [], [], []
    for f in folds:
        tr = by_fold[f][0].train
        train_ts.append(executor._cached_idx(tr, 256, np.int32))
        nts.append(len(tr))
        test_ts.append(executor._cached_idx(by_fold[f][0].test, 0,
                                            np.int64))
    if return_train_score:
        for f in folds:
            test_ts.append(executor._cached_idx(by_fold[f][0].train, 0,
                                                np.int64))

    from .ops.device_executor import _METRIC_ID

    if dbg:
        torch.cuda.synchronize()
        _t2 = time.perf_counter()
    t0 = time.perf_counter()
    out = executor.ext.linear_search_run(
        executor.X, executor.Xt, executor.y, train_ts, nts, test_ts,
        loss_t, lr_t, l1_t, l2_t, mom_t, tol_t, mi_t, tgt_t, ic_t,
        C, _METRIC_ID[metric], n_cand, 1 if return_train_score else 0,
        32)
    test_sums, train_sums, fit_ms, score_ms = out[0], out[1], out[2], out[3]
    if dbg:
        torch.cuda.synchronize()
        _t3 = time.perf_counter()
    test_sums = test_sums.cpu().numpy()
    train_sums = train_sums.cpu().numpy() if return_train_score else None
    fit_ms = fit_ms.numpy()
    score_ms = score_ms.numpy()

    def finalize(sums, idx_len, idx):
        if metric in ("accuracy", "neg_log_loss"):
            return sums / idx_len
        y_sel = _y_np()[np.asarray(idx)]
        ss_tot = float(np.sum((y_sel - y_sel.mean()) ** 2))
        return 1.0 - sums / max(ss_tot, 1e-12)

    _y_cache = []

    def _y_np():
        if not _y_cache:
            _y_cache.append(executor.y.cpu().numpy())
        return _y_cache[0]

    results = []
    for fi, f in enumerate(folds):
        te = by_fold[f][0].test
        tr = by_fold[f][0].train
        ts = finalize(test_sums[fi], len(te), te)
        if return_train_score:
            trs = finalize(train_sums[fi], len(tr), tr)
        for slot, task in enumerate(by_fold[f]):
            chunk = slot // cand_chunk
            per_fit = float(fit_ms[fi][chunk]) / 1000.0 / cand_chunk
            per_score = float(score_ms[fi][chunk]) / 1000.0 / cand_chunk
            if slot in bad:
                if not isinstance(error_score, (int, float)):
                    raise bad[slot]
                v = tv = float(error_score)
                err = (f"{type(bad[slot]).__name__}: {bad[slot]}\n"
                       f"(candidate {task.params!r} failed plan "
                       f"validation)")
            else:
                v = float(ts[slot])
                tv = float(trs[slot]) if return_train_score else float("nan")
                err = None
            results.append(TaskResult(task.index, v, tv, len(task.test),
                                      per_fit, per_score, error=err))
    if dbg:
        _t4 = time.perf_counter()
        print(f"[ska-time] plan_build {1000*(_t1-_t0):.2f}ms "
              f"plan_upload {1000*(_t2-_t1):.2f}ms "
              f"native {1000*(_t3-_t2):.2f}ms "
              f"assemble {1000*(_t4-_t3):.2f}ms", flush=True)
    if verbose > 1:
        print(f"[engine] native search_run: {len(folds)} folds x "
              f"{n_chunks} chunks in {time.perf_counter() - t0:.2f}s")
    return results


def _host_score_fallback(family, estimator, X, y, fitted, plan, chunk,
                         scorer, classes, train, test, return_train_score):
    """Custom scorer: build fitted clones on the host and score each.
    Scores may be floats or (multimetric) dicts -- returned as lists."""
    X = np.asarray(X) if not hasattr(X, "cpu") else X.cpu().float().numpy()
    y = np.asarray(y) if y is not None else None
    test_scores = [np.nan] * len(chunk)
    train_scores = [np.nan] * len(chunk)
    for slot, task in enumerate(chunk):
        est = clone(estimator).set_params(**task.params)
        est = family.install_fitted(est, fitted, slot, plan, classes)
        if y is None:
            test_scores[slot] = scorer(est, X[test])
            if return_train_score:
                train_scores[slot] = scorer(est, X[train])
        else:
            test_scores[slot] = scorer(est, X[test], y[test])
            if return_train_score:
                train_scores[slot] = scorer(est, X[train], y[train])
    return test_scores, train_scores


# --------------------------------------------------------------------------
# single-estimator fit (standalone .fit(), and rank-0 refit)
# --------------------------------------------------------------------------

def fit_single(estimator, X, y, device=None, sample_weight=None):
    """Fit one estimator (its OVR classes still batch into one launch)."""
    family = get_family(estimator)
    classes, y_enc = _encode_cached(family, estimator, y)
    n_classes = len(classes) if classes is not None else 0
    device = _device_or_none(device)
    executor = family.make_executor(X, y_enc, classes, device)
    n = X.shape[0] if hasattr(X, "shape") else len(X)
    fp = {"sample_weight": sample_weight} if sample_weight is not None \
        else {}
    sw, msq_scale, ignored = _route_fit_params(fp, executor, X)
    if ignored:
        warnings.warn(
            "sample_weight is IGNORED by this executor tier (fits run "
            "unweighted)", stacklevel=2)
    all_idx = np.arange(n, dtype=np.int64)
    cw = _uniform_class_weight(estimator, [], classes, executor)
    sw, msq_scale = _fold_weights(sw, msq_scale, cw, classes, y_enc,
                                  all_idx)
    if msq_scale != 1.0 and family is LinearFamily:
        plan, _ = family.build_plan(estimator, [{}], n, executor,
                                    n_classes, "raise",
                                    msq_scale=msq_scale)
    else:
        plan, _ = family.build_plan(estimator, [{}], n, executor,
                                    n_classes, "raise")
    fit_kw = {"sample_weight": sw} if sw is not None else {}
    fitted, n_iter, _ = executor.fit(all_idx, plan, **fit_kw)
    family.install_fitted(estimator, fitted, 0, plan, classes)
    if n_iter is not None:
        estimator.n_iter_ = n_iter
    return estimator


from .models.gbt import GBTFamily  # noqa: E402
from .models.rf import RFFamily  # noqa: E402
from .models.kmeans import KMeansFamily  # noqa: E402  (no import cycle:
# model modules only import engine lazily inside methods)

FAMILIES = {"linear": LinearFamily, "kmeans": KMeansFamily,
            "gbt": GBTFamily, "rf": RFFamily}
