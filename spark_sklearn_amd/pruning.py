"""Early-stop pruning for very large randomized sweeps (BASELINE config 5).

A deliberate behavioral EXTENSION over the reference (SURVEY.md section
3.2): ``RandomizedSearchCV(halving_factor=f)`` runs successive halving over
the iteration budget -- all candidates are first fitted cheaply (a
``min_resource_frac`` fraction of ``max_iter``), the bottom (1 - 1/f) are
pruned, survivors are promoted with an f-times larger budget, until the
full ``max_iter`` rung.  With the batched engine each rung is still a
handful of wide kernel launches, so a 100k-candidate sweep costs little
more than its surviving tail.

Every candidate appears in ``cv_results_`` with the scores from the LAST
rung it reached, plus two extra keys: ``rung`` (0-based rung index) and
``resource_iter`` (the max_iter budget it was evaluated at).  ``best_*``
come from the final rung, so ``best_estimator_`` is refit at the full
budget, exactly as in the unpruned path.
"""

from __future__ import annotations

import time
from typing import List

import numpy as np

from sklearn.base import clone, is_classifier
from sklearn.metrics import check_scoring
from sklearn.model_selection import check_cv
from sklearn.utils.validation import indexable

from .base_search import Task

__all__ = ["halving_fit"]


def halving_fit(search, X, y=None, groups=None, **fit_params):
    """Successive-halving driver for a RandomizedSearchCV instance."""
    factor = float(search.halving_factor)
    if factor <= 1.0:
        raise ValueError("halving_factor must be > 1")
    estimator = search.estimator
    if isinstance(search.scoring, (dict, list, tuple, set)):
        raise ValueError(
            "halving pruning supports single-metric scoring only; run "
            "the unpruned search for multimetric sweeps")
    cv = check_cv(search.cv, y, classifier=is_classifier(estimator))
    search.scorer_ = check_scoring(estimator, scoring=search.scoring)
    X, y, groups = indexable(X, y, groups)
    splits = list(cv.split(X, y, groups))
    n_splits = len(splits)

    candidates = list(search._get_param_iterator())
    full_budget = int(estimator.get_params().get("max_iter", 100))
    # grids may sweep max_iter themselves; the rung budget must clamp the
    # candidate value (not be overridden by it), and the final rung must
    # reach the largest requested budget
    grid_mi = [int(p["max_iter"]) for p in candidates if "max_iter" in p]
    if grid_mi:
        full_budget = max(full_budget, max(grid_mi))
    budget = max(1, int(full_budget * float(search.min_resource_frac)))

    backend = search._get_backend()
    if getattr(backend, "resume_from_log", None):
        import warnings

        warnings.warn(
            "resume_from_log is ignored by the halving path (rungs are "
            "re-sharded per budget); run the unpruned search to resume",
            UserWarning, stacklevel=2)
    merged_fit_params = dict(search.fit_params or {})
    merged_fit_params.update(fit_params)

    # last-rung record per candidate index into `candidates`
    last = {}
    alive = list(range(len(candidates)))
    rung = 0
    while True:
        budget = min(budget, full_budget)
        rung_estimator = clone(estimator).set_params(max_iter=budget)
        if search.verbose > 0:
            print(f"[halving] rung {rung}: {len(alive)} candidates at "
                  f"max_iter={budget}")
        tasks: List[Task] = []
        idx = 0
        for slot, cand_i in enumerate(alive):
            params = candidates[cand_i]
            if "max_iter" in params:
                params = dict(params,
                              max_iter=min(int(params["max_iter"]), budget))
            for fold_i, (train, test) in enumerate(splits):
                tasks.append(Task(idx, slot, fold_i, params,
                                  np.asarray(train), np.asarray(test)))
                idx += 1
        results = backend.run_search(
            rung_estimator, X, y, tasks, scorer=search.scorer_,
            fit_params=merged_fit_params,
            return_train_score=search.return_train_score,
            error_score=search.error_score, verbose=search.verbose)
        by_index = {r.index: r for r in results}
        ordered = [by_index[i] for i in range(len(tasks))]

        means = np.array([
            np.mean([ordered[slot * n_splits + f].test_score
                     for f in range(n_splits)])
            for slot in range(len(alive))])
        for slot, cand_i in enumerate(alive):
            last[cand_i] = (rung, budget,
                            ordered[slot * n_splits:(slot + 1) * n_splits])

        if budget >= full_budget or len(alive) <= 1:
            break
        n_keep = max(1, int(np.ceil(len(alive) / factor)))
        keep_order = np.argsort(-means, kind="stable")[:n_keep]
        alive = [alive[int(i)] for i in sorted(keep_order)]
        budget = min(full_budget, max(budget + 1, int(budget * factor)))
        rung += 1

    # assemble cv_results_ over ALL candidates from their last rung
    flat = []
    t = 0
    for cand_i in range(len(candidates)):
        _, _, recs = last[cand_i]
        for r in recs:
            r.index = t
            flat.append(r)
            t += 1
    search.cv_results_ = search._assemble_cv_results(
        candidates, flat, len(candidates), n_splits)
    search.cv_results_["rung"] = np.array(
        [last[c][0] for c in range(len(candidates))])
    search.cv_results_["resource_iter"] = np.array(
        [last[c][1] for c in range(len(candidates))])

    # best = final-rung winner (guard: rank only among final-rung members)
    final_rung = max(last[c][0] for c in range(len(candidates)))
    final_members = [c for c in range(len(candidates))
                     if last[c][0] == final_rung]
    fr_scores = search.cv_results_["mean_test_score"][final_members]
    best_local = int(np.argmax(fr_scores))
    search.best_index_ = int(final_members[best_local])
    search.best_params_ = candidates[search.best_index_]
    search.best_score_ = float(
        search.cv_results_["mean_test_score"][search.best_index_])
    search.n_splits_ = n_splits
    search.multimetric_ = False

    if search.refit:
        best_estimator = clone(estimator).set_params(
            **{k: clone(v, safe=False)
               for k, v in search.best_params_.items()})
        t0 = time.perf_counter()
        refit_fn = getattr(backend, "refit", None)
        if refit_fn is not None:
            best_estimator = refit_fn(best_estimator, X, y,
                                      merged_fit_params)
        elif y is not None:
            best_estimator.fit(X, y, **merged_fit_params)
        else:
            best_estimator.fit(X, **merged_fit_params)
        search.refit_time_ = time.perf_counter() - t0
        search.best_estimator_ = best_estimator
    return search
