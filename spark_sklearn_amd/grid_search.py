"""Drop-in GridSearchCV scheduled onto MI355X GPUs.

Re-implements the behavior of the reference's ``GridSearchCV``
(spark-sklearn ``grid_search.py`` ~L30-330 [R]): the same
constructor surface as ``sklearn.model_selection.GridSearchCV`` plus a
leading context argument (the reference takes a ``SparkContext``; here a
:class:`spark_sklearn_amd.HipContext`, a backend name string, or ``None``
for auto-detection), the same ``fit()`` / ``cv_results_`` /
``best_estimator_`` contract.

The (param-candidate x CV-fold) task list the reference ships to Spark
executors is instead scheduled onto the GPUs of one node: the training
matrix is broadcast once into each GPU's HBM via RCCL over xGMI, per-GPU
task queues drain candidate batches through hand-written CDNA4 HIP kernels,
and fold scores come back via RCCL all-gather with refit on rank 0.
"""

from __future__ import annotations

import numpy as np

from sklearn.model_selection import ParameterGrid

from .base_search import BaseSearchCV

__all__ = ["GridSearchCV"]


class GridSearchCV(BaseSearchCV):
    """Exhaustive search over a parameter grid, distributed over MI355X GPUs.

    Parameters
    ----------
    ctx : HipContext, Backend, str or None
        Execution context.  ``None`` auto-selects: the torch.distributed
        process group if one is initialized (one process per GPU over RCCL),
        else the local GPU if present, else a CPU thread pool.  The string
        forms ``"threads"``/``"threads:N"``, ``"sequential"``, ``"dist"``
        and ``"gpu"`` force a backend.  This replaces the reference's
        leading ``sc`` (SparkContext) argument.
    estimator : estimator object
        Anything implementing the sklearn estimator interface.  Instances of
        the :mod:`spark_sklearn_amd.models` families additionally run their
        inner ``fit`` through batched CDNA4 HIP kernels.
    param_grid : dict or list of dicts
        Same semantics as sklearn ``ParameterGrid``.

    All remaining keyword arguments mirror the reference surface
    (``scoring``, ``fit_params``, ``n_jobs``, ``iid``, ``refit``, ``cv``,
    ``verbose``, ``pre_dispatch``, ``error_score``, ``return_train_score``)
    -- ``n_jobs``/``pre_dispatch`` are accepted-but-ignored, exactly as the
    reference ignores them.

    Examples
    --------
    >>> from sklearn import datasets, svm
    >>> import spark_sklearn_amd as ska
    >>> digits = datasets.load_digits()
    >>> gs = ska.GridSearchCV(None, svm.SVC(), {"C": [0.1, 1, 10]}, cv=3)
    >>> _ = gs.fit(digits.data, digits.target)
    >>> sorted(gs.cv_results_.keys())[:2]
    ['mean_fit_time', 'mean_score_time']
    """

    def __init__(self, ctx, estimator, param_grid, scoring=None,
                 fit_params=None, n_jobs=1, iid=False, refit=True, cv=None,
                 verbose=0, pre_dispatch="2*n_jobs", error_score=np.nan,
                 return_train_score=True):
        super().__init__(
            estimator=estimator, scoring=scoring, fit_params=fit_params,
            n_jobs=n_jobs, iid=iid, refit=refit, cv=cv, verbose=verbose,
            pre_dispatch=pre_dispatch, error_score=error_score,
            return_train_score=return_train_score)
        self.ctx = ctx
        self.param_grid = param_grid

    def _get_param_iterator(self):
        # sklearn's ParameterGrid performs the exact upstream validation
        # (typed messages for scalars / empty sequences) at fit time
        return ParameterGrid(self.param_grid)
