"""spark_sklearn_amd: MI355X-native hyperparameter-search engine.

A brand-new, from-scratch framework with the capabilities of
databricks/spark-sklearn (SURVEY.md is the spec of record): a drop-in
``GridSearchCV`` / ``RandomizedSearchCV`` whose (param-candidate x CV-fold)
task graph is scheduled onto the 8 GPUs of one MI355X node instead of a
Spark cluster -- training matrix broadcast once into each GPU's 288 GB HBM
via RCCL over xGMI, candidate-round-robin task partition per GPU,
candidate fits batched into hand-written CDNA4 MFMA kernels, fold scores
all-reduced, refit on rank 0.

Public surface mirrors the reference's ``__init__`` exports:
GridSearchCV, RandomizedSearchCV, Converter, KeyedEstimator, KeyedModel,
SparkSklearnEstimator, gapply, CSRVectorUDT -- plus the MI355X-native
estimator families under :mod:`spark_sklearn_amd.models` and the context
helpers that replace SparkContext.
"""

from .converter import Converter
from .grid_search import GridSearchCV
from .group_apply import gapply
from .keyed_models import KeyedEstimator, KeyedModel, SparkSklearnEstimator
from .random_search import RandomizedSearchCV
from .udt import CSRVectorUDT, DeviceCSRMatrix
from .util import (HipContext, createLocalContext, enable_prometheus,
                   init_distributed)

__version__ = "0.1.0"

__all__ = [
    "GridSearchCV", "RandomizedSearchCV", "Converter",
    "KeyedEstimator", "KeyedModel", "SparkSklearnEstimator",
    "gapply", "CSRVectorUDT", "DeviceCSRMatrix",
    "HipContext", "createLocalContext", "init_distributed",
    "enable_prometheus",
]
