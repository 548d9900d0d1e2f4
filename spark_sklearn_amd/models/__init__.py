"""MI355X-native estimator families (batched CDNA4 kernel fits)."""

from .gbt import GradientBoostingClassifier, GradientBoostingRegressor
from .rf import (ExtraTreesClassifier, ExtraTreesRegressor,
                 RandomForestClassifier, RandomForestRegressor)
from .kmeans import KMeans
from .linear import (LinearRegression, LinearSVC, LogisticRegression, Ridge,
                     SGDClassifier)

__all__ = ["LogisticRegression", "LinearSVC", "SGDClassifier",
           "LinearRegression", "Ridge", "KMeans",
           "GradientBoostingClassifier", "GradientBoostingRegressor",
           "RandomForestClassifier", "RandomForestRegressor",
           "ExtraTreesClassifier", "ExtraTreesRegressor"]
