"""sparkdl.xgboost — gradient-boosted trees with the reference's
pyspark-ML-shaped API (reference sparkdl/xgboost/__init__.py:17-23),
backed by the framework's native histogram GBT engine (CPU numpy /
CDNA4 HIP histogram kernel)."""

from sparkdl.xgboost.estimator import (  # noqa: F401
    XgboostClassifier, XgboostClassifierModel,
    XgboostRegressor, XgboostRegressorModel,
)

__all__ = ["XgboostClassifier", "XgboostClassifierModel",
           "XgboostRegressor", "XgboostRegressorModel"]
