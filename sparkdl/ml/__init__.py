"""sparkdl.ml — minimal pyspark.ml-shaped Estimator/Model/Param framework.

The reference's XGBoost API is written against pyspark.ml
(reference sparkdl/xgboost/xgboost.py:30-39: Estimator, Model, Param,
Params, TypeConverters, MLReadable, MLWritable).  pyspark is not a
dependency of this framework, so this module provides the same surface
natively, operating on pandas DataFrames instead of Spark DataFrames.
Only the API used by sparkdl.xgboost is implemented.
"""

from sparkdl.ml.param import (  # noqa: F401
    Param, Params, TypeConverters,
    HasFeaturesCol, HasLabelCol, HasWeightCol, HasPredictionCol,
    HasProbabilityCol, HasRawPredictionCol, HasValidationIndicatorCol,
)
from sparkdl.ml.base import (  # noqa: F401
    Estimator, Model, MLReadable, MLWritable, MLReader, MLWriter,
)
