"""Param/Params system mirroring the pyspark.ml.param API surface used
by the reference (reference sparkdl/xgboost/xgboost.py:33-39)."""

import copy


class Param:
    """A typed parameter with self-contained documentation."""

    def __init__(self, parent, name, doc, typeConverter=None):
        self.parent = parent
        self.name = name
        self.doc = doc
        self.typeConverter = typeConverter

    def __repr__(self):
        return "Param(%s)" % self.name

    def __hash__(self):
        return hash(self.name)

    def __eq__(self, other):
        return isinstance(other, Param) and self.name == other.name


class TypeConverters:
    @staticmethod
    def toInt(value):
        if isinstance(value, bool):
            raise TypeError("expected int, got bool")
        return int(value)

    @staticmethod
    def toFloat(value):
        return float(value)

    @staticmethod
    def toBoolean(value):
        if not isinstance(value, bool):
            raise TypeError("expected bool, got %r" % (value,))
        return value

    @staticmethod
    def toString(value):
        return str(value)


class Params:
    """Base class holding Param values (default + user-set maps)."""

    _dummy_token = object()

    @classmethod
    def _dummy(cls):
        return cls._dummy_token

    def __init__(self):
        self._paramMap = {}
        self._defaultParamMap = {}

    # -- declaration helpers -------------------------------------------
    @property
    def params(self):
        out = []
        for name in dir(type(self)):
            v = getattr(type(self), name, None)
            if isinstance(v, Param):
                out.append(v)
        return sorted(out, key=lambda p: p.name)

    def hasParam(self, name):
        return any(p.name == name for p in self.params)

    def getParam(self, name):
        for p in self.params:
            if p.name == name:
                return p
        raise AttributeError("no param %r" % name)

    # -- set/get -------------------------------------------------------
    def _set(self, **kwargs):
        for name, value in kwargs.items():
            p = self.getParam(name)
            if p.typeConverter is not None and value is not None:
                value = p.typeConverter(value)
            self._paramMap[p] = value
        return self

    def _setDefault(self, **kwargs):
        for name, value in kwargs.items():
            self._defaultParamMap[self.getParam(name)] = value
        return self

    def set(self, param, value):
        return self._set(**{param.name: value})

    def isSet(self, param):
        if isinstance(param, str):
            param = self.getParam(param)
        return param in self._paramMap

    def isDefined(self, param):
        if isinstance(param, str):
            param = self.getParam(param)
        return param in self._paramMap or param in self._defaultParamMap

    def getOrDefault(self, param):
        if isinstance(param, str):
            param = self.getParam(param)
        if param in self._paramMap:
            return self._paramMap[param]
        return self._defaultParamMap[param]

    def extractParamMap(self):
        m = dict(self._defaultParamMap)
        m.update(self._paramMap)
        return m

    def copy(self, extra=None):
        that = copy.deepcopy(self)
        if extra:
            that._paramMap.update(extra)
        return that

    def explainParam(self, param):
        """One-line description of a Param: name, doc, default and the
        currently set value (pyspark.ml API surface)."""
        if isinstance(param, str):
            param = self.getParam(param)
        default = self._defaultParamMap.get(param, "undefined")
        current = self._paramMap.get(param, "undefined")
        return "%s: %s (default: %s, current: %s)" % (
            param.name, param.doc, default, current)

    def explainParams(self):
        """All Params, one explainParam line each."""
        return "\n".join(self.explainParam(p) for p in self.params)


def _col_param(name, doc):
    return Param(Params._dummy(), name, doc,
                 typeConverter=TypeConverters.toString)


class HasFeaturesCol(Params):
    featuresCol = _col_param("featuresCol", "features column name.")

    def __init__(self):
        super().__init__()
        self._setDefault(featuresCol="features")

    def getFeaturesCol(self):
        return self.getOrDefault(self.featuresCol)

    def setFeaturesCol(self, value):
        return self._set(featuresCol=value)


class HasLabelCol(Params):
    labelCol = _col_param("labelCol", "label column name.")

    def __init__(self):
        super().__init__()
        self._setDefault(labelCol="label")

    def getLabelCol(self):
        return self.getOrDefault(self.labelCol)

    def setLabelCol(self, value):
        return self._set(labelCol=value)


class HasWeightCol(Params):
    weightCol = _col_param("weightCol", "weight column name.")

    def getWeightCol(self):
        return self.getOrDefault(self.weightCol)

    def setWeightCol(self, value):
        return self._set(weightCol=value)


class HasPredictionCol(Params):
    predictionCol = _col_param("predictionCol", "prediction column name.")

    def __init__(self):
        super().__init__()
        self._setDefault(predictionCol="prediction")

    def getPredictionCol(self):
        return self.getOrDefault(self.predictionCol)

    def setPredictionCol(self, value):
        return self._set(predictionCol=value)


class HasProbabilityCol(Params):
    probabilityCol = _col_param(
        "probabilityCol", "Column name for predicted class conditional "
        "probabilities.")

    def __init__(self):
        super().__init__()
        self._setDefault(probabilityCol="probability")

    def getProbabilityCol(self):
        return self.getOrDefault(self.probabilityCol)


class HasRawPredictionCol(Params):
    rawPredictionCol = _col_param(
        "rawPredictionCol", "raw prediction (a.k.a. confidence) column "
        "name (margins).")

    def __init__(self):
        super().__init__()
        self._setDefault(rawPredictionCol="rawPrediction")

    def getRawPredictionCol(self):
        return self.getOrDefault(self.rawPredictionCol)


class HasValidationIndicatorCol(Params):
    validationIndicatorCol = _col_param(
        "validationIndicatorCol", "name of the column that indicates "
        "whether each row is for training or for validation.")

    def getValidationIndicatorCol(self):
        return self.getOrDefault(self.validationIndicatorCol)

    def setValidationIndicatorCol(self, value):
        return self._set(validationIndicatorCol=value)
