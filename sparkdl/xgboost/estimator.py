"""PySpark-ML-shaped XGBoost estimators/models over the native GBT engine.

Re-implements the contract of reference sparkdl/xgboost/xgboost.py
(every docstring cite below is into that file): the full Param surface
(C5, :38-106), the abstract estimator/model with MLReadable/MLWritable
persistence (C6, :109-144), and the public
XgboostRegressor/XgboostClassifier pairs (C7/C8, :147-331) with the
sklearn-xgboost kwargs auto-forwarding and remaps (:176-199):
``gpu_id -> use_gpu``, ``base_margin -> baseMarginCol``,
``eval_set -> validationIndicatorCol``, ``sample_weight -> weightCol``,
``xgb_model -> get_booster()``.

Datasets are pandas DataFrames; the features column holds either
list/array cells or is complemented by ``feature_names`` columns.
"""

import numpy as np

from sparkdl.ml import (
    Estimator, Model, MLReadable, MLWritable, Param, Params,
    TypeConverters, HasFeaturesCol, HasLabelCol, HasWeightCol,
    HasPredictionCol, HasProbabilityCol, HasRawPredictionCol,
    HasValidationIndicatorCol,
)
from sparkdl.xgboost import gbt


class _XgboostParams(HasFeaturesCol, HasLabelCol, HasWeightCol,
                     HasPredictionCol, HasValidationIndicatorCol):
    """Shared Params (reference xgboost.py:38-106)."""

    missing = Param(
        parent=Params._dummy(),
        name="missing",
        doc="Feature value treated as absent (defaults to NaN). Training "
            "runs fastest when 0.0 is chosen here. Caveat for sparse "
            "input: positions a sparse vector leaves out are zeros, not "
            "absent values — they are only skipped when missing is set "
            "to 0.")

    callbacks = Param(
        parent=Params._dummy(),
        name="callbacks",
        doc="Optional list of per-round callback callables. They travel "
            "to workers via cloudpickle, so deserialization can break if "
            "the loading environment ships different library versions "
            "than the one that pickled them.")

    num_workers = Param(
        parent=Params._dummy(),
        name="num_workers",
        doc="How many data-parallel training workers to launch; one "
            "worker occupies one task slot.",
        typeConverter=TypeConverters.toInt)

    use_gpu = Param(
        parent=Params._dummy(),
        name="use_gpu",
        doc="Boolean switch routing the histogram build onto a GPU; a "
            "worker can drive at most a single device.")

    force_repartition = Param(
        parent=Params._dummy(),
        name="force_repartition",
        doc="Boolean switch; when true the input rows are redistributed "
            "across the workers even if already partitioned.")

    use_external_storage = Param(
        parent=Params._dummy(),
        name="use_external_storage",
        doc="Boolean switch (off by default) that spills the quantized "
            "feature matrix to disk so datasets larger than memory can "
            "train. Row weights and base margins are incompatible with "
            "this mode.")

    external_storage_precision = Param(
        parent=Params._dummy(),
        name="external_storage_precision",
        doc="Significant decimal digits kept when rows are spilled to "
            "disk in external-storage mode.",
        typeConverter=TypeConverters.toInt)

    baseMarginCol = Param(
        parent=Params._dummy(),
        name="baseMarginCol",
        doc="Name of a column holding each row's starting margin, used "
            "in place of the sklearn-style base_margin / "
            "base_margin_eval_set fit arguments. Unavailable when "
            "training with multiple workers.")

    def __init__(self):
        super().__init__()
        self._setDefault(missing=float("nan"), num_workers=1,
                         use_gpu=False, force_repartition=False,
                         use_external_storage=False,
                         external_storage_precision=5, callbacks=None)
        self._xgb_params = {}

    # sklearn-xgboost kwargs forwarded via **kwargs (reference :171-174)
    _RENAMES = {"reg_alpha": "reg_alpha", "n_estimators": "n_estimators"}
    _UNSUPPORTED = ("validate_features", "output_margin", "gpu_id",
                    "base_margin", "base_margin_eval_set", "eval_set",
                    "sample_weight", "sample_weight_eval_set",
                    "xgb_model")

    def _apply_kwargs(self, kwargs):
        for k, v in kwargs.items():
            if k in self._UNSUPPORTED:
                raise ValueError(
                    "Parameter %r is replaced in sparkdl.xgboost "
                    "(reference xgboost.py:176-199): use "
                    "use_gpu/baseMarginCol/validationIndicatorCol/"
                    "weightCol or get_booster() instead." % k)
            if self.hasParam(k):
                self._set(**{k: v})
            else:
                self._xgb_params[k] = v

    def _trainer_params(self):
        return dict(self._xgb_params)


class _XgboostEstimator(Estimator, _XgboostParams, MLReadable, MLWritable):
    """Abstract estimator (reference xgboost.py:109-123)."""

    def __init__(self, **kwargs):
        super().__init__()
        self._apply_kwargs(kwargs)

    # -- dataset plumbing ----------------------------------------------
    @staticmethod
    def _densify(cell):
        """Feature cell -> 1-D float64 array. Sparse rows (scipy)
        densify with explicit zeros at the inactive positions — the
        documented caveat: an absent entry of a sparse vector is the
        VALUE 0, not a missing value, so it is only treated as missing
        when the `missing` Param is set to 0 (which then routes those
        zeros to the learned default direction)."""
        try:
            import scipy.sparse as sp
            if sp.issparse(cell):
                return np.asarray(cell.todense(),
                                  dtype=np.float64).ravel()
        except ImportError:  # pragma: no cover
            pass
        return np.asarray(cell, dtype=np.float64)

    def _extract_xy(self, dataset):
        import pandas as pd
        fc = self.getFeaturesCol()
        if fc in dataset.columns:
            X = np.asarray([self._densify(v) for v in dataset[fc]])
        else:
            # Same skip set as _XgboostModel._features: auxiliary columns
            # (weights, margins, validation mask, outputs) must not leak
            # into the feature matrix, or train/transform would disagree
            # on feature count.
            skip = {self.getLabelCol()}
            for name in ("weightCol", "baseMarginCol",
                         "validationIndicatorCol", "predictionCol",
                         "probabilityCol", "rawPredictionCol"):
                if self.hasParam(name) and self.isDefined(name):
                    skip.add(self.getOrDefault(name))
            cols = [c for c in dataset.columns if c not in skip]
            X = dataset[cols].to_numpy(dtype=np.float64)
        y = dataset[self.getLabelCol()].to_numpy(dtype=np.float64)
        w = None
        if self.isDefined("weightCol") and \
                self.getOrDefault("weightCol") in dataset.columns:
            w = dataset[self.getOrDefault("weightCol")].to_numpy(
                dtype=np.float64)
        bm = None
        if self.isDefined("baseMarginCol") and \
                self.getOrDefault("baseMarginCol") in dataset.columns:
            bm = dataset[self.getOrDefault("baseMarginCol")].to_numpy(
                dtype=np.float64)
        vmask = None
        if self.isDefined("validationIndicatorCol") and \
                self.getOrDefault("validationIndicatorCol") in dataset.columns:
            vmask = dataset[self.getOrDefault(
                "validationIndicatorCol")].to_numpy(dtype=bool)
        return X, y, w, bm, vmask

    def _objective(self):
        raise NotImplementedError

    def _resolve_objective(self, y, params):
        return self._objective()

    def _fit(self, dataset):
        X, y, w, bm, vmask = self._extract_xy(dataset)
        eval_set = None
        if vmask is not None:
            # validation rows are held out of training; they drive early
            # stopping when early_stopping_rounds is set (the reference's
            # eval_set -> validationIndicatorCol remap, xgboost.py:189-197)
            Xt, yt = X[~vmask], y[~vmask]
            wt = w[~vmask] if w is not None else None
            bmt = bm[~vmask] if bm is not None else None
            eval_set = (X[vmask], y[vmask])
        else:
            Xt, yt, wt, bmt = X, y, w, bm
        params = self._trainer_params()
        params["objective"] = self._resolve_objective(yt, params)
        missing = self.getOrDefault("missing")
        xgb_model = params.pop("booster_warm_start", None)
        num_workers = self.getOrDefault("num_workers")
        use_gpu = bool(self.getOrDefault("use_gpu"))
        ext_dir = None
        if self.getOrDefault("use_external_storage"):
            # reference xgboost.py:81-90: disk spill for huge datasets;
            # base margin and weights are unsupported with it.
            if wt is not None or bmt is not None:
                raise ValueError(
                    "use_external_storage does not support weightCol or "
                    "baseMarginCol (reference xgboost.py:86)")
            import tempfile
            ext_dir = tempfile.mkdtemp(prefix="sparkdl_xgb_ext_")
        if self.getOrDefault("force_repartition"):
            # re-shard rows before training (reference xgboost.py:72-80)
            rng = np.random.RandomState(0)
            perm = rng.permutation(len(yt))
            Xt, yt = Xt[perm], yt[perm]
            wt = wt[perm] if wt is not None else None
            bmt = bmt[perm] if bmt is not None else None
        if num_workers > 1:
            booster = _fit_distributed(
                Xt, yt, params, wt, bmt, missing, use_gpu, num_workers,
                self.getOrDefault("callbacks"), xgb_model=xgb_model,
                eval_set=eval_set)
        else:
            booster = gbt.train(
                Xt, yt, params, sample_weight=wt, base_margin=bmt,
                missing=missing, use_gpu=use_gpu,
                callbacks=self.getOrDefault("callbacks"),
                xgb_model=xgb_model, external_storage_dir=ext_dir,
                external_storage_precision=self.getOrDefault(
                    "external_storage_precision"),
                eval_set=eval_set)
        if ext_dir is not None:
            import shutil
            shutil.rmtree(ext_dir, ignore_errors=True)
        model = self._model_class()(booster=booster)
        model._paramMap = dict(self._paramMap)
        model._defaultParamMap = dict(self._defaultParamMap)
        return model

    def _model_class(self):
        raise NotImplementedError

    def _to_json_dict(self):
        import cloudpickle
        import base64
        payload = {
            "params": {p.name: self._paramMap[p] for p in self._paramMap
                       if p.name != "callbacks"},
            "xgb_params": self._xgb_params,
        }
        cbs = self.getOrDefault("callbacks")
        if cbs is not None:
            payload["callbacks_pkl"] = base64.b64encode(
                cloudpickle.dumps(cbs)).decode()
        return payload

    @classmethod
    def _from_json_dict(cls, payload):
        inst = cls()
        inst._apply_kwargs(payload.get("xgb_params", {}))
        for k, v in payload.get("params", {}).items():
            inst._set(**{k: v})
        if "callbacks_pkl" in payload:
            import cloudpickle
            import base64
            inst._set(callbacks=cloudpickle.loads(
                base64.b64decode(payload["callbacks_pkl"])))
        return inst


class _XgboostModel(Model, _XgboostParams, MLReadable, MLWritable):
    """Abstract model (reference xgboost.py:125-144)."""

    def __init__(self, booster=None, **kwargs):
        super().__init__()
        self._apply_kwargs(kwargs)
        self._booster = booster

    def get_booster(self):
        """Return the underlying Booster of this model
        (reference xgboost.py:130-134)."""
        return self._booster

    def _features(self, dataset):
        fc = self.getFeaturesCol()
        if fc in dataset.columns:
            return np.asarray([_XgboostEstimator._densify(v)
                               for v in dataset[fc]])
        # columnar fallback: everything except label/output-ish columns
        skip = {self.getLabelCol(), self.getPredictionCol()}
        for name in ("weightCol", "baseMarginCol",
                     "validationIndicatorCol", "probabilityCol",
                     "rawPredictionCol"):
            if self.hasParam(name) and self.isDefined(name):
                skip.add(self.getOrDefault(name))
        cols = [c for c in dataset.columns if c not in skip]
        return dataset[cols].to_numpy(dtype=np.float64)

    def _to_json_dict(self):
        return {
            "params": {p.name: self._paramMap[p] for p in self._paramMap
                       if p.name != "callbacks"},
            "xgb_params": self._xgb_params,
            "booster": self._booster.to_dict(),
        }

    @classmethod
    def _from_json_dict(cls, payload):
        inst = cls(booster=gbt.Booster.from_dict(payload["booster"]))
        inst._apply_kwargs(payload.get("xgb_params", {}))
        for k, v in payload.get("params", {}).items():
            inst._set(**{k: v})
        return inst


def _fit_distributed(X, y, params, w, bm, missing, use_gpu, num_workers,
                     callbacks, xgb_model=None, eval_set=None):
    """Data-parallel GBT: shard rows over a HorovodRunner gang and sum
    per-node histograms across workers each depth (the xgboost
    num_workers contract, reference xgboost.py:58-64).

    callbacks run on rank 0 only; warm start (xgb_model) and the
    early-stopping validation split are shipped whole to every worker
    (validation margins come from the identical synchronized trees, so
    all ranks stop at the same round).
    """
    from sparkdl import HorovodRunner

    # Shared quantile bins fitted on the FULL dataset: per-shard edges
    # would make summed histograms refer to different boundaries. A
    # warm-start booster already carries its binner; reuse that one.
    from sparkdl.xgboost.gbt import Binner, MAX_BINS
    if xgb_model is not None:
        binner = xgb_model.binner
    else:
        binner = Binner(params.get("max_bins") or MAX_BINS).fit(X, missing)

    def worker_main(X, y, params, w, bm, missing, use_gpu, binner_dict,
                    callbacks, xgb_model, eval_set):
        import numpy as _np
        import torch
        import torch.distributed as dist
        import sparkdl.torch as hvd
        from sparkdl.xgboost import gbt as _gbt
        hvd.init()
        rank, size = hvd.rank(), hvd.size()
        shard = slice(rank, None, size)

        def allreduce_hist(hist):
            t = torch.from_numpy(_np.ascontiguousarray(hist))
            dist.all_reduce(t, op=dist.ReduceOp.SUM)
            return t.numpy()

        booster = _gbt.train(
            X[shard], y[shard], params,
            sample_weight=w[shard] if w is not None else None,
            base_margin=bm[shard] if bm is not None else None,
            missing=missing, use_gpu=use_gpu,
            callbacks=callbacks if rank == 0 else None,
            xgb_model=xgb_model,
            comm=allreduce_hist,
            binner=_gbt.Binner.from_dict(binner_dict),
            eval_set=eval_set)
        return booster if rank == 0 else None

    hr = HorovodRunner(np=-num_workers, driver_log_verbosity="log_callback_only")
    return hr.run(worker_main, X=X, y=y, params=params, w=w, bm=bm,
                  missing=missing, use_gpu=use_gpu,
                  binner_dict=binner.to_dict(), callbacks=callbacks,
                  xgb_model=xgb_model, eval_set=eval_set)


# ---------------------------------------------------------------------------
# Public classes (reference xgboost.py:147-331)
# ---------------------------------------------------------------------------

class XgboostRegressor(_XgboostEstimator):
    """XgboostRegressor is a pyspark-ML-shaped estimator API for the
    native GBT regressor (reference xgboost.py:147-240).

    Most xgboost-sklearn estimator parameters pass straight through as
    keyword arguments (``n_estimators``, ``max_depth``,
    ``learning_rate``, ``reg_lambda``, ...).

    .. Note:: ``gpu_id`` is replaced by ``use_gpu``; ``base_margin`` by
      ``baseMarginCol``; ``eval_set`` by ``validationIndicatorCol``;
      ``sample_weight`` by ``weightCol``; ``xgb_model`` by passing the
      booster from ``model.get_booster()``.

    >>> import pandas as pd, numpy as np
    >>> df = pd.DataFrame({"features": list(np.random.rand(32, 4)),
    ...                    "label": np.random.rand(32)})
    >>> xgb = XgboostRegressor(n_estimators=10, missing=0.0)
    >>> model = xgb.fit(df)
    >>> out = model.transform(df)
    """

    def _objective(self):
        return "reg:squarederror"

    def _model_class(self):
        return XgboostRegressorModel


class XgboostRegressorModel(_XgboostModel):
    """Model fitted by :class:`XgboostRegressor`."""

    def _transform(self, dataset):
        X = self._features(dataset)
        pred = self._booster.predict(X, self.getOrDefault("missing"))
        out = dataset.copy()
        out[self.getPredictionCol()] = pred
        return out


class XgboostClassifier(_XgboostEstimator, HasProbabilityCol,
                        HasRawPredictionCol):
    """Binary classifier estimator (reference xgboost.py:247-331).

    The model's ``rawPredictionCol`` output column holds the margins
    (``output_margin=True`` equivalent, reference xgboost.py:264-276).

    >>> import pandas as pd, numpy as np
    >>> df = pd.DataFrame({"features": list(np.random.rand(64, 4)),
    ...                    "label": np.random.randint(0, 2, 64)})
    >>> xgb = XgboostClassifier(n_estimators=10, missing=0.0)
    >>> model = xgb.fit(df)
    >>> out = model.transform(df)
    """

    def _objective(self):
        return "binary:logistic"

    def _resolve_objective(self, y, params):
        # xgboost-style: >2 label values switch to softprob multiclass
        n_classes = int(np.max(y)) + 1 if len(y) else 2
        if n_classes > 2:
            params.setdefault("num_class", n_classes)
            return "multi:softprob"
        return "binary:logistic"

    def _model_class(self):
        return XgboostClassifierModel


class XgboostClassifierModel(_XgboostModel, HasProbabilityCol,
                             HasRawPredictionCol):
    """Model fitted by :class:`XgboostClassifier`.  transform() appends
    prediction, probability and rawPrediction(=margin) columns
    (reference xgboost.py:264-276)."""

    def _transform(self, dataset):
        X = self._features(dataset)
        missing = self.getOrDefault("missing")
        out = dataset.copy()
        if self._booster.n_classes > 2:
            margin = self._booster.predict_margin(X, missing)
            prob = self._booster.predict_proba(X, missing)
            out[self.getPredictionCol()] = \
                np.argmax(prob, axis=1).astype(np.float64)
            out[self.getOrDefault("probabilityCol")] = list(prob)
            out[self.getOrDefault("rawPredictionCol")] = list(margin)
            return out
        margin = self._booster.predict_margin(X, missing)
        prob1 = 1.0 / (1.0 + np.exp(-margin))
        out[self.getPredictionCol()] = (prob1 >= 0.5).astype(np.float64)
        out[self.getOrDefault("probabilityCol")] = \
            [np.array([1 - p, p]) for p in prob1]
        out[self.getOrDefault("rawPredictionCol")] = \
            [np.array([-m, m]) for m in margin]
        return out
