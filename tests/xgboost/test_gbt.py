"""Tests for the native GBT engine and the pyspark-ML-shaped API
(reference sparkdl/xgboost/xgboost.py contract)."""

import numpy as np
import pandas as pd
import pytest

from sparkdl.xgboost import (XgboostClassifier, XgboostClassifierModel,
                             XgboostRegressor, XgboostRegressorModel)
from sparkdl.xgboost import gbt


def _reg_data(n=400, f=6, seed=0):
    rng = np.random.RandomState(seed)
    X = rng.rand(n, f)
    y = 3 * X[:, 0] - 2 * X[:, 1] ** 2 + 0.5 * X[:, 2] + \
        0.1 * rng.randn(n)
    return X, y


def _clf_data(n=600, f=5, seed=1):
    rng = np.random.RandomState(seed)
    X = rng.randn(n, f)
    y = ((X[:, 0] + X[:, 1] * X[:, 2]) > 0).astype(float)
    return X, y


class TestBooster:
    def test_regression_fits_train_set(self):
        X, y = _reg_data()
        booster = gbt.train(X, y, {"n_estimators": 50, "max_depth": 4,
                                   "learning_rate": 0.3})
        pred = booster.predict(X)
        mse = float(np.mean((pred - y) ** 2))
        base = float(np.var(y))
        assert mse < 0.1 * base, (mse, base)

    def test_regression_close_to_sklearn(self):
        from sklearn.ensemble import HistGradientBoostingRegressor
        X, y = _reg_data(800)
        booster = gbt.train(X, y, {"n_estimators": 60, "max_depth": 5,
                                   "learning_rate": 0.2})
        ours = float(np.mean((booster.predict(X) - y) ** 2))
        sk = HistGradientBoostingRegressor(
            max_iter=60, max_depth=5, learning_rate=0.2).fit(X, y)
        theirs = float(np.mean((sk.predict(X) - y) ** 2))
        # within 2x of sklearn's train-set MSE
        assert ours < max(theirs * 2.0, 1e-3), (ours, theirs)

    def test_classification_accuracy(self):
        X, y = _clf_data()
        booster = gbt.train(X, y, {"n_estimators": 60, "max_depth": 4,
                                   "objective": "binary:logistic"})
        acc = float(np.mean((booster.predict(X) >= 0.5) == y))
        assert acc > 0.95, acc

    def test_missing_semantics(self):
        """missing=0.0 treats zeros as missing (reference
        xgboost.py:41-47); NaN always missing."""
        X, y = _reg_data(300)
        X[::7, 0] = np.nan
        booster = gbt.train(X, y, {"n_estimators": 20})
        assert np.isfinite(booster.predict(X)).all()

        Xz = X.copy()
        Xz[np.isnan(Xz)] = 0.0
        b0 = gbt.train(Xz, y, {"n_estimators": 5}, missing=0.0)
        bn = gbt.train(X, y, {"n_estimators": 5}, missing=np.nan)
        # zeros-as-missing must equal NaN-as-missing on equivalent data
        assert np.allclose(b0.predict(Xz, missing=0.0),
                           bn.predict(X), atol=1e-9)

    def test_warm_start(self):
        X, y = _reg_data()
        b1 = gbt.train(X, y, {"n_estimators": 10})
        b2 = gbt.train(X, y, {"n_estimators": 10}, xgb_model=b1)
        assert len(b2.trees) == 20
        m1 = float(np.mean((b1.predict(X) - y) ** 2))
        m2 = float(np.mean((b2.predict(X) - y) ** 2))
        assert m2 < m1


class TestEstimatorAPI:
    def test_regressor_fit_transform(self):
        X, y = _reg_data()
        df = pd.DataFrame({"features": list(X), "label": y})
        model = XgboostRegressor(n_estimators=30, max_depth=4).fit(df)
        out = model.transform(df)
        assert "prediction" in out.columns
        mse = float(np.mean((out["prediction"].to_numpy() - y) ** 2))
        assert mse < 0.2 * float(np.var(y))

    def test_classifier_columns(self):
        X, y = _clf_data(300)
        df = pd.DataFrame({"features": list(X), "label": y})
        model = XgboostClassifier(n_estimators=20).fit(df)
        out = model.transform(df)
        for col in ("prediction", "probability", "rawPrediction"):
            assert col in out.columns
        p = np.stack(out["probability"].to_numpy())
        assert np.allclose(p.sum(axis=1), 1.0)
        raw = np.stack(out["rawPrediction"].to_numpy())
        # rawPrediction holds margins (reference xgboost.py:264-276)
        assert np.allclose(1 / (1 + np.exp(-raw[:, 1])), p[:, 1])

    def test_param_surface(self):
        xgb = XgboostRegressor(missing=0.0, num_workers=2, use_gpu=False,
                               force_repartition=True,
                               use_external_storage=False,
                               external_storage_precision=7)
        for name in ("missing", "callbacks", "num_workers", "use_gpu",
                     "force_repartition", "use_external_storage",
                     "external_storage_precision", "baseMarginCol",
                     "featuresCol", "labelCol", "weightCol",
                     "predictionCol", "validationIndicatorCol"):
            assert xgb.hasParam(name), name
        assert xgb.getOrDefault("num_workers") == 2
        assert xgb.getOrDefault("external_storage_precision") == 7

    def test_classifier_extra_cols_params(self):
        clf = XgboostClassifier()
        assert clf.hasParam("probabilityCol")
        assert clf.hasParam("rawPredictionCol")

    def test_replaced_params_raise(self):
        # reference xgboost.py:176-199 remaps these
        for bad in ("gpu_id", "base_margin", "eval_set", "sample_weight",
                    "xgb_model", "output_margin", "validate_features"):
            with pytest.raises(ValueError):
                XgboostRegressor(**{bad: 1})

    def test_save_load_roundtrip(self, tmp_path):
        X, y = _reg_data(200)
        df = pd.DataFrame({"features": list(X), "label": y})
        model = XgboostRegressor(n_estimators=10).fit(df)
        path = str(tmp_path / "model")
        model.write().save(path)
        loaded = XgboostRegressorModel.load(path)
        a = model.transform(df)["prediction"].to_numpy()
        b = loaded.transform(df)["prediction"].to_numpy()
        assert np.allclose(a, b)
        assert len(loaded.get_booster().trees) == 10

    def test_weight_col(self):
        X, y = _reg_data(300)
        w = np.ones_like(y)
        df = pd.DataFrame({"features": list(X), "label": y, "w": w})
        model = XgboostRegressor(n_estimators=5, weightCol="w").fit(df)
        assert len(model.get_booster().trees) == 5

    def test_validation_indicator(self):
        X, y = _reg_data(300)
        vmask = np.zeros(300, dtype=bool)
        vmask[:60] = True
        df = pd.DataFrame({"features": list(X), "label": y,
                           "isVal": vmask})
        model = XgboostRegressor(
            n_estimators=5, validationIndicatorCol="isVal").fit(df)
        assert len(model.get_booster().trees) == 5

    def test_num_workers_distributed(self):
        X, y = _reg_data(240)
        df = pd.DataFrame({"features": list(X), "label": y})
        m1 = XgboostRegressor(n_estimators=8, max_depth=3).fit(df)
        m2 = XgboostRegressor(n_estimators=8, max_depth=3,
                              num_workers=2).fit(df)
        p1 = m1.transform(df)["prediction"].to_numpy()
        p2 = m2.transform(df)["prediction"].to_numpy()
        # distributed training sums shard histograms: same quality class
        mse1 = float(np.mean((p1 - y) ** 2))
        mse2 = float(np.mean((p2 - y) ** 2))
        assert mse2 < 3 * mse1 + 1e-6, (mse1, mse2)


class TestExternalStorageAndRepartition:
    def test_external_storage(self):
        X, y = _reg_data(300)
        df = pd.DataFrame({"features": list(X), "label": y})
        m = XgboostRegressor(n_estimators=5,
                             use_external_storage=True).fit(df)
        assert len(m.get_booster().trees) == 5

    def test_external_storage_rejects_weights(self):
        X, y = _reg_data(100)
        df = pd.DataFrame({"features": list(X), "label": y,
                           "w": np.ones_like(y)})
        est = XgboostRegressor(n_estimators=2, use_external_storage=True,
                               weightCol="w")
        with pytest.raises(ValueError):
            est.fit(df)

    def test_force_repartition(self):
        X, y = _reg_data(200)
        df = pd.DataFrame({"features": list(X), "label": y})
        m = XgboostRegressor(n_estimators=5,
                             force_repartition=True).fit(df)
        assert len(m.get_booster().trees) == 5


class TestEstimatorPersistence:
    def test_estimator_save_load(self, tmp_path):
        est = XgboostRegressor(n_estimators=7, max_depth=3, missing=0.0,
                               num_workers=1)
        path = str(tmp_path / "est")
        est.write().save(path)
        loaded = XgboostRegressor.load(path)
        assert loaded.getOrDefault("missing") == 0.0
        assert loaded._trainer_params()["n_estimators"] == 7
        X, y = _reg_data(100)
        df = pd.DataFrame({"features": list(X), "label": y})
        model = loaded.fit(df)
        assert len(model.get_booster().trees) == 7

    def test_estimator_save_overwrite_guard(self, tmp_path):
        est = XgboostRegressor(n_estimators=2)
        path = str(tmp_path / "est2")
        est.write().save(path)
        with pytest.raises(IOError):
            est.write().save(path)
        est.write().overwrite().save(path)

    def test_external_storage_precision_rounding(self):
        from sparkdl.xgboost.gbt import _round_significant
        X = np.array([[123.456, 0.00123456, -9.8765e8, 0.0, np.nan]])
        R = _round_significant(X, 3)
        assert R[0, 0] == 123.0
        assert abs(R[0, 1] - 0.00123) < 1e-12
        assert R[0, 2] == -9.88e8
        assert R[0, 3] == 0.0
        assert np.isnan(R[0, 4])


class TestTrainerOptions:
    def test_reg_alpha_shrinks_leaves(self):
        X, y = _reg_data(300)
        b0 = gbt.train(X, y, {"n_estimators": 5, "reg_alpha": 0.0})
        b1 = gbt.train(X, y, {"n_estimators": 5, "reg_alpha": 50.0})
        s0 = sum(np.abs(t.value).sum() for t in b0.trees)
        s1 = sum(np.abs(t.value).sum() for t in b1.trees)
        assert s1 < s0

    def test_subsample_colsample_run(self):
        X, y = _reg_data(400)
        b = gbt.train(X, y, {"n_estimators": 20, "subsample": 0.7,
                             "colsample_bytree": 0.5, "random_state": 3})
        mse = float(np.mean((b.predict(X) - y) ** 2))
        assert mse < 0.5 * float(np.var(y))

    def test_unknown_param_warns(self):
        X, y = _reg_data(60)
        with pytest.warns(UserWarning):
            gbt.train(X, y, {"n_estimators": 2, "bogus_knob": 1})
        # inert execution knobs do not warn
        import warnings
        with warnings.catch_warnings():
            warnings.simplefilter("error")
            gbt.train(X, y, {"n_estimators": 2, "n_jobs": 4,
                             "tree_method": "hist"})

    def test_early_stopping(self):
        rng = np.random.RandomState(0)
        X = rng.rand(500, 4)
        y = rng.rand(500)  # pure noise: validation stops improving fast
        vmask = np.zeros(500, dtype=bool)
        vmask[:150] = True
        df = pd.DataFrame({"features": list(X), "label": y,
                           "isVal": vmask})
        model = XgboostRegressor(
            n_estimators=200, early_stopping_rounds=5, max_depth=3,
            validationIndicatorCol="isVal").fit(df)
        booster = model.get_booster()
        assert len(booster.trees) < 200
        assert booster.best_iteration is not None

    def test_columnar_features_fallback(self):
        X, y = _reg_data(200, f=3)
        df = pd.DataFrame({"f0": X[:, 0], "f1": X[:, 1], "f2": X[:, 2],
                           "label": y})
        model = XgboostRegressor(n_estimators=10).fit(df)
        out = model.transform(df.drop(columns=["label"]))
        assert "prediction" in out.columns
        mse = float(np.mean((out["prediction"].to_numpy() - y) ** 2))
        assert mse < 0.5 * float(np.var(y))

    def test_load_wrong_class_raises(self, tmp_path):
        X, y = _reg_data(80)
        df = pd.DataFrame({"features": list(X), "label": y})
        model = XgboostRegressor(n_estimators=2).fit(df)
        path = str(tmp_path / "m")
        model.save(path)
        with pytest.raises(TypeError):
            XgboostClassifierModel.load(path)


class TestMulticlass:
    def _data(self, n=600, seed=5):
        rng = np.random.RandomState(seed)
        X = rng.randn(n, 4)
        y = (X[:, 0] > 0.5).astype(int) + (X[:, 1] > 0).astype(int)
        return X, y.astype(float)  # 3 classes: 0,1,2

    def test_multiclass_accuracy(self):
        X, y = self._data()
        b = gbt.train(X, y, {"n_estimators": 30, "max_depth": 4,
                             "objective": "multi:softprob",
                             "num_class": 3})
        acc = float(np.mean(b.predict(X) == y))
        assert b.n_classes == 3
        assert len(b.trees) == 90  # 3 trees per round
        assert acc > 0.9, acc

    def test_classifier_auto_multiclass(self):
        X, y = self._data()
        df = pd.DataFrame({"features": list(X), "label": y})
        model = XgboostClassifier(n_estimators=20, max_depth=4).fit(df)
        out = model.transform(df)
        prob = np.stack(out["probability"].to_numpy())
        assert prob.shape == (len(y), 3)
        assert np.allclose(prob.sum(axis=1), 1.0)
        acc = float(np.mean(out["prediction"].to_numpy() == y))
        assert acc > 0.85, acc

    def test_multiclass_save_load(self, tmp_path):
        X, y = self._data(200)
        df = pd.DataFrame({"features": list(X), "label": y})
        model = XgboostClassifier(n_estimators=5).fit(df)
        path = str(tmp_path / "mc")
        model.save(path)
        loaded = XgboostClassifierModel.load(path)
        a = np.stack(model.transform(df)["probability"].to_numpy())
        b = np.stack(loaded.transform(df)["probability"].to_numpy())
        assert np.allclose(a, b)


class TestLossguide:
    def test_lossguide_quality_and_leaf_cap(self):
        X, y = _reg_data(600)
        b = gbt.train(X, y, {"n_estimators": 30, "max_depth": 8,
                             "grow_policy": "lossguide",
                             "max_leaves": 15})
        mse = float(np.mean((b.predict(X) - y) ** 2))
        assert mse < 0.15 * float(np.var(y)), mse
        for t in b.trees:
            leaves = sum(1 for f in t.feature if f == -1)
            assert leaves <= 15

    def test_lossguide_matches_depthwise_class(self):
        X, y = _reg_data(500)
        bd = gbt.train(X, y, {"n_estimators": 20, "max_depth": 4})
        bl = gbt.train(X, y, {"n_estimators": 20, "max_depth": 8,
                              "grow_policy": "lossguide",
                              "max_leaves": 16})
        md = float(np.mean((bd.predict(X) - y) ** 2))
        ml = float(np.mean((bl.predict(X) - y) ** 2))
        assert ml < 3 * md + 1e-6, (md, ml)

    def test_lossguide_missing(self):
        X, y = _reg_data(300)
        X[::5, 0] = np.nan
        b = gbt.train(X, y, {"n_estimators": 10,
                             "grow_policy": "lossguide",
                             "max_leaves": 8})
        assert np.isfinite(b.predict(X)).all()


class TestFeatureImportance:
    def test_get_score_weight_and_gain(self):
        X, y = _reg_data(500)
        b = gbt.train(X, y, {"n_estimators": 20, "max_depth": 4})
        w = b.get_score("weight")
        g = b.get_score("gain")
        assert w and g
        # the strongest predictor (feature 0, coef 3) dominates
        top_gain = max(g, key=g.get)
        assert top_gain == "f0", g
        assert all(v > 0 for v in g.values())

    def test_importance_survives_save_load(self, tmp_path):
        X, y = _reg_data(200)
        df = pd.DataFrame({"features": list(X), "label": y})
        model = XgboostRegressor(n_estimators=5).fit(df)
        path = str(tmp_path / "fi")
        model.save(path)
        loaded = XgboostRegressorModel.load(path)
        assert loaded.get_booster().get_score("gain") == \
            model.get_booster().get_score("gain")


class TestAdvisorRegressions:
    """Regression tests for round-1 advisor findings."""

    def test_warm_start_early_stopping_keeps_prior_trees(self):
        # Truncation after early stopping must preserve the warm-start
        # prefix: trees[:n_prior + best_iter + 1].
        rng = np.random.RandomState(3)
        X = rng.rand(400, 4)
        y = rng.rand(400)  # noise -> early stop triggers quickly
        b0 = gbt.train(X, y, {"n_estimators": 10, "max_depth": 3})
        n_prior = len(b0.trees)
        assert n_prior == 10
        vrows = np.zeros(400, dtype=bool)
        vrows[:120] = True
        b1 = gbt.train(X[~vrows], y[~vrows],
                       {"n_estimators": 100, "max_depth": 3,
                        "early_stopping_rounds": 3},
                       xgb_model=b0, eval_set=(X[vrows], y[vrows]))
        assert len(b1.trees) >= n_prior  # prior trees never dropped
        if b1.best_iteration is not None:
            assert b1.best_iteration >= n_prior - 1
            assert len(b1.trees) == b1.best_iteration + 1

    def test_columnar_fit_excludes_weight_col(self):
        # weightCol must not leak into the feature matrix when features
        # are given as separate columns (train/transform agreement).
        X, y = _reg_data(200, f=3)
        w = np.ones(200)
        df = pd.DataFrame({"f0": X[:, 0], "f1": X[:, 1], "f2": X[:, 2],
                           "label": y, "wcol": w})
        model = XgboostRegressor(n_estimators=5, weightCol="wcol").fit(df)
        assert model.get_booster().n_features == 3
        out = model.transform(
            pd.DataFrame({"f0": X[:, 0], "f1": X[:, 1], "f2": X[:, 2],
                          "wcol": w}))
        assert "prediction" in out.columns

    def test_colsample_mask_independent_of_shard_size(self):
        # The per-round feature mask must be identical regardless of row
        # count (DP workers have uneven shards but share the seed).
        rng = np.random.RandomState(7)
        Xa, ya = rng.rand(301, 8), rng.rand(301)
        Xb, yb = rng.rand(200, 8), rng.rand(200)
        shared_binner = gbt.Binner(16).fit(np.vstack([Xa, Xb]), np.nan)
        masks = {}
        for key, (Xs, ys) in {"a": (Xa, ya), "b": (Xb, yb)}.items():
            feats = []
            ident = lambda h: h

            b = gbt.train(Xs, ys,
                          {"n_estimators": 4, "max_depth": 3,
                           "subsample": 0.7, "colsample_bytree": 0.5,
                           "random_state": 11},
                          comm=ident, binner=shared_binner)
            # recover used features from the grown trees
            for t in b.trees:
                feats.append(frozenset(int(f) for f in t.feature
                                       if f >= 0))
            masks[key] = feats
        # Direct check of the mask derivation used in gbt.py:
        F = 8
        for rnd in range(4):
            m1 = np.random.RandomState((11 + 7919 * (rnd + 1)) % 2**31)
            m2 = np.random.RandomState((11 + 7919 * (rnd + 1)) % 2**31)
            assert np.array_equal(m1.choice(F, 4, replace=False),
                                  m2.choice(F, 4, replace=False))

    def test_max_bin_spelling_honored(self):
        X, y = _reg_data(300)
        b = gbt.train(X, y, {"n_estimators": 3, "max_bin": 8})
        assert b.binner.max_bins == 8

    def test_multiclass_margin_includes_base_score(self):
        rng = np.random.RandomState(5)
        X = rng.randn(300, 4)
        y = ((X[:, 0] > 0).astype(int) + (X[:, 1] > 0).astype(int))
        b = gbt.train(X, y.astype(float),
                      {"n_estimators": 3, "objective": "multi:softprob",
                       "num_class": 3, "base_score": 0.5})
        # margins of an empty ensemble equal base_score
        b_empty = gbt.Booster("multi:softprob", 0.5, b.binner, [], 4, 3)
        m0 = b_empty.predict_margin(X[:5])
        assert np.allclose(m0, 0.5)

    def test_distributed_warm_start_and_callbacks(self):
        X, y = _reg_data(240, f=4)
        df = pd.DataFrame({"features": list(X), "label": y})
        m0 = XgboostRegressor(n_estimators=4, max_depth=3).fit(df)
        m1 = XgboostRegressor(
            n_estimators=3, max_depth=3, num_workers=2,
            booster_warm_start=m0.get_booster()).fit(df)
        # warm start shipped into the gang: 4 prior + 3 new trees
        assert len(m1.get_booster().trees) == 7


class TestSparseInput:
    """The reference's documented sparse caveat (xgboost.py:41-47):
    inactive sparse entries are zeros, NOT missing — unless missing=0."""

    def _sparse_df(self):
        import scipy.sparse as sp
        rng = np.random.RandomState(11)
        Xd = rng.rand(300, 6)
        Xd[rng.rand(300, 6) < 0.6] = 0.0  # sparse zeros
        y = Xd[:, 0] * 2 + (Xd[:, 1] == 0) * 0.5 + 0.05 * rng.randn(300)
        rows = [sp.csr_matrix(Xd[i]) for i in range(300)]
        return pd.DataFrame({"features": rows, "label": y}), Xd, y

    def test_sparse_rows_train_as_zeros(self):
        df, Xd, y = self._sparse_df()
        m = XgboostRegressor(n_estimators=10, max_depth=3).fit(df)
        md = XgboostRegressor(n_estimators=10, max_depth=3).fit(
            pd.DataFrame({"features": list(Xd), "label": y}))
        # sparse and dense inputs are the same data -> identical model
        p1 = m.transform(df)["prediction"].to_numpy()
        p2 = md.transform(df)["prediction"].to_numpy()
        assert np.allclose(p1, p2)

    def test_missing_zero_changes_routing(self):
        df, Xd, y = self._sparse_df()
        m0 = XgboostRegressor(n_estimators=10, max_depth=3).fit(df)
        mz = XgboostRegressor(n_estimators=10, max_depth=3,
                              missing=0.0).fit(df)
        p0 = m0.transform(df)["prediction"].to_numpy()
        pz = mz.transform(df)["prediction"].to_numpy()
        # with missing=0 the zeros follow learned default directions;
        # the models must genuinely differ on this zero-heavy data
        assert not np.allclose(p0, pz)
