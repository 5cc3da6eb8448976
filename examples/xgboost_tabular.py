"""sparkdl.xgboost quick-start: distributed GBT on synthetic tabular
data (BASELINE.json config 5 — runs on CPU; use_gpu=True for the HIP
histogram path).

    python examples/xgboost_tabular.py
"""
import sys, os; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))  # noqa

import numpy as np
import pandas as pd

from sparkdl.xgboost import XgboostClassifier, XgboostClassifierModel


def make_data(n=4000, f=12, seed=0):
    rng = np.random.RandomState(seed)
    X = rng.randn(n, f)
    X[rng.rand(n, f) < 0.05] = np.nan  # missing values
    y = ((X[:, 0] > 0) ^ (np.nan_to_num(X[:, 1]) * X[:, 2] > 0)) \
        .astype(float)
    return pd.DataFrame({"features": list(X), "label": y})


def run(tmpdir="/tmp"):
    df = make_data()
    train, test = df.iloc[:3000], df.iloc[3000:]

    clf = XgboostClassifier(n_estimators=60, max_depth=5,
                            learning_rate=0.2, num_workers=2)
    model = clf.fit(train)
    scored = model.transform(test)
    acc = float((scored["prediction"] == test["label"]).mean())
    print("test accuracy: %.3f" % acc)

    path = os.path.join(tmpdir, "xgb_model")
    model.write().overwrite().save(path)
    reloaded = XgboostClassifierModel.load(path)
    assert np.allclose(
        np.stack(reloaded.transform(test)["probability"].to_numpy()),
        np.stack(scored["probability"].to_numpy()))
    print("save/load roundtrip ok ->", path)
    return acc


if __name__ == "__main__":
    run()
