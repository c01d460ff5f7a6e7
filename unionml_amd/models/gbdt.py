"""Gradient-boosted-trees app — the non-neural built-in model family.

sklearn's HistGradientBoostingClassifier on the digits shape through
the same decorator API (the reference's sklearn surface is exactly this
kind of estimator; its quickstart uses LogisticRegression —
tests/integration/sklearn_app/quickstart.py). Trees don't map to MFMA,
so this app runs CPU-side estimators by design; it exists so the
framework's "any sklearn estimator" path has a first-class, tested
boosting example with warm-start iteration control.
"""

from typing import List

import numpy as np
import pandas as pd
from sklearn.ensemble import HistGradientBoostingClassifier

from unionml_amd import Dataset, Model

FEATURES = [f"p{i}" for i in range(64)]

dataset = Dataset(name="gbdt_digits", features=FEATURES, targets=["target"], test_size=0.2)
model = Model(name="digits_gbdt", dataset=dataset)


@dataset.reader
def reader(n: int = 0, synthetic: bool = False, seed: int = 17) -> pd.DataFrame:
    if synthetic:
        rng = np.random.RandomState(seed)
        X = rng.rand(n or 1797, 64).astype(np.float32) * 16.0
        y = rng.randint(0, 10, size=len(X))
    else:
        from sklearn.datasets import load_digits

        digits = load_digits()
        X, y = digits.data, digits.target
        if n:
            X, y = X[:n], y[:n]
    frame = pd.DataFrame(X, columns=FEATURES)
    frame["target"] = y
    return frame


@model.init
def init(hyperparameters: dict) -> HistGradientBoostingClassifier:
    return HistGradientBoostingClassifier(
        **{"max_iter": 100, "random_state": 0, **(hyperparameters or {})}
    )


@model.trainer
def trainer(
    clf: HistGradientBoostingClassifier,
    features: pd.DataFrame,
    target: pd.DataFrame,
    *,
    max_iter: int = None,
) -> HistGradientBoostingClassifier:
    """One training pass; ``max_iter`` overrides the boosting-iteration
    count per run (a "step" for this family = one boosting iteration)."""
    if max_iter is not None:
        clf.set_params(max_iter=max_iter)
    return clf.fit(features, target.squeeze())


@model.predictor
def predictor(clf: HistGradientBoostingClassifier, features: pd.DataFrame) -> List[int]:
    return [int(x) for x in clf.predict(features)]


@model.evaluator
def evaluator(
    clf: HistGradientBoostingClassifier, features: pd.DataFrame, target: pd.DataFrame
) -> float:
    return float((np.asarray(predictor(clf, features)) == target.squeeze().to_numpy()).mean())
