#!/usr/bin/env python3
"""BASELINE config 1: the sklearn LogisticRegression digits app
(README example) through model.train/predict on CPU — measures API
plumbing, no GPU. Prints one JSON line."""

import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from typing import List

import pandas as pd
from sklearn.linear_model import LogisticRegression

from unionml_amd import Dataset, Model


def build_app():
    dataset = Dataset(name="digits_dataset", test_size=0.2, shuffle=True, targets=["target"])
    model = Model(name="digits_classifier", init=LogisticRegression, dataset=dataset)

    @dataset.reader
    def reader() -> pd.DataFrame:
        from sklearn.datasets import load_digits

        return load_digits(as_frame=True).frame

    @model.trainer
    def trainer(
        estimator: LogisticRegression, features: pd.DataFrame, target: pd.DataFrame
    ) -> LogisticRegression:
        return estimator.fit(features, target.squeeze())

    @model.predictor
    def predictor(estimator: LogisticRegression, features: pd.DataFrame) -> List[float]:
        return [float(x) for x in estimator.predict(features)]

    @model.evaluator
    def evaluator(
        estimator: LogisticRegression, features: pd.DataFrame, target: pd.DataFrame
    ) -> float:
        from sklearn.metrics import accuracy_score

        return float(accuracy_score(target.squeeze(), estimator.predict(features)))

    return model, reader


def pct(xs, q):
    xs = sorted(xs)
    return xs[min(len(xs) - 1, int(q * len(xs)))]


def main():
    model, reader = build_app()

    t0 = time.perf_counter()
    _, metrics = model.train(hyperparameters={"C": 1.0, "max_iter": 1000})
    train_s = time.perf_counter() - t0

    one = reader().drop(columns=["target"]).head(1)
    feats = model._dataset.get_features(one)
    for _ in range(20):
        model.predict(features=feats)
    lat = []
    for _ in range(300):
        t0 = time.perf_counter()
        model.predict(features=feats)
        lat.append((time.perf_counter() - t0) * 1000.0)

    print(
        json.dumps(
            {
                "metric": "sklearn_cpu_plumbing",
                "value": train_s,
                "unit": "s (train wall)",
                "n_gpus": 0,
                "higher_is_better": False,
                "config": {
                    "model": "LogisticRegression digits (README example)",
                    "train_wall_s": train_s,
                    "test_accuracy": metrics["test"],
                    "local_predict_p50_ms": pct(lat, 0.50),
                    "local_predict_p99_ms": pct(lat, 0.99),
                },
            }
        )
    )


if __name__ == "__main__":
    main()
