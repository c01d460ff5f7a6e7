"""{{app_name}} — a unionml_amd app: sklearn digits classifier + FastAPI.

Template parity: reference templates/basic/{{cookiecutter.app_name}}/app.py:1-56.
Train locally, serve over HTTP, or deploy to the backend:

    unionml-amd train app:model --local --output model.joblib
    unionml-amd serve app:fastapi_app --model-path model.joblib --port 8000
"""

from typing import List

import pandas as pd
from fastapi import FastAPI
from sklearn.linear_model import LogisticRegression

from unionml_amd import Dataset, Model

dataset = Dataset(name="{{app_name}}_dataset", test_size=0.2, shuffle=True, targets=["target"])
model = Model(name="{{app_name}}", init=LogisticRegression, dataset=dataset)


@dataset.reader
def reader() -> pd.DataFrame:
    from sklearn.datasets import load_digits

    return load_digits(as_frame=True).frame


@model.trainer
def trainer(
    estimator: LogisticRegression,
    features: pd.DataFrame,
    target: pd.DataFrame,
    *,
    max_iter: int = 1000,
) -> LogisticRegression:
    estimator.set_params(max_iter=max_iter)
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


# serve with: unionml-amd serve app:fastapi_app --model-path <artifact>
fastapi_app = FastAPI()
model.serve(fastapi_app)


if __name__ == "__main__":
    model_object, metrics = model.train(hyperparameters={"C": 1.0})
    print(model_object, metrics)
    predictions = model.predict(features=reader().sample(5, random_state=42).drop(columns=["target"]))
    print(predictions)
    model.save("model.joblib")
