"""{{app_name}} — unionml_amd app served on AWS Lambda via Mangum.

Template parity: reference templates/basic-aws-lambda (the same FastAPI
app wrapped in a Mangum ASGI adapter; pattern of the reference's
tests/unit/aws_lambda_app/app.py:36-40). The model artifact ships inside
the Lambda package and loads from $UNIONML_MODEL_PATH at cold start.
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


fastapi_app = FastAPI()
model.serve(fastapi_app)

# the Lambda entrypoint: Mangum when installed, else the built-in
# dependency-free API-Gateway adapter
from unionml_amd.services.awslambda import lambda_handler_for

lambda_handler = lambda_handler_for(fastapi_app)
