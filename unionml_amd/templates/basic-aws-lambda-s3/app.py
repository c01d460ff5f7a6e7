"""{{app_name}} — S3-event-reactive predictions on AWS Lambda.

Template parity: reference templates/basic-aws-lambda-s3 (pattern of
docs/source/reacting_to_s3_events.md:40-50): a Lambda handler triggered
by S3 object-created events downloads the features file, runs the
feature pipeline + prediction, and uploads the predictions next to it.
"""

import json
import os
import urllib.parse
from typing import List

import pandas as pd
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


def lambda_handler(event, context):
    """S3 object-created -> download features -> predict -> upload
    predictions as `<key>.predictions.json`."""
    import boto3

    if model.artifact is None:
        model.load_from_env()  # UNIONML_MODEL_PATH ships in the package

    s3 = boto3.client("s3")
    results = []
    for record in event.get("Records", []):
        bucket = record["s3"]["bucket"]["name"]
        key = urllib.parse.unquote_plus(record["s3"]["object"]["key"])
        local = os.path.join("/tmp", os.path.basename(key))
        s3.download_file(bucket, key, local)
        features = dataset.get_features(local)
        predictions = model.predict(features=features)
        out_key = f"{key}.predictions.json"
        s3.put_object(Bucket=bucket, Key=out_key, Body=json.dumps(predictions))
        results.append({"input": key, "output": out_key, "n": len(predictions)})
    return {"statusCode": 200, "body": json.dumps(results)}
