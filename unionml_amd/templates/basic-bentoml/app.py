"""{{app_name}} — unionml_amd app served through BentoML.

Template parity: reference templates/basic-bentoml. Train locally, save
to the bento store, then `bentoml serve app:service.svc`.
"""

from typing import List

import pandas as pd
from sklearn.linear_model import LogisticRegression

from unionml_amd import Dataset, Model
from unionml_amd.services.bentoml import BentoMLService

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


service = BentoMLService(model, name="{{app_name}}")


if __name__ == "__main__":
    model.train()
    service.save_model()
    service.configure(features=list, predictions=list)
    print("saved + configured; serve with: bentoml serve app:service.svc")
