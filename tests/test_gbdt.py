"""Gradient-boosted-trees built-in app (non-neural model family)."""


def test_gbdt_app_trains_and_serves(tmp_path):
    from fastapi import FastAPI
    from fastapi.testclient import TestClient

    from unionml_amd.models.gbdt import model

    model.artifact = None
    clf, metrics = model.train(trainer_kwargs={"max_iter": 20}, n=400)
    assert metrics["test"] > 0.5, metrics

    preds = model.predict(features=[{f"p{i}": float(i % 16) for i in range(64)}])
    assert len(preds) == 1 and 0 <= preds[0] <= 9

    # artifact round-trip (joblib default path for sklearn estimators)
    p = tmp_path / "gbdt.joblib"
    model.save(p)
    loaded = model._loader(str(p))
    assert type(loaded).__name__ == "HistGradientBoostingClassifier"

    app = FastAPI()
    model.serve(app)
    with TestClient(app) as client:
        r = client.post(
            "/predict", json={"features": [{f"p{i}": 1.0 for i in range(64)}]}
        )
        assert r.status_code == 200
