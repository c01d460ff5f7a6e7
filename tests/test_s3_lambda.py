"""S3-event reactive predictions: the basic-aws-lambda-s3 template's
handler driven with a canned S3 ObjectCreated event against a faked
boto3 (reference covers this flow in docs only —
docs/source/reacting_to_s3_events.md; here it is executable)."""

import json
import sys
import types
from pathlib import Path

import pytest
from typer.testing import CliRunner

from unionml_amd.cli import app as cli_app

runner = CliRunner()


class _FakeS3:
    """In-memory bucket store honoring the two calls the handler makes."""

    def __init__(self, store):
        self.store = store

    def download_file(self, bucket, key, local):
        Path(local).write_bytes(self.store[(bucket, key)])

    def put_object(self, Bucket, Key, Body):
        self.store[(Bucket, Key)] = Body.encode() if isinstance(Body, str) else Body


@pytest.mark.timeout(240)
def test_s3_event_handler_round_trip(tmp_path, monkeypatch):
    monkeypatch.chdir(tmp_path)
    result = runner.invoke(cli_app, ["init", "s3app", "--template", "basic-aws-lambda-s3"])
    assert result.exit_code == 0, result.output

    store = {}
    boto3 = types.ModuleType("boto3")
    boto3.client = lambda name: _FakeS3(store)
    monkeypatch.setitem(sys.modules, "boto3", boto3)

    sys.path.insert(0, str(tmp_path / "s3app"))
    try:
        import app as s3_app

        # train + persist the artifact the handler loads at cold start
        s3_app.model.train()
        artifact = tmp_path / "model.joblib"
        s3_app.model.save(artifact)
        s3_app.model.artifact = None
        monkeypatch.setenv("UNIONML_MODEL_PATH", str(artifact))

        # an uploaded features file (JSON records of digit pixels)
        feats = [{f"pixel_{r}_{c}": 0.0 for r in range(8) for c in range(8)}]
        store[("my-bucket", "incoming/batch1.json")] = json.dumps(feats).encode()

        event = {
            "Records": [
                {
                    "s3": {
                        "bucket": {"name": "my-bucket"},
                        "object": {"key": "incoming/batch1.json"},
                    }
                }
            ]
        }
        resp = s3_app.lambda_handler(event, context=None)
        assert resp["statusCode"] == 200
        body = json.loads(resp["body"])
        assert body[0]["n"] == 1
        out = json.loads(store[("my-bucket", "incoming/batch1.json.predictions.json")])
        assert len(out) == 1 and 0 <= out[0] <= 9
    finally:
        sys.path.remove(str(tmp_path / "s3app"))
        sys.modules.pop("app", None)
