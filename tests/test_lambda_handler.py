"""AWS-Lambda handler tests with canned API-Gateway events (coverage
shape of the reference's tests/unit/test_aws_lambda_handler.py, which
invokes a Mangum-wrapped app with a recorded event). Runs with the
dependency-free built-in adapter; exercises mangum instead when it is
installed."""

import json

import pytest
from fastapi import FastAPI

from tests.model_fixtures import build_sklearn_app
from unionml_amd.services.awslambda import (
    MiniAsgiLambdaAdapter,
    lambda_handler_for,
    make_api_gateway_event,
)


@pytest.fixture()
def handler():
    model = build_sklearn_app()
    model.train()
    app = FastAPI()
    model.serve(app)
    return MiniAsgiLambdaAdapter(app)


def test_health_route(handler):
    resp = handler(make_api_gateway_event("/health"), context=None)
    assert resp["statusCode"] == 200, resp
    assert json.loads(resp["body"]) == {"status": "ok"}


@pytest.mark.parametrize("version", ["1.0", "2.0"])
def test_predict_route_canned_event(handler, version):
    feats = [{"x1": 0.5, "x2": 0.1, "x3": 0.9}, {"x1": 0.1, "x2": 0.9, "x3": 0.2}]
    event = make_api_gateway_event(
        "/predict", method="POST", body={"features": feats}, version=version
    )
    resp = handler(event, context=None)
    assert resp["statusCode"] == 200, resp
    preds = json.loads(resp["body"])
    assert isinstance(preds, list) and len(preds) == 2


def test_banner_route(handler):
    resp = handler(make_api_gateway_event("/"), context=None)
    assert resp["statusCode"] == 200


def test_bad_body_is_422_not_crash(handler):
    event = make_api_gateway_event("/predict", method="POST", body=None)
    event["body"] = "not json {"
    resp = handler(event, context=None)
    assert resp["statusCode"] in (400, 422)


def test_lambda_handler_for_falls_back_without_mangum():
    model = build_sklearn_app()
    model.train()
    app = FastAPI()
    model.serve(app)
    h = lambda_handler_for(app)
    try:
        import mangum  # noqa: F401

        from mangum import Mangum

        assert isinstance(h, Mangum)
    except ImportError:
        assert isinstance(h, MiniAsgiLambdaAdapter)
    resp = h(make_api_gateway_event("/health"), None)
    assert resp["statusCode"] == 200


def test_base64_encoded_event_body(handler):
    """API Gateway may deliver base64-encoded bodies; the adapter must
    decode them before handing the bytes to the ASGI app."""
    import base64

    feats = [{"x1": 0.5, "x2": 0.1, "x3": 0.9}]
    raw = json.dumps({"features": feats}).encode()
    event = make_api_gateway_event("/predict", method="POST")
    event["body"] = base64.b64encode(raw).decode()
    event["isBase64Encoded"] = True
    resp = handler(event, context=None)
    assert resp["statusCode"] == 200, resp
    assert len(json.loads(resp["body"])) == 1
