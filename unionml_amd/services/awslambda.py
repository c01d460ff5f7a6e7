"""AWS-Lambda serving — adapt the FastAPI serving app to API-Gateway
events.

The reference serves on Lambda by wrapping its FastAPI app in Mangum
(reference pattern: tests/unit/aws_lambda_app/app.py:36-40 and the
basic-aws-lambda template). Here ``lambda_handler_for(app)`` returns a
Mangum handler when mangum is installed and otherwise falls back to
:class:`MiniAsgiLambdaAdapter`, a dependency-free adapter for
API-Gateway REST (payload v1) and HTTP API (payload v2) events — enough
for the ``/``, ``/predict`` and ``/health`` routes, and unit-testable
with canned events (reference: tests/unit/test_aws_lambda_handler.py).
"""

import asyncio
import base64
import json
from typing import Any, Dict, Optional
from urllib.parse import urlencode


class MiniAsgiLambdaAdapter:
    """Minimal synchronous ASGI <-> API-Gateway adapter.

    Drives one ASGI request/response cycle per Lambda invocation. The
    FastAPI app's lifespan startup (artifact load) runs once on the
    first event — Lambda cold-start semantics.
    """

    def __init__(self, app):
        self.app = app
        self._lifespan_started = False

    # -- event parsing ------------------------------------------------

    @staticmethod
    def _parse_event(event: Dict[str, Any]):
        if event.get("version") == "2.0":  # HTTP API payload v2
            ctx = event.get("requestContext", {}).get("http", {})
            method = ctx.get("method", "GET")
            path = ctx.get("path", event.get("rawPath", "/"))
            query = event.get("rawQueryString", "")
        else:  # REST API payload v1
            method = event.get("httpMethod", "GET")
            path = event.get("path", "/")
            qs = event.get("queryStringParameters") or {}
            query = urlencode(qs)
        headers = {
            k.lower(): v for k, v in (event.get("headers") or {}).items()
        }
        body = event.get("body") or ""
        if event.get("isBase64Encoded"):
            body = base64.b64decode(body)
        elif isinstance(body, str):
            body = body.encode()
        return method, path, query, headers, body

    # -- lifespan -----------------------------------------------------

    async def _startup(self):
        receive_q = [
            {"type": "lifespan.startup"},
        ]
        done = asyncio.Event()

        async def receive():
            if receive_q:
                return receive_q.pop(0)
            await done.wait()
            return {"type": "lifespan.shutdown"}

        started = asyncio.Event()

        async def send(message):
            if message["type"] in ("lifespan.startup.complete", "lifespan.startup.failed"):
                started.set()

        task = asyncio.ensure_future(
            self.app({"type": "lifespan", "asgi": {"version": "3.0"}}, receive, send)
        )
        await started.wait()
        done.set()
        task.cancel()

    # -- invocation ---------------------------------------------------

    def __call__(self, event: Dict[str, Any], context: Any = None) -> Dict[str, Any]:
        loop = asyncio.new_event_loop()
        try:
            return loop.run_until_complete(self._handle(event))
        finally:
            loop.close()

    async def _handle(self, event: Dict[str, Any]) -> Dict[str, Any]:
        if not self._lifespan_started:
            await self._startup()
            self._lifespan_started = True

        method, path, query, headers, body = self._parse_event(event)
        scope = {
            "type": "http",
            "asgi": {"version": "3.0"},
            "http_version": "1.1",
            "method": method,
            "path": path,
            "raw_path": path.encode(),
            "query_string": query.encode(),
            "headers": [(k.encode(), v.encode()) for k, v in headers.items()],
            "scheme": "https",
            "server": ("lambda", 443),
            "client": ("0.0.0.0", 0),
        }

        sent = False
        response = {"statusCode": 500, "headers": {}, "body": ""}
        chunks = []

        async def receive():
            nonlocal sent
            if not sent:
                sent = True
                return {"type": "http.request", "body": body, "more_body": False}
            return {"type": "http.disconnect"}

        async def send(message):
            if message["type"] == "http.response.start":
                response["statusCode"] = message["status"]
                response["headers"] = {
                    k.decode(): v.decode() for k, v in message.get("headers", [])
                }
            elif message["type"] == "http.response.body":
                chunks.append(message.get("body", b""))

        await self.app(scope, receive, send)
        payload = b"".join(chunks)
        try:
            response["body"] = payload.decode()
            response["isBase64Encoded"] = False
        except UnicodeDecodeError:
            response["body"] = base64.b64encode(payload).decode()
            response["isBase64Encoded"] = True
        return response


def lambda_handler_for(app):
    """Mangum when installed, the built-in adapter otherwise."""
    try:
        from mangum import Mangum

        return Mangum(app)
    except ImportError:
        return MiniAsgiLambdaAdapter(app)


def make_api_gateway_event(
    path: str,
    method: str = "GET",
    body: Optional[Any] = None,
    version: str = "1.0",
) -> Dict[str, Any]:
    """A canned API-Gateway event for tests (reference test fixture
    shape: tests/unit/test_aws_lambda_handler.py)."""
    raw = json.dumps(body) if body is not None else None
    if version == "2.0":
        return {
            "version": "2.0",
            "rawPath": path,
            "rawQueryString": "",
            "headers": {"content-type": "application/json"},
            "requestContext": {"http": {"method": method, "path": path}},
            "body": raw,
            "isBase64Encoded": False,
        }
    return {
        "httpMethod": method,
        "path": path,
        "headers": {"Content-Type": "application/json"},
        "queryStringParameters": None,
        "body": raw,
        "isBase64Encoded": False,
    }
