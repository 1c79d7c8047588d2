"""REST facade — TF Serving's HTTP API, both tiers.

URL grammar matches the reference's regex (case-insensitive, version
optional): pkg/tfservingproxy/tfservingproxy.go:24
    ^/v1/models/(?P<modelName>[^/]+)(/versions/(?P<version>[0-9]+))?
with verbs :predict / :classify / :regress, plus /metadata and the bare
status GET. (The reference's proxy returned 400 when the version was
missing — a documented quirk; here version is optional
and means "latest", matching TF Serving itself.)

Two roles, like the reference's shared RestProxy with different
directors (taskhandler.go:95 vs cachemanager.go:268):
  * cache tier  -> LocalServingHandler (in-process engine),
  * proxy tier  -> forward to the owning node over HTTP.
"""
from __future__ import annotations

import asyncio
import json
import logging
import re
from typing import Callable, Optional

from aiohttp import ClientSession, ClientTimeout, web

import numpy as np

from ..utils import metrics as mt
from ..wire import messages as m
from .json_codec import (RestCodecError, parse_predict_body,
                         render_predict_response)
from .servinghandler import LocalServingHandler, ServingError

log = logging.getLogger("tfsc.rest")

URL_RE = re.compile(
    r"^/v1/models/(?P<modelName>[^/:]+)"
    r"(/versions/(?P<version>[0-9]+)|/labels/(?P<label>[^/:]+))?"
    r"(?P<rest>.*)$", re.IGNORECASE)

_CODE_TO_HTTP = {
    m.ERROR_INVALID_ARGUMENT: 400,
    m.ERROR_NOT_FOUND: 404,
    m.ERROR_UNAVAILABLE: 503,
    m.ERROR_UNKNOWN: 500,
}


def _error_response(msg: str, code: int = 400) -> web.Response:
    return web.json_response({"error": msg}, status=code)


def parse_model_url(path: str):
    """Returns (model_name, version, label, verb) or None. verb in
    {'predict','classify','regress','metadata',''(status)}. The
    reference's grammar has /versions/N only (tfservingproxy.go:24);
    /labels/<label> is TF Serving's own REST grammar, resolved through
    the ReloadConfig version_labels map."""
    match = URL_RE.match(path)
    if not match:
        return None
    name = match.group("modelName")
    version = int(match.group("version") or 0)
    label = match.group("label") or ""
    rest = match.group("rest") or ""
    verb = ""
    if rest.startswith(":"):
        verb = rest[1:].lower()
    elif rest.lower() == "/metadata":
        verb = "metadata"
    elif rest not in ("", "/"):
        return None
    return name, version, label, verb


# ---------------------------------------------------------------------------
# Cache tier: serve locally
# ---------------------------------------------------------------------------
def make_cache_rest_app(handler: LocalServingHandler,
                        metrics_path: str = "/monitoring/prometheus/metrics",
                        metrics_render: Optional[Callable[[], bytes]] = None,
                        ) -> web.Application:
    app = web.Application(client_max_size=256 * 1024 * 1024)

    async def handle(request: web.Request) -> web.Response:
        parsed = parse_model_url(request.path)
        if parsed is None:
            mt.proxy_requests_failed.labels("http").inc()
            return _error_response(
                f"Malformed url: {request.path}", 404)
        name, version, label, verb = parsed
        loop = asyncio.get_running_loop()
        try:
            if verb == "predict":
                body = await request.read()
                return web.json_response(await loop.run_in_executor(
                    None, _predict_sync, handler, name, version, body,
                    label))
            if verb in ("classify", "regress"):
                body = await request.read()
                return web.json_response(await loop.run_in_executor(
                    None, _classify_regress_sync, handler, name, version,
                    verb, body, label))
            if verb == "metadata":
                return web.json_response(await loop.run_in_executor(
                    None, _metadata_sync, handler, name, version, label))
            if verb == "":
                return web.json_response(await loop.run_in_executor(
                    None, _status_sync, handler, name, version, label))
            return _error_response(f"unsupported method :{verb}", 400)
        except ServingError as e:
            mt.proxy_requests_failed.labels("http").inc()
            return _error_response(str(e), _CODE_TO_HTTP.get(e.code, 500))
        except Exception as e:      # noqa: BLE001
            log.exception("REST handler error")
            mt.proxy_requests_failed.labels("http").inc()
            return _error_response(str(e), 500)

    app.router.add_route("*", "/v1/models/{tail:.*}", handle)

    async def healthz(request: web.Request) -> web.Response:
        return web.Response(text="ok")

    app.router.add_get("/healthz", healthz)

    if metrics_render is not None:
        async def metrics_handler(request: web.Request) -> web.Response:
            data = await asyncio.get_running_loop().run_in_executor(
                None, metrics_render)
            return web.Response(body=data, content_type="text/plain")
        app.router.add_get(metrics_path, metrics_handler)
    return app


def _predict_sync(handler, name, version, body: bytes,
                  label: str = "") -> dict:
    mt.proxy_requests_total.labels("http").inc()
    try:
        payload = json.loads(body or b"{}")
    except json.JSONDecodeError as e:
        raise ServingError(f"invalid JSON: {e}")
    try:
        inputs, fmt, _sig = parse_predict_body(payload)
    except RestCodecError as e:
        raise ServingError(str(e))
    outputs, _version = handler.predict_arrays(
        name, version, inputs, version_label=label)
    return render_predict_response(outputs, fmt)


def _classify_regress_sync(handler, name, version, verb,
                           body: bytes, label: str = "") -> dict:
    mt.proxy_requests_total.labels("http").inc()
    try:
        payload = json.loads(body or b"{}")
    except json.JSONDecodeError as e:
        raise ServingError(f"invalid JSON: {e}")
    examples = payload.get("examples")
    if not isinstance(examples, list) or not examples:
        raise ServingError('missing "examples" list')
    # build features arrays from JSON examples
    cols = {}
    for ex in examples:
        if not isinstance(ex, dict):
            raise ServingError("each example must be an object")
        for k, v in ex.items():
            cols.setdefault(k, []).append(v)
    feats = {k: np.asarray(v, dtype=np.float32) for k, v in cols.items()}
    if len(feats) == 1:
        feats = {"": next(iter(feats.values()))}
    outputs, _version = handler.predict_arrays(
        name, version, feats, version_label=label)
    vals = next(iter(outputs.values()))
    if verb == "regress":
        return {"results": np.asarray(vals, dtype=np.float64).reshape(-1)
                .tolist()}
    scores = np.atleast_2d(np.asarray(vals, dtype=np.float64))
    return {"results": [
        [[str(i), float(s)] for i, s in enumerate(row)] for row in scores]}


def _status_sync(handler, name, version, label: str = "") -> dict:
    mt.proxy_requests_total.labels("http").inc()
    req = m.GetModelStatusRequest(model_spec=m.ModelSpec(
        name=name,
        version=m.Int64Value(value=version) if version else None,
        version_label=label))
    resp = handler.get_model_status(req)
    return {"model_version_status": [
        {"version": str(s.version),
         "state": m.STATE_NAMES.get(s.state, "UNKNOWN"),
         "status": {"error_code": "OK" if not s.status or
                    s.status.error_code == 0 else "UNKNOWN",
                    "error_message": s.status.error_message
                    if s.status else ""}}
        for s in resp.model_version_status]}


def _metadata_sync(handler, name, version, label: str = "") -> dict:
    mt.proxy_requests_total.labels("http").inc()
    req = m.GetModelMetadataRequest(model_spec=m.ModelSpec(
        name=name,
        version=m.Int64Value(value=version) if version else None,
        version_label=label),
        metadata_field=["signature_def"])
    resp = handler.get_model_metadata(req)
    model, _v = handler.get_model(name, version, version_label=label)
    sig = model.signature_def
    sig_json = {}
    if sig is not None:
        def ti_json(ti):
            return {"dtype": m.DTYPE_NAMES.get(ti.dtype, "DT_INVALID"),
                    "tensor_shape": {"dim": [
                        {"size": str(d.size)} for d in
                        (ti.tensor_shape.dim if ti.tensor_shape else [])]},
                    "name": ti.name}
        sig_json = {"serving_default": {
            "inputs": {k: ti_json(v) for k, v in sig.inputs.items()},
            "outputs": {k: ti_json(v) for k, v in sig.outputs.items()},
            "method_name": sig.method_name}}
    return {
        "model_spec": {"name": name, "version": str(
            resp.model_spec.version_value() if resp.model_spec else 0),
            "signature_name": ""},
        "metadata": {"signature_def": {"signature_def": sig_json}}}


# ---------------------------------------------------------------------------
# Proxy tier: forward to the owning node
# ---------------------------------------------------------------------------
def make_proxy_rest_app(pick_node: Callable[[str, int], str],
                        metrics_path: str = "/monitoring/prometheus/metrics",
                        metrics_render: Optional[Callable[[], bytes]] = None,
                        timeout_s: float = 60.0) -> web.Application:
    """pick_node(model, version) -> 'host:restPort' of the cache node."""
    app = web.Application(client_max_size=256 * 1024 * 1024)
    session: dict = {}

    async def _get_session() -> ClientSession:
        if "s" not in session:
            session["s"] = ClientSession(
                timeout=ClientTimeout(total=timeout_s))
        return session["s"]

    async def handle(request: web.Request) -> web.Response:
        mt.proxy_requests_total.labels("http").inc()
        parsed = parse_model_url(request.path)
        if parsed is None:
            mt.proxy_requests_failed.labels("http").inc()
            return _error_response(f"Malformed url: {request.path}", 404)
        name, version, _label, _verb = parsed
        try:
            target = pick_node(name, version)
        except Exception as e:      # noqa: BLE001
            mt.proxy_requests_failed.labels("http").inc()
            return _error_response(f"no node for model {name}: {e}", 503)
        url = f"http://{target}{request.path_qs}"
        body = await request.read()
        try:
            s = await _get_session()
            async with s.request(request.method, url, data=body,
                                 headers={"Content-Type":
                                          request.content_type or
                                          "application/json"}) as resp:
                data = await resp.read()
                return web.Response(
                    body=data, status=resp.status,
                    content_type=resp.content_type or "application/json")
        except Exception as e:      # noqa: BLE001
            mt.proxy_requests_failed.labels("http").inc()
            return _error_response(f"forwarding to {target} failed: {e}", 502)

    app.router.add_route("*", "/v1/models/{tail:.*}", handle)

    async def on_cleanup(app_):
        if "s" in session:
            await session["s"].close()
    app.on_cleanup.append(on_cleanup)

    if metrics_render is not None:
        async def metrics_handler(request: web.Request) -> web.Response:
            data = await asyncio.get_running_loop().run_in_executor(
                None, metrics_render)
            return web.Response(body=data, content_type="text/plain")
        app.router.add_get(metrics_path, metrics_handler)
    return app


# ---------------------------------------------------------------------------
# Sync dispatcher for the NATIVE REST front-end (engine/csrc/
# rest_frontend.cpp): (method, path, body) -> (status, content_type,
# body bytes). Routes exactly like make_cache_rest_app.
# ---------------------------------------------------------------------------
def make_rest_dispatcher(handler: LocalServingHandler,
                         metrics_path: str =
                         "/monitoring/prometheus/metrics",
                         metrics_render: Optional[Callable[[], bytes]]
                         = None):
    def dispatch(method: str, path: str, body: bytes):
        try:
            if path == "/healthz":
                return 200, "text/plain", b"ok"
            if metrics_render is not None and path == metrics_path:
                return 200, "text/plain", metrics_render()
            parsed = parse_model_url(path)
            if parsed is None:
                mt.proxy_requests_failed.labels("http").inc()
                return (404, "application/json", json.dumps(
                    {"error": f"Malformed url: {path}"}).encode())
            name, version, label, verb = parsed
            if verb == "predict":
                payload = _predict_sync(handler, name, version, body,
                                        label)
            elif verb in ("classify", "regress"):
                payload = _classify_regress_sync(handler, name, version,
                                                 verb, body)
            elif verb == "metadata":
                payload = _metadata_sync(handler, name, version)
            elif verb == "":
                payload = _status_sync(handler, name, version)
            else:
                return (400, "application/json", json.dumps(
                    {"error": f"unsupported method :{verb}"}).encode())
            return 200, "application/json", json.dumps(payload).encode()
        except ServingError as e:
            mt.proxy_requests_failed.labels("http").inc()
            return (_CODE_TO_HTTP.get(e.code, 500), "application/json",
                    json.dumps({"error": str(e)}).encode())
        except Exception as e:      # noqa: BLE001
            log.exception("REST dispatcher error")
            mt.proxy_requests_failed.labels("http").inc()
            return (500, "application/json",
                    json.dumps({"error": str(e)}).encode())
    return dispatch
