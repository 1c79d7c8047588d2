"""Python side of the native (nghttp2) gRPC front-end.

The C++ server (engine/csrc/frontend.cpp) serves registered Predicts
entirely in C++; every other method — and any Predict it cannot handle —
lands here as `(path, payload_bytes) -> (grpc_status, message, bytes)`,
dispatched to the same LocalServingHandler the grpcio server uses, so
the two server implementations are wire-identical.
"""
from __future__ import annotations

import logging
from typing import Callable, Optional, Tuple

from ..utils import metrics as mt
from ..wire import messages as m
from .grpc_server import (HealthCheckRequest, HealthCheckResponse,
                          HealthState)
from .servinghandler import LocalServingHandler, ServingError

log = logging.getLogger("tfsc.native_frontend")

# grpc numeric status codes (google.rpc.Code)
_OK, _UNKNOWN, _INVALID, _NOT_FOUND = 0, 2, 3, 5
_UNIMPLEMENTED, _INTERNAL, _UNAVAILABLE = 12, 13, 14

_CODE_MAP = {
    m.ERROR_INVALID_ARGUMENT: _INVALID,
    m.ERROR_NOT_FOUND: _NOT_FOUND,
    m.ERROR_UNAVAILABLE: _UNAVAILABLE,
    m.ERROR_UNKNOWN: _UNKNOWN,
}


def make_dispatcher(handler: LocalServingHandler,
                    health: Optional[HealthState] = None
                    ) -> Callable[[str, bytes], Tuple[int, str, bytes]]:
    health = health or HealthState()
    P = f"/{m.PREDICTION_SERVICE}/"
    M = f"/{m.MODEL_SERVICE}/"
    S = f"/{m.SESSION_SERVICE}/"

    def uu(fn, req_cls):
        def call(payload: bytes) -> bytes:
            return fn(req_cls.decode(payload)).encode()
        return call

    def health_check(payload: bytes) -> bytes:
        HealthCheckRequest.decode(payload)
        return HealthCheckResponse(status=health.get()).encode()

    routes = {
        P + "Predict": handler.predict_bytes,
        P + "Classify": uu(handler.classify, m.ClassificationRequest),
        P + "Regress": uu(handler.regress, m.RegressionRequest),
        P + "GetModelMetadata": uu(handler.get_model_metadata,
                                   m.GetModelMetadataRequest),
        M + "GetModelStatus": uu(handler.get_model_status,
                                 m.GetModelStatusRequest),
        M + "HandleReloadConfigRequest": uu(handler.handle_reload_config,
                                            m.ReloadConfigRequest),
        S + "SessionRun": uu(handler.session_run, m.SessionRunRequest),
        "/grpc.health.v1.Health/Check": health_check,
    }

    def dispatch(path: str, payload: bytes) -> Tuple[int, str, bytes]:
        fn = routes.get(path)
        if fn is None:
            if path == P + "MultiInference":
                return (_UNIMPLEMENTED,
                        "MultiInference not supported by TFServingCache",
                        b"")
            return _UNIMPLEMENTED, f"unknown method {path}", b""
        mt.proxy_requests_total.labels("grpc").inc()
        try:
            return _OK, "", fn(payload)
        except ServingError as e:
            mt.proxy_requests_failed.labels("grpc").inc()
            return _CODE_MAP.get(e.code, _UNKNOWN), str(e), b""
        except Exception as e:      # noqa: BLE001
            log.exception("native frontend handler error (%s)", path)
            mt.proxy_requests_failed.labels("grpc").inc()
            return _INTERNAL, str(e), b""

    return dispatch


class NativeGrpcServer:
    """grpc.Server-shaped facade over the C++ front-end so the Server
    composition root can swap implementations by config."""

    def __init__(self, handler: LocalServingHandler,
                 health: Optional[HealthState] = None,
                 workers: int = 16):
        import torch  # noqa: F401  (loads libc10 for the extension)
        from ..engine import _tfsc_engine as ext
        self._fe = ext.GrpcFrontend(make_dispatcher(handler, health))
        self._workers = workers
        self._port: Optional[int] = None

    # grpc.Server-compatible surface (the subset Server uses)
    def add_insecure_port(self, addr: str) -> int:
        self._port = int(addr.rsplit(":", 1)[1])
        return self._port

    def start(self) -> None:
        self._port = self._fe.start(self._port or 0, self._workers)

    def stop(self, grace: float = 0.0) -> None:     # noqa: ARG002
        self._fe.stop()

    # registry plumbing (Server wires these to the model pool)
    def register_model(self, name: str, version: int, fast) -> None:
        self._fe.register_model(name, int(version), fast._ptr(), fast)

    def unregister_model(self, name: str, version: int) -> None:
        self._fe.unregister_model(name, int(version))

    @property
    def port(self) -> int:
        return self._fe.port()

    def native_hits(self) -> int:
        return self._fe.native_hits()

    def fallback_calls(self) -> int:
        return self._fe.fallback_calls()


class NativeRestServer:
    """HTTP/1.1 + JSON front-end (engine/csrc/rest_frontend.cpp):
    registered numeric Predicts run fully in C++; everything else hits
    the same sync dispatcher logic as the aiohttp app
    (rest.make_rest_dispatcher), keeping the two REST servers
    wire-identical."""

    def __init__(self, handler: LocalServingHandler,
                 metrics_path: str = "/monitoring/prometheus/metrics",
                 metrics_render=None):
        import torch  # noqa: F401
        from ..engine import _tfsc_engine as ext
        from .rest import make_rest_dispatcher
        self._fe = ext.RestFrontendNative(
            make_rest_dispatcher(handler, metrics_path, metrics_render))
        self._port: Optional[int] = None

    def add_insecure_port(self, addr: str) -> int:
        self._port = int(addr.rsplit(":", 1)[1])
        return self._port

    def start(self) -> None:
        self._port = self._fe.start(self._port or 0)

    def stop(self, grace: float = 0.0) -> None:     # noqa: ARG002
        self._fe.stop()

    def register_model(self, name: str, version: int, fast) -> None:
        self._fe.register_model(name, int(version), fast._ptr(), fast)

    def unregister_model(self, name: str, version: int) -> None:
        self._fe.unregister_model(name, int(version))

    @property
    def port(self) -> int:
        return self._fe.port()

    def native_hits(self) -> int:
        return self._fe.native_hits()

    def fallback_calls(self) -> int:
        return self._fe.fallback_calls()
