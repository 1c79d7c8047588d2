"""gRPC facade — PredictionService / ModelService / SessionService / health.

Wire-compatible with TF Serving clients: service + method names match the
reference's generated stubs (prediction_service.pb.go:260,
model_service.pb.go:162, session_service.pb.go:272). No protoc in this
environment, so servicers are registered via grpc generic handlers with
the hand-rolled codec (wire/) as (de)serializer.

Two roles mirroring the reference's GrpcProxy with different directors:
  * cache tier: decode -> LocalServingHandler -> encode;
  * proxy tier: decode (for model_spec routing), pick node on the ring,
    re-send to the owning node — message-level proxying with unknown-field
    preservation, like the reference's decode/re-send proxying
    (tfservingproxy.go:168-244).
MultiInference returns UNIMPLEMENTED like the reference
(tfservingproxy.go:215-217).
"""
from __future__ import annotations

import logging
import threading
from concurrent import futures
from typing import Callable, Dict, Optional, Tuple

import grpc

from ..utils import metrics as mt
from ..wire import messages as m
from ..wire.pb import Message
from .servinghandler import LocalServingHandler, ServingError

log = logging.getLogger("tfsc.grpc")

_CODE_MAP = {
    m.ERROR_INVALID_ARGUMENT: grpc.StatusCode.INVALID_ARGUMENT,
    m.ERROR_NOT_FOUND: grpc.StatusCode.NOT_FOUND,
    m.ERROR_UNAVAILABLE: grpc.StatusCode.UNAVAILABLE,
    m.ERROR_UNKNOWN: grpc.StatusCode.UNKNOWN,
}

GRPC_MAX_MSG = 16 * 1024 * 1024     # reference default (cachemanager.go:230)


# ---------------------------------------------------------------------------
# health service (grpc.health.v1) — no grpc_health package in this env
# ---------------------------------------------------------------------------
class HealthCheckRequest(Message):
    FIELDS = [("service", 1, "string")]


class HealthCheckResponse(Message):
    FIELDS = [("status", 1, "enum")]


HEALTH_UNKNOWN, HEALTH_SERVING, HEALTH_NOT_SERVING = 0, 1, 2


class HealthState:
    def __init__(self):
        self._status = HEALTH_SERVING
        self._lock = threading.Lock()

    def set_serving(self, serving: bool) -> None:
        with self._lock:
            self._status = HEALTH_SERVING if serving else HEALTH_NOT_SERVING

    def get(self) -> int:
        with self._lock:
            return self._status


def _health_handler(state: HealthState):
    def check(request: HealthCheckRequest, context):
        return HealthCheckResponse(status=state.get())

    return grpc.method_handlers_generic_handler("grpc.health.v1.Health", {
        "Check": grpc.unary_unary_rpc_method_handler(
            check, HealthCheckRequest.decode,
            lambda r: r.encode()),
    })


def _wrap(fn, protocol="grpc"):
    def call(request, context):
        mt.proxy_requests_total.labels(protocol).inc()
        try:
            return fn(request)
        except ServingError as e:
            mt.proxy_requests_failed.labels(protocol).inc()
            context.abort(_CODE_MAP.get(e.code, grpc.StatusCode.UNKNOWN),
                          str(e))
        except Exception as e:      # noqa: BLE001
            log.exception("grpc handler error")
            mt.proxy_requests_failed.labels(protocol).inc()
            context.abort(grpc.StatusCode.INTERNAL, str(e))
    return call


def _unimplemented(request, context):
    context.abort(grpc.StatusCode.UNIMPLEMENTED,
                  "MultiInference not supported by TFServingCache")


# ---------------------------------------------------------------------------
# cache tier (local serving)
# ---------------------------------------------------------------------------
def make_cache_grpc_server(handler: LocalServingHandler,
                           health: Optional[HealthState] = None,
                           max_workers: int = 32,
                           max_msg: int = GRPC_MAX_MSG) -> Tuple[grpc.Server, HealthState]:
    health = health or HealthState()

    def uu(fn, req_cls):
        return grpc.unary_unary_rpc_method_handler(
            _wrap(fn), req_cls.decode, lambda r: r.encode())

    prediction = grpc.method_handlers_generic_handler(
        m.PREDICTION_SERVICE, {
            # Predict runs at the bytes level: the C++ fast path consumes
            # and produces raw wire bytes (servinghandler.predict_bytes)
            "Predict": grpc.unary_unary_rpc_method_handler(
                _wrap(handler.predict_bytes), bytes, lambda b: b),
            "Classify": uu(handler.classify, m.ClassificationRequest),
            "Regress": uu(handler.regress, m.RegressionRequest),
            "GetModelMetadata": uu(handler.get_model_metadata,
                                   m.GetModelMetadataRequest),
            "MultiInference": grpc.unary_unary_rpc_method_handler(
                _unimplemented, bytes, bytes),
        })
    modelsvc = grpc.method_handlers_generic_handler(
        m.MODEL_SERVICE, {
            "GetModelStatus": uu(handler.get_model_status,
                                 m.GetModelStatusRequest),
            "HandleReloadConfigRequest": uu(handler.handle_reload_config,
                                            m.ReloadConfigRequest),
        })
    session = grpc.method_handlers_generic_handler(
        m.SESSION_SERVICE, {
            "SessionRun": uu(handler.session_run, m.SessionRunRequest),
        })

    server = grpc.server(
        futures.ThreadPoolExecutor(max_workers=max_workers),
        options=[("grpc.max_receive_message_length", max_msg),
                 ("grpc.max_send_message_length", max_msg)])
    server.add_generic_rpc_handlers(
        (prediction, modelsvc, session, _health_handler(health)))
    return server, health


# ---------------------------------------------------------------------------
# proxy tier (ring-routed forwarding)
# ---------------------------------------------------------------------------
class GrpcForwarder:
    """Per-host channel cache (the reference's grpcConnMap,
    taskhandler.go:117-147 — including the dedup the reference missed on
    lock upgrade)."""

    def __init__(self, max_msg: int = GRPC_MAX_MSG, timeout_s: float = 10.0):
        self._channels: Dict[str, grpc.Channel] = {}
        self._lock = threading.Lock()
        self.max_msg = max_msg
        self.timeout_s = timeout_s

    def channel(self, target: str) -> grpc.Channel:
        with self._lock:
            ch = self._channels.get(target)
            if ch is None:
                ch = grpc.insecure_channel(
                    target,
                    options=[("grpc.max_receive_message_length", self.max_msg),
                             ("grpc.max_send_message_length", self.max_msg)])
                self._channels[target] = ch
            return ch

    def call(self, target: str, full_method: str, request_bytes: bytes
             ) -> bytes:
        ch = self.channel(target)
        fn = ch.unary_unary(full_method,
                            request_serializer=lambda b: b,
                            response_deserializer=lambda b: b)
        return fn(request_bytes, timeout=self.timeout_s)

    def close(self) -> None:
        with self._lock:
            for ch in self._channels.values():
                ch.close()
            self._channels.clear()


def make_proxy_grpc_server(pick_node: Callable[[str, int], str],
                           forwarder: Optional[GrpcForwarder] = None,
                           health: Optional[HealthState] = None,
                           max_workers: int = 32,
                           max_msg: int = GRPC_MAX_MSG
                           ) -> Tuple[grpc.Server, HealthState, GrpcForwarder]:
    """pick_node(model, version) -> 'host:grpcPort' of the cache node."""
    health = health or HealthState()
    fwd = forwarder or GrpcForwarder(max_msg=max_msg)

    def forward(service: str, method: str, req_cls):
        full_method = f"/{service}/{method}"

        def call(request_bytes: bytes, context):
            mt.proxy_requests_total.labels("grpc").inc()
            try:
                req = req_cls.decode(request_bytes)
                spec = getattr(req, "model_spec", None) or m.ModelSpec()
                target = pick_node(spec.name, spec.version_value())
                return fwd.call(target, full_method, request_bytes)
            except grpc.RpcError as e:
                mt.proxy_requests_failed.labels("grpc").inc()
                context.abort(e.code() if hasattr(e, "code")
                              else grpc.StatusCode.UNAVAILABLE, str(e))
            except Exception as e:      # noqa: BLE001
                log.exception("grpc proxy error")
                mt.proxy_requests_failed.labels("grpc").inc()
                context.abort(grpc.StatusCode.UNAVAILABLE, str(e))
        return grpc.unary_unary_rpc_method_handler(
            call, lambda b: b, lambda b: b)

    prediction = grpc.method_handlers_generic_handler(
        m.PREDICTION_SERVICE, {
            name: forward(m.PREDICTION_SERVICE, name, req_cls)
            for name, (req_cls, _resp) in m.PREDICTION_METHODS.items()
        } | {"MultiInference": grpc.unary_unary_rpc_method_handler(
            _unimplemented, bytes, bytes)})
    session = grpc.method_handlers_generic_handler(
        m.SESSION_SERVICE, {
            "SessionRun": forward(m.SESSION_SERVICE, "SessionRun",
                                  m.SessionRunRequest),
        })

    server = grpc.server(
        futures.ThreadPoolExecutor(max_workers=max_workers),
        options=[("grpc.max_receive_message_length", max_msg),
                 ("grpc.max_send_message_length", max_msg)])
    server.add_generic_rpc_handlers(
        (prediction, session, _health_handler(health)))
    return server, health, fwd
