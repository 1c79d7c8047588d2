"""Native (nghttp2) gRPC front-end against a real grpcio client:
protocol correctness, error mapping, concurrency, large payloads, and
the Server config switch. On CPU every request goes through the Python
fallback dispatcher — the HTTP/2 + gRPC layer under test is identical
to the GPU path; the in-C++ registry hit is covered by the GPU tests."""
import queue
import socket
import threading
import time
from concurrent.futures import ThreadPoolExecutor

import grpc
import numpy as np
import pytest

from tfservingcache_amd.cachemanager import (CacheManager, LRUCache,
                                             ModelPool, make_cpu_loader)
from tfservingcache_amd.cachemanager.providers import DiskModelProvider
from tfservingcache_amd.models import write_model_repo
from tfservingcache_amd.tfservingproxy import LocalServingHandler
from tfservingcache_amd.tfservingproxy.native_frontend import \
    NativeGrpcServer
from tfservingcache_amd.wire import messages as m
from tfservingcache_amd.wire.tensor import (numpy_to_tensorproto,
                                            tensorproto_to_numpy)

pytest.importorskip("torch")


@pytest.fixture()
def served(tmp_path):
    write_model_repo(str(tmp_path / "repo"), [("mlp", 1, "mlp")])
    cache = LRUCache(str(tmp_path / "cache"), 10 ** 8)
    pool = ModelPool(make_cpu_loader(cache), max_concurrent_models=2)
    cm = CacheManager(DiskModelProvider(str(tmp_path / "repo")), cache,
                      pool)
    handler = LocalServingHandler(cm)
    srv = NativeGrpcServer(handler, workers=4)
    srv.add_insecure_port("[::]:0")
    srv.start()
    ch = grpc.insecure_channel(f"127.0.0.1:{srv.port}")
    yield srv, ch
    ch.close()
    srv.stop()


def _predict_rpc(ch):
    return ch.unary_unary(
        "/tensorflow.serving.PredictionService/Predict",
        request_serializer=lambda r: r.encode(),
        response_deserializer=m.PredictResponse.decode)


def test_native_predict_and_errors(served):
    srv, ch = served
    predict = _predict_rpc(ch)
    x = np.random.default_rng(0).standard_normal((3, 16)).astype(
        np.float32)
    req = m.PredictRequest(
        model_spec=m.ModelSpec(name="mlp", version=m.Int64Value(value=1)),
        inputs={"x": numpy_to_tensorproto(x)})
    out = tensorproto_to_numpy(predict(req, timeout=30).outputs["probs"])
    np.testing.assert_allclose(out.sum(-1), np.ones(3), rtol=1e-4)

    with pytest.raises(grpc.RpcError) as ei:
        predict(m.PredictRequest(model_spec=m.ModelSpec(name="absent")),
                timeout=10)
    assert ei.value.code() == grpc.StatusCode.NOT_FOUND

    multi = ch.unary_unary(
        "/tensorflow.serving.PredictionService/MultiInference",
        request_serializer=lambda b: b,
        response_deserializer=lambda b: b)
    with pytest.raises(grpc.RpcError) as ei:
        multi(b"", timeout=10)
    assert ei.value.code() == grpc.StatusCode.UNIMPLEMENTED


def test_native_status_metadata_health(served):
    srv, ch = served
    predict = _predict_rpc(ch)
    x = np.zeros((1, 16), np.float32)
    predict(m.PredictRequest(
        model_spec=m.ModelSpec(name="mlp"),
        inputs={"x": numpy_to_tensorproto(x)}), timeout=30)

    status = ch.unary_unary(
        "/tensorflow.serving.ModelService/GetModelStatus",
        request_serializer=lambda r: r.encode(),
        response_deserializer=m.GetModelStatusResponse.decode)
    st = status(m.GetModelStatusRequest(
        model_spec=m.ModelSpec(name="mlp")), timeout=10)
    assert st.model_version_status[0].state == m.STATE_AVAILABLE

    meta = ch.unary_unary(
        "/tensorflow.serving.PredictionService/GetModelMetadata",
        request_serializer=lambda r: r.encode(),
        response_deserializer=m.GetModelMetadataResponse.decode)
    md = meta(m.GetModelMetadataRequest(
        model_spec=m.ModelSpec(name="mlp"),
        metadata_field=["signature_def"]), timeout=10)
    assert md.model_spec.name == "mlp"

    from tfservingcache_amd.tfservingproxy.grpc_server import (
        HealthCheckRequest, HealthCheckResponse)
    health = ch.unary_unary(
        "/grpc.health.v1.Health/Check",
        request_serializer=lambda r: r.encode(),
        response_deserializer=HealthCheckResponse.decode)
    assert health(HealthCheckRequest(), timeout=10).status == 1


def test_native_concurrent_and_large(served):
    srv, ch = served
    predict = _predict_rpc(ch)
    rng = np.random.default_rng(1)
    # ~4MB request exercises DATA flow control both ways
    big = rng.standard_normal((8192, 16)).astype(np.float32)
    req_big = m.PredictRequest(
        model_spec=m.ModelSpec(name="mlp"),
        inputs={"x": numpy_to_tensorproto(big)})
    out = tensorproto_to_numpy(
        predict(req_big, timeout=60).outputs["probs"])
    assert out.shape == (8192, 8)

    small = m.PredictRequest(
        model_spec=m.ModelSpec(name="mlp"),
        inputs={"x": numpy_to_tensorproto(
            rng.standard_normal((2, 16)).astype(np.float32))})
    with ThreadPoolExecutor(12) as ex:
        outs = list(ex.map(lambda i: predict(small, timeout=30),
                           range(150)))
    assert len(outs) == 150
    assert srv.fallback_calls() >= 150   # CPU models: all via Python


def test_connection_churn(served):
    """Many short-lived connections: threads are reaped, serving keeps
    working (long-lived-server hygiene)."""
    srv, _ch = served
    req = m.PredictRequest(
        model_spec=m.ModelSpec(name="mlp"),
        inputs={"x": numpy_to_tensorproto(
            np.zeros((1, 16), np.float32))})
    for i in range(30):
        ch = grpc.insecure_channel(f"127.0.0.1:{srv.port}",
                                   options=[("tfsc.conn", i)])
        out = _predict_rpc(ch)(req, timeout=30)
        assert out.outputs["probs"].tensor_shape.dim[0].size == 1
        ch.close()


def test_forwarder_interop_with_native_server(served):
    """The proxy tier's GrpcForwarder (bytes-level grpcio client, the
    node-to-node hop) speaks to the native front-end."""
    from tfservingcache_amd.tfservingproxy.grpc_server import \
        GrpcForwarder
    srv, _ch = served
    fwd = GrpcForwarder(timeout_s=30.0)
    x = np.random.default_rng(2).standard_normal((2, 16)).astype(
        np.float32)
    req = m.PredictRequest(
        model_spec=m.ModelSpec(name="mlp"),
        inputs={"x": numpy_to_tensorproto(x)})
    raw = fwd.call(f"127.0.0.1:{srv.port}",
                   "/tensorflow.serving.PredictionService/Predict",
                   req.encode())
    out = tensorproto_to_numpy(
        m.PredictResponse.decode(raw).outputs["probs"])
    np.testing.assert_allclose(out.sum(-1), np.ones(2), rtol=1e-4)
    fwd.close()


def test_server_native_frontend_config(tmp_path):
    """serving.nativeFrontend switches the cache gRPC implementation;
    the pool registry hooks are attached."""
    from tfservingcache_amd.config import Config
    from tfservingcache_amd.main import Server

    def free_port():
        s = socket.socket()
        s.bind(("127.0.0.1", 0))
        p = s.getsockname()[1]
        s.close()
        return p

    write_model_repo(str(tmp_path / "repo"),
                     [("half_plus_two", 1, "half_plus_two")])
    cfg = Config({
        "cacheRestPort": free_port(),
        "cacheGrpcPort": free_port(),
        "modelProvider": {"type": "diskProvider",
                          "diskProvider": {"baseDir":
                                           str(tmp_path / "repo")}},
        "modelCache": {"hostModelPath": str(tmp_path / "cache"),
                       "size": 10 ** 8},
        "serving": {"maxConcurrentModels": 2, "nativeFrontend": True},
    })
    server = Server(cfg)
    server.start()
    try:
        assert server.cm.pool.on_available is not None
        ch = grpc.insecure_channel(
            f"127.0.0.1:{cfg.get_int('cacheGrpcPort')}")
        predict = _predict_rpc(ch)
        req = m.PredictRequest(
            model_spec=m.ModelSpec(name="half_plus_two"),
            inputs={"x": numpy_to_tensorproto(
                np.array([1.0, 4.0], np.float32))})
        out = tensorproto_to_numpy(predict(req, timeout=30).outputs["y"])
        np.testing.assert_allclose(out, [2.5, 4.0], rtol=1e-5)
        ch.close()
    finally:
        server.stop()
