"""GPU end-to-end serving test: full Server (REST + gRPC) over the
CDNA4 engine — the round-end check that the native path is what serves
real requests."""
import socket
import time

import grpc
import numpy as np
import pytest
import requests

pytestmark = pytest.mark.gpu

from tfservingcache_amd.config import Config  # noqa: E402
from tfservingcache_amd.main import Server  # noqa: E402
from tfservingcache_amd.models import write_model_repo  # noqa: E402
from tfservingcache_amd.wire import messages as m  # noqa: E402
from tfservingcache_amd.wire.tensor import (numpy_to_tensorproto,  # noqa: E402
                                            tensorproto_to_numpy)


def free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def test_server_gpu_end_to_end(tmp_path):
    repo = tmp_path / "repo"
    write_model_repo(str(repo), [("half_plus_two", 1, "half_plus_two"),
                                 ("mlp", 1, "mlp")])
    cfg = Config({
        "cacheRestPort": free_port(),
        "cacheGrpcPort": free_port(),
        "modelProvider": {"type": "diskProvider",
                          "diskProvider": {"baseDir": str(repo)}},
        "modelCache": {"hostModelPath": str(tmp_path / "cache"),
                       "size": 10 ** 9},
        "serving": {"maxConcurrentModels": 4},
        "engine": {"gpus": 1, "maxBatch": 8},
    })
    server = Server(cfg)
    server.start()
    try:
        deadline = time.time() + 15
        while time.time() < deadline:
            try:
                r = requests.get(
                    f"http://127.0.0.1:{server.cache_rest_port}/healthz",
                    timeout=1)
                if r.status_code == 200:
                    break
            except requests.RequestException:
                time.sleep(0.1)

        # the engine must be the GPU engine (no CPU fallback on a GPU box)
        assert server.cm.pool.device.startswith("cuda")

        r = requests.post(
            f"http://127.0.0.1:{server.cache_rest_port}"
            "/v1/models/half_plus_two:predict",
            json={"instances": [1.0, 2.0, 5.0]}, timeout=60)
        assert r.status_code == 200, r.text
        np.testing.assert_allclose(r.json()["predictions"], [2.5, 3.0, 4.5],
                                   rtol=1e-2)
        # the loaded model really is GPU-resident
        lm = server.cm.pool.get_model("half_plus_two", 1)
        assert lm is not None and lm._gpu is not None

        # gRPC Predict (bytes-level handler -> C++ fast path once warmed)
        ch = grpc.insecure_channel(
            f"127.0.0.1:{server.cache_grpc_port}")
        predict = ch.unary_unary(
            "/tensorflow.serving.PredictionService/Predict",
            request_serializer=lambda x: x.encode(),
            response_deserializer=m.PredictResponse.decode)
        x = np.random.default_rng(0).standard_normal((4, 16)).astype(
            np.float32)
        req = m.PredictRequest(
            model_spec=m.ModelSpec(name="mlp", version=m.Int64Value(value=1)),
            inputs={"x": numpy_to_tensorproto(x)})
        first = tensorproto_to_numpy(predict(req, timeout=60).outputs["probs"])
        second = tensorproto_to_numpy(
            predict(req, timeout=60).outputs["probs"])   # fast path now
        assert first.shape == (4, 8)
        np.testing.assert_allclose(first, second, rtol=1e-3, atol=1e-4)
        np.testing.assert_allclose(first.sum(-1), np.ones(4), rtol=1e-2)
        mlp = server.cm.pool.get_model("mlp", 1)
        assert mlp._gpu._fast.has_bucket(4)   # C++ path registered
        ch.close()
    finally:
        server.stop()
