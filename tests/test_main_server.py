"""End-to-end Server (composition root) tests on real ports."""
import os
import socket
import time

import grpc
import numpy as np
import pytest
import requests

from tfservingcache_amd.config import Config
from tfservingcache_amd.main import Server
from tfservingcache_amd.models import write_model_repo
from tfservingcache_amd.wire import messages as m
from tfservingcache_amd.wire.tensor import (numpy_to_tensorproto,
                                            tensorproto_to_numpy)


def free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def make_cfg(tmp_path, with_discovery=False, name="n1"):
    repo = tmp_path / "repo"
    if not repo.exists():
        write_model_repo(str(repo), [("half_plus_two", 1, "half_plus_two")])
    values = {
        "cacheRestPort": free_port(),
        "cacheGrpcPort": free_port(),
        "proxyRestPort": free_port(),
        "proxyGrpcPort": free_port(),
        "modelProvider": {"type": "diskProvider",
                          "diskProvider": {"baseDir": str(repo)}},
        "modelCache": {"hostModelPath": str(tmp_path / f"cache_{name}"),
                       "size": 10 ** 8},
        "serving": {"maxConcurrentModels": 2},
        "proxy": {"replicasPerModel": 1, "advertiseHost": "127.0.0.1"},
    }
    if with_discovery:
        values["serviceDiscovery"] = {
            "type": "file", "heartbeatTTL": 0.5,
            "file": {"directory": str(tmp_path / "cluster")},
        }
    return Config(values)


def wait_http(port, path="/healthz", timeout=30):
    deadline = time.time() + timeout
    while time.time() < deadline:
        try:
            r = requests.get(f"http://127.0.0.1:{port}{path}", timeout=1)
            if r.status_code == 200:
                return
        except requests.RequestException:
            time.sleep(0.05)
    raise TimeoutError(f"port {port} never became ready")


def test_server_cache_tier_only(tmp_path):
    cfg = make_cfg(tmp_path)
    server = Server(cfg)
    server.start()
    try:
        assert server.cluster is None     # proxy disabled (main.go:103-105)
        wait_http(server.cache_rest_port)
        r = requests.post(
            f"http://127.0.0.1:{server.cache_rest_port}"
            "/v1/models/half_plus_two:predict",
            json={"instances": [1.0, 2.0, 5.0]}, timeout=10)
        assert r.status_code == 200
        assert r.json() == {"predictions": [2.5, 3.0, 4.5]}

        # metrics endpoint
        r = requests.get(
            f"http://127.0.0.1:{server.cache_rest_port}"
            "/monitoring/prometheus/metrics", timeout=5)
        assert "tfservingcache_cache_total" in r.text

        # gRPC health
        ch = grpc.insecure_channel(
            f"127.0.0.1:{server.cache_grpc_port}")
        from tfservingcache_amd.tfservingproxy.grpc_server import (
            HealthCheckRequest, HealthCheckResponse)
        check = ch.unary_unary(
            "/grpc.health.v1.Health/Check",
            request_serializer=lambda x: x.encode(),
            response_deserializer=HealthCheckResponse.decode)
        assert check(HealthCheckRequest(), timeout=5).status == 1
        ch.close()
    finally:
        server.stop()


def test_server_proxy_tier_two_nodes(tmp_path):
    cfg1 = make_cfg(tmp_path, with_discovery=True, name="n1")
    cfg2 = make_cfg(tmp_path, with_discovery=True, name="n2")
    s1 = Server(cfg1)
    s2 = Server(cfg2)
    s1.start()
    s2.start()
    try:
        wait_http(s1.cache_rest_port)
        wait_http(s2.cache_rest_port)
        deadline = time.time() + 10
        while time.time() < deadline and (
                s1.cluster.n_members() < 2 or s2.cluster.n_members() < 2):
            time.sleep(0.1)
        assert s1.cluster.n_members() == 2
        assert s2.cluster.n_members() == 2

        # REST predict through EITHER proxy reaches the owning node
        for srv in (s1, s2):
            r = requests.post(
                f"http://127.0.0.1:{srv.proxy_rest_port}"
                "/v1/models/half_plus_two/versions/1:predict",
                json={"instances": [4.0]}, timeout=15)
            assert r.status_code == 200, r.text
            assert r.json() == {"predictions": [4.0]}

        # gRPC predict through the proxy tier
        ch = grpc.insecure_channel(f"127.0.0.1:{s1.proxy_grpc_port}")
        predict = ch.unary_unary(
            "/tensorflow.serving.PredictionService/Predict",
            request_serializer=lambda x: x.encode(),
            response_deserializer=m.PredictResponse.decode)
        resp = predict(m.PredictRequest(
            model_spec=m.ModelSpec(name="half_plus_two",
                                   version=m.Int64Value(value=1)),
            inputs={"x": numpy_to_tensorproto(
                np.array([10.0], dtype=np.float32))}), timeout=15)
        np.testing.assert_allclose(
            tensorproto_to_numpy(resp.outputs["y"]), [7.0])
        ch.close()

        # both proxies route the same model to the SAME owner (ring
        # determinism across nodes)
        owner1 = s1.cluster.node_for_key("some_model", 1)
        owner2 = s2.cluster.node_for_key("some_model", 1)
        assert owner1.serialize() == owner2.serialize()
    finally:
        s1.stop()
        s2.stop()


def test_config_env_override(tmp_path, monkeypatch):
    monkeypatch.setenv("TFSC_SERVING_GRPCHOST", "otherhost:9999")
    monkeypatch.setenv("TFSC_MODELCACHE_SIZE", "12345")
    cfg = Config({"serving": {"grpcHost": "localhost:8500"}})
    assert cfg.get_string("serving.grpcHost") == "otherhost:9999"
    assert cfg.get_int("modelCache.size") == 12345
    assert cfg.get_string("healthProbe.modelName") == \
        "__TFSERVINGCACHE_PROBE_CHECK__"


def test_warm_handoff_on_ring_change(tmp_path):
    """When membership shifts a cached model's ownership away from this
    node, the old owner nudges the new owner (GetModelMetadata over the
    cache gRPC surface) so it loads warm instead of missing cold."""
    from tfservingcache_amd.taskhandler.discovery.base import ServingService

    def mock_cfg(name):
        cfg = make_cfg(tmp_path, name=name, with_discovery=False)
        values = dict(cfg._values)      # noqa: SLF001
        values["serviceDiscovery"] = {"type": "mock"}
        return Config(values)

    a = Server(mock_cfg("hand_a"))
    b = Server(mock_cfg("hand_b"))
    a.start()
    b.start()
    try:
        wait_http(a.cache_rest_port)
        wait_http(b.cache_rest_port)
        a_svc = a._self_service         # noqa: SLF001
        b_svc = ServingService("127.0.0.1", b.cache_rest_port,
                               b.cache_grpc_port)
        # seed: A alone owns everything
        a.discovery.push([a_svc])
        # load the model on A
        r = requests.post(
            f"http://127.0.0.1:{a.cache_rest_port}"
            "/v1/models/half_plus_two:predict",
            json={"instances": [1.0]}, timeout=30)
        assert r.status_code == 200, r.text
        assert a.cm.pool.get_model("half_plus_two", 1) is not None
        assert b.cm.pool.get_model("half_plus_two", 1) is None

        # membership flips: only B in the ring -> A hands the model off
        a.discovery.push([b_svc])
        deadline = time.time() + 15
        while time.time() < deadline:
            if b.cm.pool.get_model("half_plus_two", 1) is not None:
                break
            time.sleep(0.1)
        assert b.cm.pool.get_model("half_plus_two", 1) is not None
    finally:
        a.stop()
        b.stop()


def test_logging_config(capsys):
    """logging.{level,format} config surface (cfg.go:28-60 analog)."""
    import json as json_mod
    import logging
    from tfservingcache_amd.utils.logsetup import setup_logging

    setup_logging(Config({"logging": {"level": "debug",
                                      "format": "json"}}))
    assert logging.getLogger().level == logging.DEBUG
    logging.getLogger("tfsc.test").warning("hello %s", "world")
    line = capsys.readouterr().err.strip().splitlines()[-1]
    entry = json_mod.loads(line)
    assert entry["msg"] == "hello world"
    assert entry["level"] == "warning"

    setup_logging(Config({"logging": {"level": "warning"}}))
    assert logging.getLogger().level == logging.WARNING
    # restore defaults for other tests
    setup_logging(Config({}))


def test_example_config_parses_and_reads():
    """config.yaml.example ships as the Docker image's default config —
    it must load through Config and yield the documented values."""
    from pathlib import Path
    from tfservingcache_amd.config import Config
    path = Path(__file__).resolve().parent.parent / "config.yaml.example"
    cfg = Config.load(str(path))
    assert cfg.get_int("proxyRestPort") == 8093
    assert cfg.get_int("cacheGrpcPort") == 8095
    assert cfg.get_string("modelProvider.type") == "diskProvider"
    assert cfg.get_int("serving.maxConcurrentModels") == 16
    assert cfg.get_int("serving.grpcMaxMsgSize") == 16 * 1024 * 1024
    assert cfg.get_int("engine.maxBatch") == 64
    assert cfg.get_int("proxy.replicasPerModel") == 2
    assert cfg.get_string("healthProbe.modelName") == \
        "__TFSERVINGCACHE_PROBE_CHECK__"
    assert cfg.get_string("logging.level") == "info"
