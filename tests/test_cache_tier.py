"""End-to-end cache-tier tests: CacheManager + LRU + pool + REST + gRPC.

Mirrors the reference's component-test style (tfservingproxy_test.go uses
real localhost sockets against a fake backend; here the backend is the
real in-process engine over a temp model repo).
"""
import json
import threading
import time

import grpc
import numpy as np
import pytest
from aiohttp.test_utils import TestClient, TestServer

from tfservingcache_amd.cachemanager import (CacheManager, LRUCache,
                                             ModelPool, make_cpu_loader)
from tfservingcache_amd.cachemanager.providers import DiskModelProvider
from tfservingcache_amd.models import write_model_repo
from tfservingcache_amd.tfservingproxy import (LocalServingHandler,
                                               make_cache_grpc_server,
                                               make_cache_rest_app)
from tfservingcache_amd.utils import metrics as mt
from tfservingcache_amd.wire import messages as m
from tfservingcache_amd.wire.tensor import (numpy_to_tensorproto,
                                            tensorproto_to_numpy)


@pytest.fixture()
def stack(tmp_path):
    repo = tmp_path / "repo"
    cache_dir = tmp_path / "cache"
    write_model_repo(str(repo), [
        ("half_plus_two", 123, "half_plus_two"),
        ("half_plus_two", 124, "half_plus_two"),
        ("mlp", 1, "mlp"),
    ])
    provider = DiskModelProvider(str(repo))
    cache = LRUCache(str(cache_dir), max_size_bytes=10 * 1024 * 1024)
    pool = ModelPool(make_cpu_loader(cache), max_concurrent_models=2)
    cm = CacheManager(provider, cache, pool, model_fetch_timeout=10.0)
    handler = LocalServingHandler(cm)
    return cm, handler


def test_predict_via_handler(stack):
    cm, handler = stack
    req = m.PredictRequest(
        model_spec=m.ModelSpec(name="half_plus_two",
                               version=m.Int64Value(value=123)),
        inputs={"x": numpy_to_tensorproto(
            np.array([1.0, 2.0, 5.0], dtype=np.float32))})
    resp = handler.predict(req)
    out = tensorproto_to_numpy(resp.outputs["y"])
    np.testing.assert_allclose(out, [2.5, 3.0, 4.5])
    assert resp.model_spec.version.value == 123


def test_version_latest_resolution(stack):
    cm, handler = stack
    req = m.PredictRequest(
        model_spec=m.ModelSpec(name="half_plus_two"),
        inputs={"x": numpy_to_tensorproto(
            np.array([4.0], dtype=np.float32))})
    resp = handler.predict(req)
    assert resp.model_spec.version.value == 124  # latest


def test_model_status_and_reload(stack):
    cm, handler = stack
    handler.predict(m.PredictRequest(
        model_spec=m.ModelSpec(name="mlp", version=m.Int64Value(value=1)),
        inputs={"x": numpy_to_tensorproto(
            np.zeros((1, 16), dtype=np.float32))}))
    st = handler.get_model_status(m.GetModelStatusRequest(
        model_spec=m.ModelSpec(name="mlp")))
    assert st.model_version_status[0].state == m.STATE_AVAILABLE

    with pytest.raises(Exception):
        handler.get_model_status(m.GetModelStatusRequest(
            model_spec=m.ModelSpec(name="missing_model")))


def test_lru_eviction_on_pool(stack):
    """maxConcurrentModels=2: a third model evicts the LRU one from the
    pool (the reference pushed only the MRU prefix — cachemanager.go:168)."""
    cm, handler = stack

    def predict(name, version, x):
        return handler.predict(m.PredictRequest(
            model_spec=m.ModelSpec(name=name,
                                   version=m.Int64Value(value=version)),
            inputs={"x": numpy_to_tensorproto(x)}))

    predict("half_plus_two", 123, np.array([1.0], dtype=np.float32))
    predict("mlp", 1, np.zeros((1, 16), dtype=np.float32))
    predict("half_plus_two", 124, np.array([1.0], dtype=np.float32))
    # pool only holds 2: (hpt,123) must be gone
    time.sleep(0.1)
    states = {(n, v): s for (n, v), s in cm.pool.model_states().items()}
    assert (("half_plus_two", 123) not in states or
            states[("half_plus_two", 123)] != m.STATE_AVAILABLE)
    assert states[("half_plus_two", 124)] == m.STATE_AVAILABLE
    assert states[("mlp", 1)] == m.STATE_AVAILABLE


def test_concurrent_misses_single_flight(stack):
    cm, handler = stack
    results = []
    errs = []

    def worker():
        try:
            resp = handler.predict(m.PredictRequest(
                model_spec=m.ModelSpec(name="mlp",
                                       version=m.Int64Value(value=1)),
                inputs={"x": numpy_to_tensorproto(
                    np.zeros((2, 16), dtype=np.float32))}))
            results.append(resp)
        except Exception as e:      # noqa: BLE001
            errs.append(e)

    threads = [threading.Thread(target=worker) for _ in range(8)]
    for t in threads:
        t.start()
    for t in threads:
        t.join()
    assert not errs
    assert len(results) == 8


# ---------------------------------------------------------------------------
# REST round trip over a real socket
# ---------------------------------------------------------------------------
async def test_rest_predict(stack, aiohttp_client=None):
    cm, handler = stack
    app = make_cache_rest_app(handler, metrics_render=mt.render)
    server = TestServer(app)
    client = TestClient(server)
    await client.start_server()
    try:
        resp = await client.post(
            "/v1/models/half_plus_two/versions/123:predict",
            json={"instances": [1.0, 2.0, 5.0]})
        assert resp.status == 200
        body = await resp.json()
        assert body == {"predictions": [2.5, 3.0, 4.5]}

        # columnar + named
        resp = await client.post(
            "/v1/models/mlp:predict",
            json={"inputs": {"x": [[0.0] * 16]}})
        assert resp.status == 200
        body = await resp.json()
        assert len(body["outputs"][0]) == 8

        # status endpoint
        resp = await client.get("/v1/models/half_plus_two/versions/123")
        body = await resp.json()
        assert body["model_version_status"][0]["state"] == "AVAILABLE"

        # metadata
        resp = await client.get("/v1/models/half_plus_two/metadata")
        body = await resp.json()
        assert "signature_def" in body["metadata"]

        # malformed URL -> 404 (tfservingproxy_test.go:142-170 analog)
        resp = await client.post("/v1/foo/bar:predict", json={})
        assert resp.status == 404

        # unknown model -> 404
        resp = await client.post("/v1/models/nope:predict",
                                 json={"instances": [1.0]})
        assert resp.status == 404

        # metrics scrape
        resp = await client.get("/monitoring/prometheus/metrics")
        text = await resp.text()
        assert "tfservingcache_cache_total" in text
        assert "tfservingcache_proxy_requests_total" in text
    finally:
        await client.close()


# ---------------------------------------------------------------------------
# gRPC round trip over a real socket
# ---------------------------------------------------------------------------
def test_grpc_predict_and_status(stack):
    cm, handler = stack
    server, health = make_cache_grpc_server(handler)
    port = server.add_insecure_port("127.0.0.1:0")
    server.start()
    try:
        ch = grpc.insecure_channel(f"127.0.0.1:{port}")
        predict = ch.unary_unary(
            "/tensorflow.serving.PredictionService/Predict",
            request_serializer=lambda r: r.encode(),
            response_deserializer=m.PredictResponse.decode)
        req = m.PredictRequest(
            model_spec=m.ModelSpec(name="half_plus_two",
                                   version=m.Int64Value(value=123)),
            inputs={"x": numpy_to_tensorproto(
                np.array([1.0, 2.0, 5.0], dtype=np.float32))})
        resp = predict(req, timeout=10)
        np.testing.assert_allclose(
            tensorproto_to_numpy(resp.outputs["y"]), [2.5, 3.0, 4.5])

        status = ch.unary_unary(
            "/tensorflow.serving.ModelService/GetModelStatus",
            request_serializer=lambda r: r.encode(),
            response_deserializer=m.GetModelStatusResponse.decode)
        st = status(m.GetModelStatusRequest(
            model_spec=m.ModelSpec(name="half_plus_two")), timeout=10)
        assert any(s.state == m.STATE_AVAILABLE
                   for s in st.model_version_status)

        # health
        from tfservingcache_amd.tfservingproxy.grpc_server import (
            HealthCheckRequest, HealthCheckResponse)
        check = ch.unary_unary(
            "/grpc.health.v1.Health/Check",
            request_serializer=lambda r: r.encode(),
            response_deserializer=HealthCheckResponse.decode)
        hr = check(HealthCheckRequest(), timeout=10)
        assert hr.status == 1  # SERVING

        # MultiInference -> UNIMPLEMENTED (tfservingproxy.go:215-217)
        multi = ch.unary_unary(
            "/tensorflow.serving.PredictionService/MultiInference",
            request_serializer=lambda b: b,
            response_deserializer=lambda b: b)
        with pytest.raises(grpc.RpcError) as ei:
            multi(b"", timeout=10)
        assert ei.value.code() == grpc.StatusCode.UNIMPLEMENTED

        # SessionRun
        run = ch.unary_unary(
            "/tensorflow.serving.SessionService/SessionRun",
            request_serializer=lambda r: r.encode(),
            response_deserializer=m.SessionRunResponse.decode)
        rr = run(m.SessionRunRequest(
            model_spec=m.ModelSpec(name="half_plus_two",
                                   version=m.Int64Value(value=123)),
            feed=[m.NamedTensorProto(
                name="x", tensor=numpy_to_tensorproto(
                    np.array([10.0], dtype=np.float32)))],
            fetch=["y:0"]), timeout=10)
        np.testing.assert_allclose(
            tensorproto_to_numpy(rr.tensor[0].tensor), [7.0])

        ch.close()
    finally:
        server.stop(None)


def test_classify_regress(stack):
    cm, handler = stack
    ex = m.Example(features=m.Features())
    ex.features.feature["x"] = m.Feature(
        float_list=m.FloatList(value=[3.0]))
    req = m.RegressionRequest(
        model_spec=m.ModelSpec(name="half_plus_two",
                               version=m.Int64Value(value=123)),
        input=m.Input(example_list=m.ExampleList(examples=[ex])))
    resp = handler.regress(req)
    assert abs(resp.result.regressions[0].value - 3.5) < 1e-6

    creq = m.ClassificationRequest(
        model_spec=m.ModelSpec(name="mlp", version=m.Int64Value(value=1)),
        input=m.Input(example_list=m.ExampleList(examples=[
            m.Example(features=m.Features(feature={
                "x": m.Feature(float_list=m.FloatList(
                    value=[0.1] * 16))}))])))
    cresp = handler.classify(creq)
    classes = cresp.result.classifications[0].classes
    assert len(classes) == 8
    assert abs(sum(c.score for c in classes) - 1.0) < 1e-4


def test_version_labels(stack):
    cm, handler = stack
    req = m.ReloadConfigRequest(config=m.ModelServerConfig(
        model_config_list=m.ModelConfigList(config=[m.ModelConfig(
            name="half_plus_two", base_path="/x",
            model_version_policy=m.ServableVersionPolicy(
                specific=m.ServableVersionPolicySpecific(versions=[123])),
            version_labels={"stable": 123})])))
    handler.handle_reload_config(req)
    presp = handler.predict(m.PredictRequest(
        model_spec=m.ModelSpec(name="half_plus_two",
                               version_label="stable"),
        inputs={"x": numpy_to_tensorproto(
            np.array([2.0], dtype=np.float32))}))
    assert presp.model_spec.version.value == 123


def test_pool_byte_budget(tmp_path):
    """engine.hbmPoolBytes caps resident bytes, not just model count."""
    repo = tmp_path / "r2"
    write_model_repo(str(repo), [("m_a", 1, "mlp"), ("m_b", 1, "mlp"),
                                 ("m_c", 1, "mlp")])
    provider = DiskModelProvider(str(repo))
    cache = LRUCache(str(tmp_path / "c2"), 10 ** 8)
    sizes = {}

    def size_hint(name, version):
        e = cache.get(name, version)
        return e.size_on_disk if e else 0

    pool = ModelPool(make_cpu_loader(cache), max_concurrent_models=10,
                     max_bytes=None, size_hint=size_hint)
    cm = CacheManager(provider, cache, pool)
    handler = LocalServingHandler(cm)
    x = np.zeros((1, 16), dtype=np.float32)
    handler.predict(m.PredictRequest(
        model_spec=m.ModelSpec(name="m_a", version=m.Int64Value(value=1)),
        inputs={"x": numpy_to_tensorproto(x)}))
    one_size = cache.get("m_a", 1).size_on_disk
    # now budget for ~1.5 models
    pool.max_bytes = int(one_size * 1.5)
    for name in ("m_b", "m_c"):
        handler.predict(m.PredictRequest(
            model_spec=m.ModelSpec(name=name, version=m.Int64Value(value=1)),
            inputs={"x": numpy_to_tensorproto(x)}))
    time.sleep(0.2)
    avail = [mid for mid, s in cm.pool.model_states().items()
             if s == m.STATE_AVAILABLE]
    assert len(avail) == 1          # byte budget, not count, limited it


def test_warmup_requests_executed(tmp_path):
    """assets.extra/tf_serving_warmup_requests run at load (TF Serving
    warmup parity)."""
    import os
    from tfservingcache_amd.engine import warmup as wu
    from tfservingcache_amd.engine.model import load_model_from_dir
    from tfservingcache_amd.engine.savedmodel import write_saved_model
    from tfservingcache_amd.models import build_mlp

    vdir = tmp_path / "wm" / "1"
    write_saved_model(build_mlp(seed=4), str(vdir))
    req = m.PredictRequest(
        model_spec=m.ModelSpec(name="wm", version=m.Int64Value(value=1)),
        inputs={"x": numpy_to_tensorproto(
            np.zeros((2, 16), dtype=np.float32))})
    rec = wu.PredictionLog(predict_log=wu.PredictLog(request=req)).encode()
    wu.write_tfrecord(str(vdir / wu.WARMUP_PATH), [rec, rec])

    lm = load_model_from_dir(str(vdir), "wm", 1)
    calls = []
    orig = lm.predict
    lm.predict = lambda *a, **k: (calls.append(1), orig(*a, **k))[1]
    n = wu.run_warmup(lm, str(vdir))
    assert n == 2
    assert len(calls) == 2
    # roundtrip of the tfrecord reader
    assert len(list(wu.read_tfrecords(str(vdir / wu.WARMUP_PATH)))) == 2


def test_plan_cache_shares_hardlinked_models(tmp_path):
    """Hardlinked copies of one SavedModel compile once (inode-keyed
    plan cache); byte-identical fresh copies (distinct inodes, e.g.
    S3 downloads) ALSO share via the content-hash fallback key; a
    model with different bytes gets its own plan."""
    import os
    import shutil
    from tfservingcache_amd.engine.model import load_model_from_dir
    from tfservingcache_amd.models import write_model_repo

    write_model_repo(str(tmp_path), [("a", 1, "half_plus_two"),
                                     ("d", 1, "mlp")])
    src = tmp_path / "a" / "1"
    dst = tmp_path / "b" / "1"
    os.makedirs(dst.parent, exist_ok=True)
    shutil.copytree(src, dst, copy_function=os.link)

    lm_a = load_model_from_dir(str(src), "a", 1)
    lm_b = load_model_from_dir(str(dst), "b", 1)
    assert lm_a.plan is lm_b.plan       # same inode -> shared plan

    # identical content, fresh inode -> shared via the content key
    dst2 = tmp_path / "c" / "1"
    os.makedirs(dst2.parent, exist_ok=True)
    shutil.copytree(src, dst2)
    lm_c = load_model_from_dir(str(dst2), "c", 1)
    assert lm_c.plan is lm_a.plan

    # genuinely different bytes -> different plan
    lm_d = load_model_from_dir(str(tmp_path / "d" / "1"), "d", 1)
    assert lm_d.plan is not lm_a.plan

    import numpy as np
    out_a = lm_a.predict({"x": np.array([2.0], np.float32)})
    out_c = lm_c.predict({"x": np.array([2.0], np.float32)})
    np.testing.assert_allclose(list(out_a.values())[0],
                               list(out_c.values())[0])


def test_rebuild_from_disk_after_restart(tmp_path):
    """modelCache.rebuildFromDisk re-indexes an existing cache dir: a
    restarted node serves what is already on disk instead of leaking
    the files (SURVEY.md §2.3 — the reference leaves stale files and
    an empty map)."""
    from tfservingcache_amd.cachemanager import LRUCache
    from tfservingcache_amd.models import write_model_repo

    cache_dir = tmp_path / "c"
    write_model_repo(str(cache_dir), [("m1", 1, "half_plus_two"),
                                      ("m1", 2, "half_plus_two"),
                                      ("m2", 7, "mlp")])
    # fresh process, same dir
    cache = LRUCache(str(cache_dir), 10 ** 9, rebuild_from_disk=True)
    ids = {(e.name, e.version) for e in cache.list_models()}
    assert ids == {("m1", 1), ("m1", 2), ("m2", 7)}
    assert cache.current_size > 0
    assert all(e.size_on_disk > 0 for e in cache.list_models())
    # without the flag the index starts empty (reference behavior)
    cache2 = LRUCache(str(cache_dir), 10 ** 9)
    assert cache2.list_models() == []


@pytest.mark.slow
def test_lru_thrash_soak(tmp_path):
    """CPU miniature of the headline workload: more models than the
    pool can hold, 8 concurrent threads, EVERY response checked against
    that model's own numpy forward. Catches races in the single-flight
    cold-load path under constant eviction (the reference's whole-
    download mutex makes this trivially serial — cachemanager.go:114)."""
    n_models, pool_cap = 10, 3
    repo = tmp_path / "repo"
    write_model_repo(
        str(repo), [(f"mlp{i}", 1, "mlp") for i in range(n_models)],
        builder_kwargs={f"mlp{i}": {"seed": i} for i in range(n_models)})
    provider = DiskModelProvider(str(repo))
    cache = LRUCache(str(tmp_path / "cache"), max_size_bytes=10 ** 9)
    pool = ModelPool(make_cpu_loader(cache),
                     max_concurrent_models=pool_cap)
    cm = CacheManager(provider, cache, pool, model_fetch_timeout=60.0)
    handler = LocalServingHandler(cm)

    x = np.random.default_rng(99).standard_normal((2, 16)).astype(
        np.float32)
    want = {}
    for i in range(n_models):
        rng = np.random.default_rng(i)
        w1 = rng.standard_normal((16, 32), dtype=np.float32) * 0.3
        b1 = rng.standard_normal(32, dtype=np.float32) * 0.1
        w2 = rng.standard_normal((32, 8), dtype=np.float32) * 0.3
        b2 = rng.standard_normal(8, dtype=np.float32) * 0.1
        h = np.maximum(x @ w1 + b1, 0.0)
        logits = h @ w2 + b2
        e = np.exp(logits - logits.max(axis=-1, keepdims=True))
        want[f"mlp{i}"] = e / e.sum(axis=-1, keepdims=True)

    order = [f"mlp{int(i)}"
             for i in np.random.default_rng(0).integers(0, n_models, 400)]
    it = iter(order)
    lock = threading.Lock()
    errors = []

    def worker():
        while True:
            with lock:
                name = next(it, None)
            if name is None:
                return
            try:
                req = m.PredictRequest(
                    model_spec=m.ModelSpec(
                        name=name, version=m.Int64Value(value=1)),
                    inputs={"x": numpy_to_tensorproto(x)})
                resp = handler.predict(req)
                out = tensorproto_to_numpy(
                    next(iter(resp.outputs.values())))
                np.testing.assert_allclose(out, want[name],
                                           rtol=2e-4, atol=2e-5)
            except Exception as ex:       # noqa: BLE001
                errors.append((name, repr(ex)))
                return

    threads = [threading.Thread(target=worker) for _ in range(8)]
    for t in threads:
        t.start()
    for t in threads:
        t.join(timeout=120)
    assert not errors, errors[:3]
    # the pool may transiently exceed the cap by in-flight loads it
    # refuses to cancel (modelpool.reload). A hit-path request never
    # reloads, so drive reconciliation directly: once the in-flight
    # loads land, a declarative reload must converge to the cap
    deadline = time.time() + 20
    while time.time() < deadline:
        cm._reload_pool()               # noqa: SLF001
        if len(pool._entries) <= pool_cap:  # noqa: SLF001
            break
        time.sleep(0.05)
    assert len(pool._entries) <= pool_cap  # noqa: SLF001


def test_shipped_testclient_runs(stack):
    """The shipped scripts/testclient.py (parity with the reference's
    cmd/testclient) works against a live cache gRPC server."""
    import subprocess
    import sys
    from pathlib import Path
    cm, handler = stack
    server, _health = make_cache_grpc_server(handler)
    port = server.add_insecure_port("127.0.0.1:0")
    server.start()
    try:
        repo_root = Path(__file__).resolve().parent.parent
        out = subprocess.run(
            [sys.executable, str(repo_root / "scripts" / "testclient.py"),
             f"127.0.0.1:{port}", "half_plus_two", "123"],
            capture_output=True, text=True, timeout=60)
        assert out.returncode == 0, out.stderr[-2000:]
        assert "Predict output" in out.stdout
        assert "2.5" in out.stdout and "4.5" in out.stdout
        assert "Classify:" in out.stdout
    finally:
        server.stop(0)


async def test_rest_label_routing(stack, aiohttp_client=None):
    """TF Serving's /labels/<label> REST grammar (beyond the
    reference's /versions-only regex): the label resolves through the
    ReloadConfig version_labels map."""
    cm, handler = stack
    handler.handle_reload_config(m.ReloadConfigRequest(
        config=m.ModelServerConfig(
            model_config_list=m.ModelConfigList(config=[m.ModelConfig(
                name="half_plus_two", base_path="/x",
                model_version_policy=m.ServableVersionPolicy(
                    specific=m.ServableVersionPolicySpecific(
                        versions=[123])),
                version_labels={"canary": 123})]))))
    app = make_cache_rest_app(handler, metrics_render=mt.render)
    server = TestServer(app)
    client = TestClient(server)
    await client.start_server()
    try:
        resp = await client.post(
            "/v1/models/half_plus_two/labels/canary:predict",
            json={"instances": [1.0, 2.0, 5.0]})
        assert resp.status == 200
        body = await resp.json()
        assert body == {"predictions": [2.5, 3.0, 4.5]}
        # unknown label -> error, not silently latest
        resp = await client.post(
            "/v1/models/half_plus_two/labels/nope:predict",
            json={"instances": [1.0]})
        assert resp.status in (400, 404)
    finally:
        await client.close()
