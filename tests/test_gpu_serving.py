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
        # per-stage engine metrics appear on the scrape endpoint
        for _ in range(10):
            predict(req, timeout=60)
        scrape = requests.get(
            f"http://127.0.0.1:{server.cache_rest_port}"
            "/monitoring/prometheus/metrics", timeout=10).text
        assert "tfservingcache_engine_stage_seconds" in scrape
        gpu_line = [ln for ln in scrape.splitlines()
                    if ln.startswith(
                        "tfservingcache_engine_stage_seconds_total"
                        '{stage="gpu"}')]
        assert gpu_line and float(gpu_line[0].split()[-1]) > 0
        ch.close()
    finally:
        server.stop()


def test_dynamic_batching_merges_and_matches(tmp_path):
    """Concurrent batch-1 Predicts with server-side batching on: every
    response matches the unbatched result AND the C++ fast path actually
    merged (plan executions < requests served)."""
    import threading
    from tfservingcache_amd.cachemanager import (CacheManager, LRUCache,
                                                 ModelPool, make_gpu_loader)
    from tfservingcache_amd.cachemanager.providers import DiskModelProvider
    from tfservingcache_amd.tfservingproxy import LocalServingHandler

    repo = tmp_path / "repo3"
    write_model_repo(str(repo), [("mlp", 1, "mlp")])
    provider = DiskModelProvider(str(repo))
    cache = LRUCache(str(tmp_path / "cache3"), 10 ** 9)
    loader = make_gpu_loader(cache, device="cuda:0", max_batch=8,
                             batching=True, batch_timeout_s=0.002,
                             n_streams=2)
    pool = ModelPool(loader, max_concurrent_models=2)
    cm = CacheManager(provider, cache, pool, model_fetch_timeout=60.0)
    handler = LocalServingHandler(cm)

    rng = np.random.default_rng(1)
    xs = [rng.standard_normal((1, 16)).astype(np.float32)
          for _ in range(16)]
    reqs = [m.PredictRequest(
        model_spec=m.ModelSpec(name="mlp", version=m.Int64Value(value=1)),
        inputs={"x": numpy_to_tensorproto(x)}).encode() for x in xs]
    # serial pass (merging may or may not kick in) = expected values
    expected = [tensorproto_to_numpy(m.PredictResponse.decode(
        handler.predict_bytes(r)).outputs["probs"]) for r in reqs]

    lm = pool.get_model("mlp", 1)
    runs0, reqs0, _rows0 = lm._gpu._fast.stats()
    errors, mismatches = [], []

    def worker(i):
        for it in range(25):
            j = (i * 25 + it) % len(reqs)
            try:
                out = tensorproto_to_numpy(m.PredictResponse.decode(
                    handler.predict_bytes(reqs[j])).outputs["probs"])
                if not np.allclose(out, expected[j], rtol=1e-2,
                                   atol=1e-3):
                    mismatches.append((i, it))
            except Exception as e:      # noqa: BLE001
                errors.append((i, it, repr(e)))

    threads = [threading.Thread(target=worker, args=(i,))
               for i in range(12)]
    for t in threads:
        t.start()
    for t in threads:
        t.join()
    assert not errors, errors[:5]
    assert not mismatches, mismatches[:5]
    runs1, reqs1, rows1 = lm._gpu._fast.stats()
    served = reqs1 - reqs0
    assert served > 0
    # 12 threads x 25 batch-1 requests against a 2 ms window: plan
    # executions must be well below request count if merging works
    assert runs1 - runs0 < served, (runs1 - runs0, served)


def test_native_frontend_gpu_registry(tmp_path):
    """Registered GPU Predicts bypass Python entirely: native_hits
    must advance once the model's fast contexts are warm."""
    import grpc
    from concurrent.futures import ThreadPoolExecutor
    from tfservingcache_amd.cachemanager import (CacheManager, LRUCache,
                                                 ModelPool,
                                                 make_gpu_loader)
    from tfservingcache_amd.cachemanager.providers import \
        DiskModelProvider
    from tfservingcache_amd.tfservingproxy import LocalServingHandler
    from tfservingcache_amd.tfservingproxy.native_frontend import \
        NativeGrpcServer

    repo = tmp_path / "repo_nf"
    write_model_repo(str(repo), [("mlp", 1, "mlp")])
    cache = LRUCache(str(tmp_path / "cache_nf"), 10 ** 9)
    pool = ModelPool(make_gpu_loader(cache, device="cuda:0", max_batch=8,
                                     n_streams=2),
                     max_concurrent_models=2)
    cm = CacheManager(DiskModelProvider(str(repo)), cache, pool,
                      model_fetch_timeout=60.0)
    handler = LocalServingHandler(cm)
    srv = NativeGrpcServer(handler, workers=4)
    srv.add_insecure_port("[::]:0")
    srv.start()

    def on_avail(name, version, model):
        fast = getattr(getattr(model, "_gpu", None), "_fast", None)
        if fast is not None:
            srv.register_model(name, version, fast)
    pool.on_available = on_avail
    pool.on_unload = (lambda name, version, model:
                      srv.unregister_model(name, version))
    try:
        ch = grpc.insecure_channel(f"127.0.0.1:{srv.port}")
        predict = ch.unary_unary(
            "/tensorflow.serving.PredictionService/Predict",
            request_serializer=lambda r: r.encode(),
            response_deserializer=m.PredictResponse.decode)
        x = np.random.default_rng(0).standard_normal((4, 16)).astype(
            np.float32)
        req = m.PredictRequest(
            model_spec=m.ModelSpec(name="mlp",
                                   version=m.Int64Value(value=1)),
            inputs={"x": numpy_to_tensorproto(x)})
        first = tensorproto_to_numpy(
            predict(req, timeout=60).outputs["probs"])
        # hammer until the fast contexts warm and register
        with ThreadPoolExecutor(8) as ex:
            for _ in range(5):
                outs = [tensorproto_to_numpy(r.outputs["probs"])
                        for r in ex.map(
                            lambda i: predict(req, timeout=60),
                            range(40))]
                for o in outs:
                    np.testing.assert_allclose(o, first, rtol=1e-3,
                                               atol=1e-4)
                if srv.native_hits() > 0:
                    break
        assert srv.native_hits() > 0, (srv.native_hits(),
                                       srv.fallback_calls())
        ch.close()
    finally:
        srv.stop()


def test_eviction_under_concurrent_load(tmp_path):
    """Hammer models from many threads while the pool evicts/reloads:
    no crashes, no wrong results, every request eventually served."""
    import threading
    from tfservingcache_amd.cachemanager import (CacheManager, LRUCache,
                                                 ModelPool, make_gpu_loader)
    from tfservingcache_amd.cachemanager.providers import DiskModelProvider
    from tfservingcache_amd.tfservingproxy import LocalServingHandler

    repo = tmp_path / "repo2"
    names = [f"m{i}" for i in range(4)]
    write_model_repo(str(repo), [(n, 1, "mlp") for n in names])
    provider = DiskModelProvider(str(repo))
    cache = LRUCache(str(tmp_path / "cache2"), 10 ** 9)
    loader = make_gpu_loader(cache, device="cuda:0", max_batch=8,
                             n_streams=2)
    pool = ModelPool(loader, max_concurrent_models=2)   # forces eviction
    cm = CacheManager(provider, cache, pool, model_fetch_timeout=60.0)
    handler = LocalServingHandler(cm)

    rng = np.random.default_rng(0)
    xs = {n: rng.standard_normal((3, 16)).astype(np.float32)
          for n in names}
    expected = {}
    for n in names:
        req = m.PredictRequest(
            model_spec=m.ModelSpec(name=n, version=m.Int64Value(value=1)),
            inputs={"x": numpy_to_tensorproto(xs[n])})
        resp = m.PredictResponse.decode(handler.predict_bytes(req.encode()))
        expected[n] = tensorproto_to_numpy(resp.outputs["probs"])

    errors = []
    mismatches = []

    def worker(seed):
        r = np.random.default_rng(seed)
        for i in range(30):
            n = names[int(r.integers(0, len(names)))]
            req = m.PredictRequest(
                model_spec=m.ModelSpec(name=n,
                                       version=m.Int64Value(value=1)),
                inputs={"x": numpy_to_tensorproto(xs[n])})
            try:
                resp = m.PredictResponse.decode(
                    handler.predict_bytes(req.encode()))
                out = tensorproto_to_numpy(resp.outputs["probs"])
                if not np.allclose(out, expected[n], rtol=1e-2, atol=1e-3):
                    mismatches.append((n, i))
            except Exception as e:      # noqa: BLE001
                errors.append((n, i, repr(e)))

    threads = [threading.Thread(target=worker, args=(s,))
               for s in range(12)]
    for t in threads:
        t.start()
    for t in threads:
        t.join()
    assert not errors, errors[:5]
    assert not mismatches, mismatches[:5]


@pytest.mark.gpu
def test_template_capture_only_numerics(tmp_path):
    """The LRU-churn fast load path end-to-end: content-shared models
    where the SECOND load takes template instantiation + arena restore
    + capture WITHOUT an eager warm run (capture_only). Its predictions
    must match the CPU fp32 reference — this is the exact path the
    headline bench serves from."""
    import os
    import shutil
    import numpy as np
    from tfservingcache_amd.cachemanager import (CacheManager, LRUCache,
                                                 ModelPool,
                                                 make_gpu_loader)
    from tfservingcache_amd.cachemanager.providers import \
        DiskModelProvider
    from tfservingcache_amd.engine.model import load_model_from_dir
    from tfservingcache_amd.models import write_model_repo

    repo = tmp_path / "repo"
    write_model_repo(str(repo), [("m0", 1, "mlp")])
    # hardlink content-identical copies -> shared plan -> template path
    for i in (1, 2):
        dst = repo / f"m{i}" / "1"
        os.makedirs(dst.parent, exist_ok=True)
        shutil.copytree(repo / "m0" / "1", dst,
                        copy_function=os.link)

    cache = LRUCache(str(tmp_path / "cache"), 10 ** 8)
    pool = ModelPool(
        make_gpu_loader(cache, max_batch=8, prewarm_batch=8),
        max_concurrent_models=3)
    cm = CacheManager(DiskModelProvider(str(repo)), cache, pool)

    x = np.random.default_rng(0).standard_normal((8, 16)).astype(
        np.float32)
    outs = []
    for i in range(3):
        lm = cm.ensure_loaded(f"m{i}", 1)
        outs.append(lm.predict({"x": x})["probs"])
    # CPU fp32 reference on the same content
    cpu = load_model_from_dir(str(repo / "m0" / "1"), "cpu", 1)
    want = cpu.predict({"x": x})["probs"]
    for i, got in enumerate(outs):
        np.testing.assert_allclose(got, want, rtol=0.05, atol=0.02,
                                   err_msg=f"model m{i}")
    # the fast C++ context must actually be registered (prewarm ran)
    lm2 = cm.ensure_loaded("m2", 1)
    ctxs = [c for lst in lm2._gpu._contexts.values() for c in lst]
    assert ctxs and all(c.captured for c in ctxs)
    assert any(c.fast_id is not None and c.fast_id >= 0 for c in ctxs)
    assert any(getattr(c, "_from_template", False) for c in ctxs)
