"""Native (C++ HTTP/1.1) REST front-end. On CPU every request goes
through the Python fallback dispatcher — the HTTP layer and routing
under test are identical to the GPU path; the in-C++ JSON predict is
covered by the gpu-marked tests at the bottom."""
import json
import socket
import urllib.request

import numpy as np
import pytest

pytest.importorskip("torch")

from tfservingcache_amd.cachemanager import (CacheManager, LRUCache,  # noqa: E402
                                             ModelPool, make_cpu_loader)
from tfservingcache_amd.cachemanager.providers import DiskModelProvider  # noqa: E402
from tfservingcache_amd.models import write_model_repo  # noqa: E402
from tfservingcache_amd.tfservingproxy import LocalServingHandler  # noqa: E402
from tfservingcache_amd.tfservingproxy.native_frontend import \
    NativeRestServer  # noqa: E402


@pytest.fixture()
def served(tmp_path):
    write_model_repo(str(tmp_path / "repo"),
                     [("half_plus_two", 1, "half_plus_two"),
                      ("mlp", 1, "mlp")])
    cache = LRUCache(str(tmp_path / "cache"), 10 ** 8)
    pool = ModelPool(make_cpu_loader(cache), max_concurrent_models=3)
    cm = CacheManager(DiskModelProvider(str(tmp_path / "repo")), cache,
                      pool)
    handler = LocalServingHandler(cm)
    srv = NativeRestServer(handler)
    srv.add_insecure_port("[::]:0")
    srv.start()
    yield srv
    srv.stop()


def _post(port, path, payload):
    req = urllib.request.Request(
        f"http://127.0.0.1:{port}{path}",
        data=json.dumps(payload).encode(),
        headers={"Content-Type": "application/json"})
    try:
        with urllib.request.urlopen(req, timeout=30) as r:
            return r.status, json.loads(r.read())
    except urllib.error.HTTPError as e:
        return e.code, json.loads(e.read())


def _get(port, path):
    try:
        with urllib.request.urlopen(
                f"http://127.0.0.1:{port}{path}", timeout=30) as r:
            return r.status, json.loads(r.read())
    except urllib.error.HTTPError as e:
        return e.code, json.loads(e.read())


def test_predict_row_format(served):
    status, body = _post(served.port,
                         "/v1/models/half_plus_two/versions/1:predict",
                         {"instances": [1.0, 2.0, 5.0]})
    assert status == 200
    np.testing.assert_allclose(body["predictions"], [2.5, 3.0, 4.5],
                               rtol=1e-5)


def test_predict_columnar_and_no_version(served):
    status, body = _post(served.port, "/v1/models/half_plus_two:predict",
                         {"inputs": [4.0, 6.0]})
    assert status == 200
    np.testing.assert_allclose(body["outputs"], [4.0, 5.0], rtol=1e-5)


def test_model_status_and_metadata(served):
    # touch the model so it's loaded
    _post(served.port, "/v1/models/mlp:predict",
          {"instances": [[0.0] * 16]})
    status, body = _get(served.port, "/v1/models/mlp")
    assert status == 200
    assert body["model_version_status"][0]["state"] == "AVAILABLE"
    status, body = _get(served.port, "/v1/models/mlp/metadata")
    assert status == 200
    assert "signature_def" in body["metadata"]


def test_malformed_url_404(served):
    status, body = _get(served.port, "/v2/nope")
    assert status == 404
    assert "error" in body


def test_unknown_model_404(served):
    status, body = _post(served.port, "/v1/models/ghost:predict",
                         {"instances": [1.0]})
    assert status == 404
    assert "error" in body


def test_keepalive_pipeline(served):
    """Several requests over ONE connection (HTTP/1.1 keep-alive)."""
    s = socket.create_connection(("127.0.0.1", served.port), timeout=30)
    try:
        payload = json.dumps({"instances": [3.0]}).encode()
        raw = (b"POST /v1/models/half_plus_two:predict HTTP/1.1\r\n"
               b"Host: x\r\nContent-Type: application/json\r\n"
               b"Content-Length: " + str(len(payload)).encode() +
               b"\r\n\r\n" + payload)
        for _ in range(3):
            s.sendall(raw)
            buf = b""
            while b"\r\n\r\n" not in buf:
                buf += s.recv(65536)
            head, rest = buf.split(b"\r\n\r\n", 1)
            clen = int([ln for ln in head.split(b"\r\n")
                        if ln.lower().startswith(b"content-length")
                        ][0].split(b":")[1])
            while len(rest) < clen:
                rest += s.recv(65536)
            body = json.loads(rest[:clen])
            np.testing.assert_allclose(body["predictions"], [3.5])
    finally:
        s.close()


def test_healthz(served):
    with urllib.request.urlopen(
            f"http://127.0.0.1:{served.port}/healthz", timeout=30) as r:
        assert r.status == 200 and r.read() == b"ok"


def test_concurrent_requests(served):
    from concurrent.futures import ThreadPoolExecutor
    def one(i):
        status, body = _post(
            served.port, "/v1/models/half_plus_two:predict",
            {"instances": [float(i)]})
        assert status == 200
        return body["predictions"][0]
    with ThreadPoolExecutor(max_workers=8) as pool:
        got = list(pool.map(one, range(32)))
    np.testing.assert_allclose(got, [i * 0.5 + 2.0 for i in range(32)])


# ---------------------------------------------------------------------------
# GPU: in-C++ JSON predict (no Python on the hot path)
# ---------------------------------------------------------------------------

@pytest.mark.gpu
def test_native_rest_fast_path_gpu(tmp_path):
    import torch
    if not torch.cuda.is_available():
        pytest.skip("needs GPU")
    from tfservingcache_amd.cachemanager import make_gpu_loader
    from tfservingcache_amd.engine.model import load_model_from_dir

    write_model_repo(str(tmp_path / "repo"), [("mlp", 1, "mlp")])
    cache = LRUCache(str(tmp_path / "cache"), 10 ** 8)
    pool = ModelPool(make_gpu_loader(cache, max_batch=8),
                     max_concurrent_models=2)
    cm = CacheManager(DiskModelProvider(str(tmp_path / "repo")), cache,
                      pool)
    handler = LocalServingHandler(cm)
    srv = NativeRestServer(handler)
    srv.add_insecure_port("[::]:0")
    srv.start()
    try:
        def on_avail(name, version, model):
            fast = getattr(getattr(model, "_gpu", None), "_fast", None)
            if fast is not None:
                srv.register_model(name, version, fast)
        pool.on_available = on_avail

        x = np.random.default_rng(0).standard_normal((4, 16)).astype(
            np.float32)
        # first request loads + registers (fallback), then prewarm the
        # contexts via repeats until the fast path serves
        body = {"instances": x.tolist()}
        for _ in range(12):
            status, got = _post(srv.port, "/v1/models/mlp:predict", body)
            assert status == 200
        assert srv.native_hits() > 0, "C++ fast REST path never ran"
        # numerics vs the CPU executor
        cpu = load_model_from_dir(
            str(tmp_path / "cache" / "mlp" / "1"), "mlp", 1)
        want = cpu.predict({"x": x})["probs"]
        np.testing.assert_allclose(np.array(got["predictions"]), want,
                                   rtol=0.05, atol=0.02)

        # named-row format through the fast path too
        rows = [{"x": x[i].tolist()} for i in range(4)]
        status, got2 = _post(srv.port, "/v1/models/mlp:predict",
                             {"instances": rows})
        assert status == 200
        np.testing.assert_allclose(np.array(got2["predictions"]), want,
                                   rtol=0.05, atol=0.02)
    finally:
        srv.stop()


def test_cpp_json_parser_probe():
    """The C++ dense-JSON parser (fast REST path) against known arrays
    and ragged rejects — CPU-runnable via the test binding."""
    import json
    import torch  # noqa: F401
    from tfservingcache_amd.engine import _tfsc_engine as ext
    x = np.random.default_rng(0).standard_normal((3, 5, 2)).astype(
        np.float32)
    dims, vals = ext._rest_parse_probe(json.dumps(x.tolist()).encode(),
                                       False)
    assert list(dims) == [3, 5, 2]
    np.testing.assert_allclose(np.array(vals).reshape(x.shape), x,
                               rtol=1e-6)
    ids = np.array([[1, 2, 3], [4, 5, 6]], dtype=np.int32)
    dims, vals = ext._rest_parse_probe(json.dumps(ids.tolist()).encode(),
                                       True)
    assert list(dims) == [2, 3] and list(vals) == [1, 2, 3, 4, 5, 6]
    for bad in (b"[[1,2],[3]]", b"[[1,2],3]", b"[3,[1,2]]", b'["x"]'):
        with pytest.raises(Exception):
            ext._rest_parse_probe(bad, False)
    # exponent + long-mantissa fallbacks agree with python floats
    dims, vals = ext._rest_parse_probe(
        b"[1e-5, -2.5E3, 0.12345678901234567890, -7]", False)
    np.testing.assert_allclose(
        vals, np.array([1e-5, -2.5e3, 0.12345678901234568, -7.0],
                       dtype=np.float32), rtol=1e-6)


def test_cpp_json_parser_depth_capped():
    """A body of 100k '['s must raise cleanly, not overflow the native
    stack (remote-crash DoS on the REST front-end)."""
    import torch  # noqa: F401
    from tfservingcache_amd.engine import _tfsc_engine as ext
    deep = b"[" * 100000 + b"1" + b"]" * 100000
    with pytest.raises(Exception):
        ext._rest_parse_probe(deep, False)
    # 8-D tensors (realistic rank ceiling) still parse
    body = b"[" * 8 + b"1,2" + b"]" * 8
    dims, vals = ext._rest_parse_probe(body, False)
    assert list(dims) == [1] * 7 + [2]
