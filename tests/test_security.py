"""Hardening regressions: model-name path traversal, truncated wire
bytes, and single-flight lock lifetime (round-1 advisor findings)."""
import threading
import time

import numpy as np

import pytest

from tfservingcache_amd.cachemanager import (CacheManager, LRUCache,
                                             ModelPool, make_cpu_loader)
from tfservingcache_amd.cachemanager.modelprovider import (
    InvalidModelNameError, validate_model_name)
from tfservingcache_amd.cachemanager.providers import DiskModelProvider
from tfservingcache_amd.models import write_model_repo
from tfservingcache_amd.wire import messages as m


# -- model-name validation ---------------------------------------------------

@pytest.mark.parametrize("bad", [
    "", ".", "..", "../x", "a/b", "a/../b", "\\..\\x", "a\\b",
    "x\x00y", "m" * 513,
])
def test_invalid_model_names_rejected(bad):
    with pytest.raises(InvalidModelNameError):
        validate_model_name(bad)


@pytest.mark.parametrize("ok", ["m", "half_plus_two", "model-1.2_x",
                                "UPPER", "00042"])
def test_valid_model_names_accepted(ok):
    assert validate_model_name(ok) == ok


def test_traversal_name_cannot_delete_repo(tmp_path):
    """A gRPC-style name '../model_repo/<m>' must be rejected before it
    reaches the disk provider's rmtree+copytree (advisor high #2)."""
    repo = tmp_path / "repo"
    write_model_repo(str(repo), [("victim", 1, "half_plus_two")])
    cache = LRUCache(str(tmp_path / "cache"), 10 ** 8)
    pool = ModelPool(make_cpu_loader(cache), max_concurrent_models=2)
    cm = CacheManager(DiskModelProvider(str(repo)), cache, pool)
    with pytest.raises(InvalidModelNameError):
        cm.ensure_loaded("../repo/victim", 1)
    assert (repo / "victim" / "1").is_dir()     # untouched


def test_provider_rejects_traversal_directly(tmp_path):
    prov = DiskModelProvider(str(tmp_path))
    with pytest.raises(InvalidModelNameError):
        prov.load_model("../evil", 1, str(tmp_path / "cache"))
    with pytest.raises(InvalidModelNameError):
        prov.model_size("a/b", 1)
    with pytest.raises(InvalidModelNameError):
        prov.latest_version("..")


# -- truncated protobuf bytes ------------------------------------------------

def _sample_request() -> bytes:
    import numpy as np
    from tfservingcache_amd.wire.tensor import numpy_to_tensorproto
    req = m.PredictRequest(
        model_spec=m.ModelSpec(name="mm", version=m.Int64Value(value=3)),
        inputs={"x": numpy_to_tensorproto(
            np.arange(64, dtype=np.float32).reshape(4, 16))})
    return req.encode()


def test_truncated_request_raises_cleanly():
    data = _sample_request()
    for cut in range(1, len(data)):
        try:
            m.PredictRequest.decode(data[:cut])
        except ValueError:
            continue
        # a successful decode of a prefix is only OK if the prefix
        # happens to end exactly on a field boundary — re-encode must
        # then be a prefix-consistent message, never garbage memory
    # full message still decodes
    got = m.PredictRequest.decode(data)
    assert got.model_spec.name == "mm"


def test_truncated_tensor_content_rejected():
    """Declared tensor_content length beyond the buffer must raise, not
    silently truncate (advisor low #5)."""
    data = bytearray(_sample_request())
    # chop the last 10 bytes: tensor_content's declared length now
    # exceeds the remaining buffer
    with pytest.raises(ValueError):
        m.PredictRequest.decode(bytes(data[:-10]))


def test_peek_spec_truncated():
    data = _sample_request()
    spec = m.peek_model_spec(data)
    assert spec.name == "mm"
    with pytest.raises((ValueError, IndexError)):
        m.peek_model_spec(data[:3])


def test_cpp_peek_spec_truncated_no_crash():
    pytest.importorskip("torch")        # extension links torch libs
    ext = pytest.importorskip(
        "tfservingcache_amd.engine._tfsc_engine")
    data = _sample_request()
    name, ver, _label = ext.peek_spec(data)
    assert name == "mm" and ver == 3
    for cut in range(len(data)):
        try:
            ext.peek_spec(data[:cut])
        except Exception:       # FastFallback — clean python exception
            pass


# -- single-flight lock lifetime ---------------------------------------------

class _SlowProvider(DiskModelProvider):
    def __init__(self, base_dir, delay=0.15):
        super().__init__(base_dir)
        self.delay = delay
        self.loads = 0
        self._mu = threading.Lock()

    def load_model(self, name, version, dest):
        with self._mu:
            self.loads += 1
        time.sleep(self.delay)
        return super().load_model(name, version, dest)


def test_single_flight_no_duplicate_fetch_under_waiters(tmp_path):
    """Waiters queued on the in-flight lock must not allow a later
    thread to start a second concurrent fetch (advisor medium #3)."""
    repo = tmp_path / "repo"
    write_model_repo(str(repo), [("m", 1, "half_plus_two")])
    cache = LRUCache(str(tmp_path / "cache"), 10 ** 8)
    pool = ModelPool(make_cpu_loader(cache), max_concurrent_models=4)
    prov = _SlowProvider(str(repo))
    cm = CacheManager(prov, cache, pool)

    errs = []

    def worker():
        try:
            cm.ensure_loaded("m", 1)
        except Exception as e:      # noqa: BLE001
            errs.append(e)

    threads = [threading.Thread(target=worker) for _ in range(12)]
    for t in threads:
        t.start()
    for t in threads:
        t.join()
    assert not errs
    assert prov.loads == 1
    assert cm._in_flight == {}


def test_warmup_corrupt_length_field(tmp_path):
    """A warmup TFRecord with an absurd length field must be treated as
    corrupt, not allocated."""
    from tfservingcache_amd.engine.warmup import read_tfrecords
    import struct
    p = tmp_path / "warmup"
    p.write_bytes(struct.pack("<Q", 1 << 40) + b"\0" * 12)
    assert list(read_tfrecords(str(p))) == []
    # a sane record still reads
    rec = b"hello"
    p.write_bytes(struct.pack("<Q", len(rec)) + b"\0" * 4 + rec +
                  b"\0" * 4)
    assert list(read_tfrecords(str(p))) == [rec]


def test_malformed_tensorproto_dims_rejected_cleanly():
    """dims/content mismatches and absurd dims must raise a clean
    ValueError/ServingError from the conversion layer — never allocate
    per attacker-declared dims or crash."""
    from tfservingcache_amd.wire.tensor import tensorproto_to_numpy

    # content 4 bytes, dims claim 100 elements
    tp = m.TensorProto(
        dtype=m.DT_FLOAT,
        tensor_shape=m.TensorShapeProto(
            dim=[m.TensorShapeDim(size=100)]),
        tensor_content=b"\x00\x00\x80?")
    with pytest.raises((ValueError, Exception)):
        arr = tensorproto_to_numpy(tp)
        assert arr.size == 100      # reaching here would be silent junk

    # absurd dims: must not allocate ~8 EB
    tp2 = m.TensorProto(
        dtype=m.DT_FLOAT,
        tensor_shape=m.TensorShapeProto(
            dim=[m.TensorShapeDim(size=1 << 30),
                 m.TensorShapeDim(size=1 << 30)]),
        tensor_content=b"\x00" * 16)
    with pytest.raises(Exception):
        tensorproto_to_numpy(tp2)

    # negative dim
    tp3 = m.TensorProto(
        dtype=m.DT_FLOAT,
        tensor_shape=m.TensorShapeProto(
            dim=[m.TensorShapeDim(size=-4)]),
        tensor_content=b"\x00" * 16)
    with pytest.raises(Exception):
        tensorproto_to_numpy(tp3)


def test_string_tensor_dims_bounded():
    """DT_STRING dims beyond the actual value count must raise before
    allocating (attacker-declared dims -> np.empty DoS)."""
    from tfservingcache_amd.wire.tensor import tensorproto_to_numpy
    tp = m.TensorProto(
        dtype=m.DT_STRING,
        tensor_shape=m.TensorShapeProto(
            dim=[m.TensorShapeDim(size=1 << 30),
                 m.TensorShapeDim(size=1 << 30)]),
        string_val=[b"x"])
    with pytest.raises(ValueError):
        tensorproto_to_numpy(tp)
    # sane string tensors still round-trip
    tp2 = m.TensorProto(
        dtype=m.DT_STRING,
        tensor_shape=m.TensorShapeProto(
            dim=[m.TensorShapeDim(size=2)]),
        string_val=[b"a", b"bb"])
    arr = tensorproto_to_numpy(tp2)
    assert list(arr) == [b"a", b"bb"]


def test_numeric_splat_expansion_bounded():
    """The TF splat rule (repeat last value to fill dims) must not
    expand to attacker-declared exabyte sizes."""
    from tfservingcache_amd.wire.tensor import tensorproto_to_numpy
    tp = m.TensorProto(
        dtype=m.DT_FLOAT,
        tensor_shape=m.TensorShapeProto(
            dim=[m.TensorShapeDim(size=1 << 30),
                 m.TensorShapeDim(size=1 << 30)]),
        float_val=[1.0])
    with pytest.raises(ValueError):
        tensorproto_to_numpy(tp)
    # the legitimate splat still works
    tp2 = m.TensorProto(
        dtype=m.DT_FLOAT,
        tensor_shape=m.TensorShapeProto(
            dim=[m.TensorShapeDim(size=4)]),
        float_val=[7.0])
    np.testing.assert_allclose(tensorproto_to_numpy(tp2),
                               [7.0, 7.0, 7.0, 7.0])
