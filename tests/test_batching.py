"""Dynamic batching tests (CPU engine)."""
import threading
import time

import numpy as np

from tfservingcache_amd.engine.batching import DynamicBatcher
from tfservingcache_amd.engine.model import load_model_from_dir
from tfservingcache_amd.engine.savedmodel import write_saved_model
from tfservingcache_amd.models import build_mlp


def test_batcher_merges_concurrent_requests():
    calls = []

    def run(inputs, filt):
        calls.append({k: v.shape for k, v in inputs.items()})
        x = inputs["x"]
        if hasattr(x, "materialize"):   # merged requests arrive segmented
            x = x.materialize()
        return {"y": x * 2.0}

    b = DynamicBatcher(run, {"x": 0}, max_batch=64, timeout_s=0.05)
    results = {}
    barrier = threading.Barrier(8)

    def worker(i):
        barrier.wait()
        x = np.full((2, 4), float(i), dtype=np.float32)
        results[i] = b.predict({"x": x})

    threads = [threading.Thread(target=worker, args=(i,)) for i in range(8)]
    for t in threads:
        t.start()
    for t in threads:
        t.join()
    # all callers got THEIR rows back
    for i in range(8):
        np.testing.assert_array_equal(results[i]["y"],
                                      np.full((2, 4), 2.0 * i))
    # merging happened: fewer run() calls than requests
    assert len(calls) < 8
    assert sum(s["x"][0] for s in calls) == 16


def test_batcher_overflow_spills_to_second_batch():
    sizes = []

    def run(inputs, filt):
        x = inputs["x"]
        sizes.append(x.shape[0])
        if hasattr(x, "materialize"):
            x = x.materialize()
        return {"y": x + 1.0}

    b = DynamicBatcher(run, {"x": 0}, max_batch=8, timeout_s=0.05)
    results = {}
    barrier = threading.Barrier(6)

    def worker(i):
        barrier.wait()
        results[i] = b.predict({"x": np.full((3,), float(i),
                                             dtype=np.float32)})

    threads = [threading.Thread(target=worker, args=(i,)) for i in range(6)]
    for t in threads:
        t.start()
    for t in threads:
        t.join()
    for i in range(6):
        np.testing.assert_array_equal(results[i]["y"], np.full((3,), i + 1.0))
    assert sum(sizes) == 18
    assert all(s <= 8 for s in sizes)


def test_batcher_error_propagates():
    def run(inputs, filt):
        raise ValueError("boom")

    b = DynamicBatcher(run, {"x": 0}, max_batch=8, timeout_s=0.01)
    errs = []

    def worker():
        try:
            b.predict({"x": np.zeros((1,), dtype=np.float32)})
        except ValueError as e:
            errs.append(e)

    threads = [threading.Thread(target=worker) for _ in range(4)]
    for t in threads:
        t.start()
    for t in threads:
        t.join()
    assert len(errs) == 4


def test_model_level_batching(tmp_path):
    d = tmp_path / "m" / "1"
    write_saved_model(build_mlp(seed=2), str(d))
    model = load_model_from_dir(str(d), "m", 1)
    ref = model.predict({"x": np.ones((2, 16), dtype=np.float32)})
    model.enable_batching(max_batch=16, timeout_s=0.02)
    outs = {}

    def worker(i):
        outs[i] = model.predict(
            {"x": np.ones((2, 16), dtype=np.float32) * (0.1 * i)})

    threads = [threading.Thread(target=worker, args=(i,)) for i in range(5)]
    for t in threads:
        t.start()
    for t in threads:
        t.join()
    assert all(outs[i]["probs"].shape == (2, 8) for i in range(5))
    np.testing.assert_allclose(
        model.predict({"x": np.ones((2, 16), dtype=np.float32)})["probs"],
        ref["probs"], rtol=1e-6)
