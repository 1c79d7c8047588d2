"""tensor_bundle (non-frozen SavedModel) support: SSTable round-trip,
VariableV2/ReadVariableOp lowering, save-subgraph pruning, and serving
through the cache tier."""
import os

import numpy as np
import pytest

from tfservingcache_amd.engine import tensor_bundle as tb
from tfservingcache_amd.engine.model import load_model_from_dir
from tfservingcache_amd.engine.planner import compile_graph
from tfservingcache_amd.engine.savedmodel import GraphBuilder
from tfservingcache_amd.models import write_model_repo


def test_bundle_round_trip(tmp_path):
    prefix = str(tmp_path / "variables" / "variables")
    rng = np.random.default_rng(0)
    tensors = {
        "layer/kernel": rng.standard_normal((64, 32)).astype(np.float32),
        "layer/bias": rng.standard_normal(32).astype(np.float32),
        "global_step": np.array(123, np.int64),
        "flag": np.array(True),
        # >16 entries exercises prefix compression + restart points
        **{f"block_{i:02d}/w": rng.standard_normal((3, 3)).astype(
            np.float32) for i in range(20)},
    }
    tb.write_bundle(prefix, tensors)
    assert os.path.exists(prefix + ".index")
    assert os.path.exists(prefix + ".data-00000-of-00001")
    out = tb.read_bundle(prefix)
    assert set(out) == set(tensors)
    for k, v in tensors.items():
        np.testing.assert_array_equal(out[k], v)
        assert out[k].dtype == v.dtype


def test_bundle_crc_and_magic(tmp_path):
    prefix = str(tmp_path / "v" / "variables")
    tb.write_bundle(prefix, {"t": np.arange(6, dtype=np.float32)})
    raw = open(prefix + ".index", "rb").read()
    import struct
    assert struct.unpack("<Q", raw[-8:])[0] == tb.MAGIC
    # corrupt the magic -> loud failure
    with open(prefix + ".index", "wb") as f:
        f.write(raw[:-8] + b"\x00" * 8)
    with pytest.raises(tb.BundleError):
        tb.read_bundle(prefix)


def test_vars_model_matches_frozen(tmp_path):
    """mlp_vars (VariableV2 + bundle + save subgraph) must serve
    identically to the frozen mlp with the same seed."""
    write_model_repo(str(tmp_path), [("mv", 1, "mlp_vars"),
                                     ("mf", 1, "mlp")])
    lm_v = load_model_from_dir(str(tmp_path / "mv" / "1"), "mv", 1)
    lm_f = load_model_from_dir(str(tmp_path / "mf" / "1"), "mf", 1)
    x = np.random.default_rng(3).standard_normal((4, 16)).astype(
        np.float32)
    np.testing.assert_allclose(lm_v.predict({"x": x})["probs"],
                               lm_f.predict({"x": x})["probs"],
                               rtol=1e-5)
    # the Assign save-subgraph nodes were pruned, not lowered
    assert all(t.name.rsplit(":", 1)[0].find("Assign") < 0
               for t in lm_v.plan.tensors if t.name)


def test_read_variable_op_lowering():
    """TF2-style VarHandleOp + ReadVariableOp resolves from the bundle
    dict by handle-node name."""
    gb = GraphBuilder()
    f32 = gb.a_type(1)
    x = gb.placeholder("x", np.float32, [-1, 4], signature_name="x")
    gb.node("VarHandleOp", "w", [], dtype=f32,
            shape=gb.a_shape((4, 2)))
    r = gb.node("ReadVariableOp", "w/Read/ReadVariableOp", ["w:0"],
                dtype=f32)
    y = gb.node("MatMul", "y", [x, r], T=f32)
    gb.mark_output("y", y)
    sm = gb.build()
    w = np.random.default_rng(1).standard_normal((4, 2)).astype(
        np.float32)
    mg = sm.meta_graphs[0]
    sig = next(iter(mg.signature_def.values()))
    plan = compile_graph(mg.graph_def, sig, {"w": w})
    from tfservingcache_amd.engine.executor_cpu import CpuExecutor
    x_val = np.random.default_rng(2).standard_normal((3, 4)).astype(
        np.float32)
    xin = plan.sig_inputs["x"]
    yout = plan.sig_outputs["y"]
    out = CpuExecutor(plan).run({xin: x_val}, batch=3,
                                fetch=[yout])[yout]
    np.testing.assert_allclose(out, x_val @ w, rtol=1e-5)


def test_vars_model_through_cache_tier(tmp_path):
    from tfservingcache_amd.cachemanager import (CacheManager, LRUCache,
                                                 ModelPool,
                                                 make_cpu_loader)
    from tfservingcache_amd.cachemanager.providers import \
        DiskModelProvider
    repo = tmp_path / "repo"
    write_model_repo(str(repo), [("mv", 1, "mlp_vars")])
    provider = DiskModelProvider(str(repo))
    cache = LRUCache(str(tmp_path / "cache"), 10 ** 8)
    pool = ModelPool(make_cpu_loader(cache), max_concurrent_models=2)
    cm = CacheManager(provider, cache, pool)
    lm = cm.ensure_loaded("mv", 1)
    x = np.random.default_rng(5).standard_normal((2, 16)).astype(
        np.float32)
    out = lm.predict({"x": x})["probs"]
    np.testing.assert_allclose(out.sum(-1), np.ones(2), rtol=1e-5)
