"""Planner + CPU executor tests over the model builders."""
import numpy as np
import pytest

from tfservingcache_amd.engine.model import load_model_from_dir
from tfservingcache_amd.engine.planner import compile_graph
from tfservingcache_amd.engine.savedmodel import (read_saved_model,
                                                  write_saved_model)
from tfservingcache_amd.models import (build_bert, build_half_plus_two,
                                       build_mlp, build_resnet50)


def _load(tmp_path, sm, name="m", version=1):
    d = tmp_path / name / str(version)
    write_saved_model(sm, str(d))
    return load_model_from_dir(str(d), name, version)


def test_half_plus_two(tmp_path):
    model = _load(tmp_path, build_half_plus_two())
    out = model.predict({"x": np.array([1.0, 2.0, 5.0], dtype=np.float32)})
    np.testing.assert_allclose(out["y"], [2.5, 3.0, 4.5])


def test_mlp_matches_manual(tmp_path):
    model = _load(tmp_path, build_mlp(seed=3))
    # fused plan: gemm(+bias,relu), gemm(+bias), softmax
    kinds = [op.kind for op in model.plan.ops]
    assert kinds == ["gemm", "gemm", "softmax"]
    assert model.plan.ops[0].params["act"] == "relu"
    assert model.plan.ops[0].params["has_bias"]

    x = np.random.default_rng(0).standard_normal((4, 16)).astype(np.float32)
    out = model.predict({"x": x})["probs"]
    # manual reference
    ts = {t.name: t.weight for t in model.plan.tensors if t.weight is not None}
    gd, sigs = read_saved_model(
        str(tmp_path / "m" / "1"))
    consts = {nd.name: nd for nd in gd.node if nd.op == "Const"}
    from tfservingcache_amd.wire.tensor import tensorproto_to_numpy
    w1 = tensorproto_to_numpy(consts["w1"].attr["value"].tensor)
    b1 = tensorproto_to_numpy(consts["b1"].attr["value"].tensor)
    w2 = tensorproto_to_numpy(consts["w2"].attr["value"].tensor)
    b2 = tensorproto_to_numpy(consts["b2"].attr["value"].tensor)
    h = np.maximum(x @ w1 + b1, 0)
    logits = h @ w2 + b2
    e = np.exp(logits - logits.max(-1, keepdims=True))
    ref = e / e.sum(-1, keepdims=True)
    np.testing.assert_allclose(out, ref, rtol=1e-5, atol=1e-6)
    assert out.shape == (4, 8)


def test_resnet50_tiny(tmp_path):
    # small image + reduced stages for CPU test speed
    sm = build_resnet50(image_size=32, num_classes=10,
                        stage_blocks=(1, 1, 1, 1))
    model = _load(tmp_path, sm)
    kinds = [op.kind for op in model.plan.ops]
    # all BN folded: no bn_act ops expected, conv count = 1 stem + 4*(3+1)
    assert kinds.count("conv2d") == 17
    assert "bn_act" not in kinds
    assert kinds.count("pool") == 1
    assert kinds[-1] == "softmax"
    # residual fusion happened
    assert any(op.params.get("residual") for op in model.plan.ops
               if op.kind == "conv2d")
    x = np.random.default_rng(0).standard_normal((2, 32, 32, 3)).astype(
        np.float32) * 0.1
    out = model.predict({"input": x})
    assert out["probs"].shape == (2, 10)
    np.testing.assert_allclose(out["probs"].sum(-1), [1.0, 1.0], rtol=1e-4)
    assert np.all(np.isfinite(out["logits"]))


def test_bert_tiny(tmp_path):
    sm = build_bert(seq_len=8, hidden=32, layers=2, heads=4,
                    intermediate=64, vocab=100)
    model = _load(tmp_path, sm)
    kinds = [op.kind for op in model.plan.ops]
    assert kinds.count("layernorm") == 2 * 2 + 1  # 2/layer + embeddings
    assert kinds.count("batched_gemm") == 2 * 2
    assert kinds.count("softmax") == 2
    # GELU fused into the ffn1 gemm
    gelu_gemms = [op for op in model.plan.ops
                  if op.kind == "gemm" and op.params.get("act") == "gelu"]
    assert len(gelu_gemms) == 2
    ids = np.random.default_rng(1).integers(0, 100, size=(3, 8)).astype(
        np.int32)
    out = model.predict({"input_ids": ids})
    assert out["sequence_output"].shape == (3, 8, 32)
    assert out["pooled_output"].shape == (3, 32)
    assert np.all(np.isfinite(out["sequence_output"]))
    assert np.abs(out["pooled_output"]).max() <= 1.0


def test_bert_layernorm_numerics(tmp_path):
    """LayerNorm fused op must match the unfused primitive math."""
    sm = build_bert(seq_len=4, hidden=16, layers=1, heads=2,
                    intermediate=32, vocab=50, seed=7)
    model = _load(tmp_path, sm)
    ids = np.arange(8).reshape(2, 4).astype(np.int32) % 50
    out = model.predict({"input_ids": ids})["sequence_output"]
    # last-layer output rows should be ~zero-mean/unit-var (gamma=1,beta=0)
    mean = out.mean(-1)
    var = out.var(-1)
    np.testing.assert_allclose(mean, np.zeros_like(mean), atol=1e-5)
    np.testing.assert_allclose(var, np.ones_like(var), rtol=1e-3)


def test_attention_fusion(tmp_path):
    """head_dim==64 attention collapses to one fused op; numerics match
    the unfused plan."""
    from tfservingcache_amd.engine.planner import _Lowerer
    from tfservingcache_amd.engine.savedmodel import read_saved_model
    from tfservingcache_amd.engine.executor_cpu import CpuExecutor
    from tfservingcache_amd.engine.model import LoadedModel

    sm = build_bert(seq_len=16, hidden=128, layers=1, heads=2,
                    intermediate=64, vocab=60, seed=13)
    d = tmp_path / "bf" / "1"
    write_saved_model(sm, str(d))
    model = load_model_from_dir(str(d), "bf", 1)   # fused (default path)
    kinds = [op.kind for op in model.plan.ops]
    assert kinds.count("attention") == 1
    assert "softmax" not in kinds
    assert "batched_gemm" not in kinds
    assert "transpose" not in kinds

    gd, sigs = read_saved_model(str(d))
    unfused_plan = _Lowerer(gd, next(iter(sigs.values()))).run()
    unfused = LoadedModel("bf", 1, unfused_plan)

    ids = np.random.default_rng(3).integers(0, 60, (2, 16)).astype(np.int32)
    a = model.predict({"input_ids": ids})
    b = unfused.predict({"input_ids": ids})
    np.testing.assert_allclose(a["sequence_output"], b["sequence_output"],
                               rtol=1e-5, atol=1e-6)
    np.testing.assert_allclose(a["pooled_output"], b["pooled_output"],
                               rtol=1e-5, atol=1e-6)


def test_nchw_graphs_rejected_loudly(tmp_path):
    """NCHW exports must raise PlanError, never silently compute NHWC
    math (bias/BN/pool would broadcast along the wrong axis)."""
    import numpy as np
    from tfservingcache_amd.engine.planner import PlanError
    from tfservingcache_amd.engine.savedmodel import GraphBuilder

    def build(op):
        gb = GraphBuilder()
        f32 = gb.a_type(1)
        x = gb.placeholder("x", np.float32, [-1, 4, 6, 6],
                           signature_name="x")
        if op == "BiasAdd":
            nd = gb.node("BiasAdd", "y",
                         [x, gb.const("b", np.zeros(4, np.float32))],
                         T=f32, data_format=gb.a_str("NCHW"))
        elif op == "MaxPool":
            nd = gb.node("MaxPool", "y", [x], T=f32,
                         ksize=gb.a_ints([1, 1, 2, 2]),
                         strides=gb.a_ints([1, 1, 2, 2]),
                         padding=gb.a_str("VALID"),
                         data_format=gb.a_str("NCHW"))
        else:
            c = np.ones(4, np.float32)
            nd = gb.node("FusedBatchNormV3", "y",
                         [x, gb.const("s", c), gb.const("o", c * 0),
                          gb.const("mn", c * 0), gb.const("vr", c)],
                         T=f32, U=f32, epsilon=gb.a_float(1e-3),
                         is_training=gb.a_bool(False),
                         data_format=gb.a_str("NCHW"))
        gb.mark_output("y", nd)
        return gb.build()

    for op in ("BiasAdd", "MaxPool", "FusedBatchNormV3"):
        with pytest.raises(PlanError):
            _load(tmp_path, build(op), name=f"nchw_{op.lower()}")


def test_unsupported_attrs_rejected_loudly(tmp_path):
    """Attributes whose silent omission would change numerics must
    raise PlanError: dilated convs, non-axis-0 Gather, PadV2 with a
    non-zero constant."""
    import numpy as np
    from tfservingcache_amd.engine.planner import PlanError
    from tfservingcache_amd.engine.savedmodel import GraphBuilder

    def dilated_conv():
        gb = GraphBuilder()
        f32 = gb.a_type(1)
        x = gb.placeholder("x", np.float32, [-1, 8, 8, 3],
                           signature_name="x")
        w = gb.const("w", np.zeros((3, 3, 3, 4), np.float32))
        nd = gb.node("Conv2D", "y", [x, w], T=f32,
                     strides=gb.a_ints([1, 1, 1, 1]),
                     dilations=gb.a_ints([1, 2, 2, 1]),
                     padding=gb.a_str("SAME"),
                     data_format=gb.a_str("NHWC"))
        gb.mark_output("y", nd)
        return gb.build()

    def gather_axis1():
        gb = GraphBuilder()
        f32 = gb.a_type(1)
        t = gb.const("t", np.zeros((5, 7), np.float32))
        idx = gb.placeholder("i", np.int32, [-1], signature_name="i")
        nd = gb.node("GatherV2", "y",
                     [t, idx, gb.const("ax", np.int32(1))],
                     Tparams=f32, Tindices=gb.a_type(3),
                     Taxis=gb.a_type(3))
        gb.mark_output("y", nd)
        return gb.build()

    def padv2_nonzero():
        gb = GraphBuilder()
        f32 = gb.a_type(1)
        x = gb.placeholder("x", np.float32, [-1, 4], signature_name="x")
        nd = gb.node("PadV2", "y",
                     [x, gb.const("p", np.array([[0, 0], [1, 1]],
                                                np.int32)),
                      gb.const("c", np.float32(-1.0))], T=f32)
        gb.mark_output("y", nd)
        return gb.build()

    for i, build in enumerate((dilated_conv, gather_axis1,
                               padv2_nonzero)):
        with pytest.raises(PlanError):
            _load(tmp_path, build(), name=f"bad_{i}")
