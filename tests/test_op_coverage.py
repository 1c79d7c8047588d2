"""Round-2 op-coverage widening: DepthwiseConv2dNative
(MobileNetV2 family), Einsum (transformer exports), Relu6, and
DT_STRING handling. CPU numerics compare the plan execution against
independent naive numpy computations (not the executor's own code
paths)."""
import numpy as np
import pytest

from tfservingcache_amd.engine.savedmodel import (GraphBuilder,
                                                  write_saved_model)
from tfservingcache_amd.engine.model import load_model_from_dir
from tfservingcache_amd.models.builders import build_mobilenet_v2


def _load(tmp_path, sm, name="m"):
    d = str(tmp_path / name / "1")
    write_saved_model(sm, d)
    return load_model_from_dir(d, name, 1)


# -- depthwise ---------------------------------------------------------------

def _naive_depthwise(x, w, stride, padding):
    """Independent reference: direct loops (no im2col, no executor code)."""
    N, H, W_, C = x.shape
    R, S, C2, M = w.shape
    assert C2 == C and M == 1
    if padding == "SAME":
        Ho = -(-H // stride)
        Wo = -(-W_ // stride)
        ph = max((Ho - 1) * stride + R - H, 0)
        pw = max((Wo - 1) * stride + S - W_, 0)
        pt, pl = ph // 2, pw // 2
    else:
        Ho, Wo = (H - R) // stride + 1, (W_ - S) // stride + 1
        pt = pl = 0
    y = np.zeros((N, Ho, Wo, C), dtype=np.float64)
    for n in range(N):
        for ho in range(Ho):
            for wo in range(Wo):
                for r in range(R):
                    for s in range(S):
                        hi = ho * stride - pt + r
                        wi = wo * stride - pl + s
                        if 0 <= hi < H and 0 <= wi < W_:
                            y[n, ho, wo] += x[n, hi, wi] * w[r, s, :, 0]
    return y.astype(np.float32)


@pytest.mark.parametrize("stride,padding", [(1, "SAME"), (2, "SAME"),
                                            (1, "VALID")])
def test_depthwise_numerics_vs_naive(tmp_path, stride, padding):
    rng = np.random.default_rng(3)
    C = 8
    w = (rng.standard_normal((3, 3, C, 1)) * 0.3).astype(np.float32)
    gb = GraphBuilder()
    f32 = gb.a_type(1)
    x_ph = gb.placeholder("input", np.float32, [-1, 10, 10, C],
                          signature_name="input")
    wc = gb.const("w", w)
    d = gb.node("DepthwiseConv2dNative", "dw", [x_ph, wc], T=f32,
                strides=gb.a_ints([1, stride, stride, 1]),
                padding=gb.a_str(padding),
                data_format=gb.a_str("NHWC"))
    gb.mark_output("y", d)
    model = _load(tmp_path, gb.build())
    assert [op.kind for op in model.plan.ops] == ["depthwise_conv"]
    x = (rng.standard_normal((2, 10, 10, C)) * 0.5).astype(np.float32)
    got = model.predict({"input": x})["y"]
    want = _naive_depthwise(x, w, stride, padding)
    np.testing.assert_allclose(got, want, rtol=1e-4, atol=1e-5)


def test_depthwise_bn_relu6_folds(tmp_path):
    rng = np.random.default_rng(5)
    C = 8
    gb = GraphBuilder()
    f32 = gb.a_type(1)
    x_ph = gb.placeholder("input", np.float32, [-1, 6, 6, C],
                          signature_name="input")
    w = (rng.standard_normal((3, 3, C, 1)) * 0.5).astype(np.float32)
    wc = gb.const("w", w)
    d = gb.node("DepthwiseConv2dNative", "dw", [x_ph, wc], T=f32,
                strides=gb.a_ints([1, 1, 1, 1]), padding=gb.a_str("SAME"),
                data_format=gb.a_str("NHWC"))
    scale = np.abs(rng.standard_normal(C)).astype(np.float32) + 0.5
    offset = (rng.standard_normal(C) * 0.1).astype(np.float32)
    mean = (rng.standard_normal(C) * 0.1).astype(np.float32)
    var = np.abs(rng.standard_normal(C)).astype(np.float32) + 0.5
    bn = gb.node("FusedBatchNormV3", "bn",
                 [d, gb.const("g", scale), gb.const("b", offset),
                  gb.const("m", mean), gb.const("v", var)],
                 T=f32, U=f32, epsilon=gb.a_float(1e-3),
                 is_training=gb.a_bool(False),
                 data_format=gb.a_str("NHWC"))
    r6 = gb.node("Relu6", "r6", [bn], T=f32)
    gb.mark_output("y", r6)
    model = _load(tmp_path, gb.build())
    # BN + Relu6 folded into the depthwise op
    kinds = [op.kind for op in model.plan.ops]
    assert kinds == ["depthwise_conv"]
    assert model.plan.ops[0].params["act"] == "relu6"
    x = (rng.standard_normal((2, 6, 6, C)) * 2.0).astype(np.float32)
    got = model.predict({"input": x})["y"]
    raw = _naive_depthwise(x, w, 1, "SAME")
    want = np.clip((raw - mean) / np.sqrt(var + 1e-3) * scale + offset,
                   0.0, 6.0)
    np.testing.assert_allclose(got, want, rtol=1e-4, atol=1e-5)
    assert got.max() <= 6.0


def test_mobilenet_v2_serves_cpu(tmp_path):
    sm = build_mobilenet_v2(image_size=32, num_classes=10)
    model = _load(tmp_path, sm)
    kinds = [op.kind for op in model.plan.ops]
    assert kinds.count("depthwise_conv") == 17
    assert "bn_act" not in kinds          # all BN folded
    x = (np.random.default_rng(0).standard_normal((2, 32, 32, 3))
         * 0.5).astype(np.float32)
    out = model.predict({"input": x})
    assert out["probs"].shape == (2, 10)
    np.testing.assert_allclose(out["probs"].sum(-1), [1, 1], rtol=1e-4)


# -- einsum ------------------------------------------------------------------

@pytest.mark.parametrize("eq,ashape,wshape", [
    ("ij,jk->ik", (4, 6), (6, 5)),
    ("abc,cd->abd", (-1, 3, 8), (8, 5)),
    ("abc,cde->abde", (-1, 3, 8), (8, 2, 5)),
    ("abcd,cde->abe", (-1, 3, 4, 6), (4, 6, 7)),
])
def test_einsum_dense_family(tmp_path, eq, ashape, wshape):
    rng = np.random.default_rng(7)
    gb = GraphBuilder()
    f32 = gb.a_type(1)
    x_ph = gb.placeholder("x", np.float32, list(ashape),
                          signature_name="x")
    w = (rng.standard_normal(wshape) * 0.3).astype(np.float32)
    wc = gb.const("w", w)
    e = gb.node("Einsum", "ein", [x_ph, wc], T=f32,
                equation=gb.a_str(eq), N=gb.a_ints([2]))
    gb.mark_output("y", e)
    model = _load(tmp_path, gb.build())
    assert [op.kind for op in model.plan.ops] == ["gemm"]
    concrete = tuple(2 if d == -1 else d for d in ashape)
    x = (rng.standard_normal(concrete) * 0.5).astype(np.float32)
    got = model.predict({"x": x})["y"]
    want = np.einsum(eq, x, w)
    assert got.shape == want.shape
    np.testing.assert_allclose(got, want, rtol=1e-4, atol=1e-5)


def test_einsum_with_bias_fuses(tmp_path):
    rng = np.random.default_rng(9)
    gb = GraphBuilder()
    f32 = gb.a_type(1)
    x_ph = gb.placeholder("x", np.float32, [-1, 4, 8], signature_name="x")
    w = (rng.standard_normal((8, 5)) * 0.3).astype(np.float32)
    b = (rng.standard_normal(5) * 0.1).astype(np.float32)
    e = gb.node("Einsum", "ein", [x_ph, gb.const("w", w)], T=f32,
                equation=gb.a_str("abc,cd->abd"))
    ba = gb.node("BiasAdd", "ba", [e, gb.const("b", b)], T=f32)
    r = gb.node("Relu", "r", [ba], T=f32)
    gb.mark_output("y", r)
    model = _load(tmp_path, gb.build())
    ops = model.plan.ops
    assert [op.kind for op in ops] == ["gemm"]
    assert ops[0].params.get("has_bias") and ops[0].params["act"] == "relu"
    x = (rng.standard_normal((2, 4, 8)) * 0.5).astype(np.float32)
    got = model.predict({"x": x})["y"]
    want = np.maximum(np.einsum("abc,cd->abd", x, w) + b, 0.0)
    np.testing.assert_allclose(got, want, rtol=1e-4, atol=1e-5)


def test_einsum_unsupported_is_loud(tmp_path):
    from tfservingcache_amd.engine.planner import PlanError
    gb = GraphBuilder()
    f32 = gb.a_type(1)
    x_ph = gb.placeholder("x", np.float32, [-1, 4, 8], signature_name="x")
    w = np.zeros((4, 8), dtype=np.float32)
    e = gb.node("Einsum", "ein", [x_ph, gb.const("w", w)], T=f32,
                equation=gb.a_str("abc,bc->ab"))   # K not trailing-only
    gb.mark_output("y", e)
    with pytest.raises(PlanError):
        _load(tmp_path, gb.build())


# -- relu6 standalone --------------------------------------------------------

def test_relu6_eltwise(tmp_path):
    gb = GraphBuilder()
    f32 = gb.a_type(1)
    x_ph = gb.placeholder("x", np.float32, [-1, 8], signature_name="x")
    sq = gb.node("Square", "sq", [x_ph], T=f32)   # block act fusion
    r6 = gb.node("Relu6", "r6", [sq], T=f32)
    gb.mark_output("y", r6)
    model = _load(tmp_path, gb.build())
    x = np.linspace(-4, 4, 16, dtype=np.float32).reshape(2, 8)
    got = model.predict({"x": x})["y"]
    np.testing.assert_allclose(got, np.clip(x * x, 0, 6), rtol=1e-5)


# -- DT_STRING ---------------------------------------------------------------

def test_string_tensorproto_round_trip():
    from tfservingcache_amd.wire import messages as m
    from tfservingcache_amd.wire.tensor import (numpy_to_tensorproto,
                                                tensorproto_to_numpy)
    arr = np.empty((2, 2), dtype=object)
    arr[0, 0], arr[0, 1] = b"hello", b"world"
    arr[1, 0], arr[1, 1] = b"", bytes(range(256))
    tp = numpy_to_tensorproto(arr)
    assert tp.dtype == m.DT_STRING
    enc = tp.encode()
    got = tensorproto_to_numpy(m.TensorProto.decode(enc))
    assert got.shape == (2, 2)
    assert got[1, 1] == bytes(range(256))


def test_string_predict_clean_error(tmp_path):
    from tfservingcache_amd.cachemanager import (CacheManager, LRUCache,
                                                 ModelPool,
                                                 make_cpu_loader)
    from tfservingcache_amd.cachemanager.providers import DiskModelProvider
    from tfservingcache_amd.models import write_model_repo
    from tfservingcache_amd.tfservingproxy import LocalServingHandler
    from tfservingcache_amd.tfservingproxy.servinghandler import \
        ServingError
    from tfservingcache_amd.wire import messages as m
    from tfservingcache_amd.wire.tensor import numpy_to_tensorproto

    write_model_repo(str(tmp_path / "repo"), [("mlp", 1, "mlp")])
    cache = LRUCache(str(tmp_path / "cache"), 10 ** 8)
    pool = ModelPool(make_cpu_loader(cache), max_concurrent_models=2)
    cm = CacheManager(DiskModelProvider(str(tmp_path / "repo")), cache,
                      pool)
    handler = LocalServingHandler(cm)
    s = np.empty((1,), dtype=object)
    s[0] = b"some-bytes"
    req = m.PredictRequest(
        model_spec=m.ModelSpec(name="mlp", version=m.Int64Value(value=1)),
        inputs={"x": numpy_to_tensorproto(s)})
    with pytest.raises(ServingError) as ei:
        handler.predict(req)
    assert ei.value.code == m.ERROR_INVALID_ARGUMENT
    assert "DT_STRING" in str(ei.value)


def test_rest_json_string_inputs_decode():
    from tfservingcache_amd.tfservingproxy.json_codec import \
        parse_predict_body
    inputs, fmt, _sig = parse_predict_body(
        {"instances": [{"text": "abc"}, {"text": {"b64": "aGk="}}]})
    arr = inputs["text"]
    assert arr.dtype == object
    assert arr[0] == b"abc" and arr[1] == b"hi"


# -- StridedSlice / Slice / Cast / ArgMax ------------------------------------

def _ss_graph(ashape, begin, end, strides, bm=0, em=0, sm=0):
    gb = GraphBuilder()
    f32 = gb.a_type(1)
    x_ph = gb.placeholder("x", np.float32, list(ashape),
                          signature_name="x")
    ss = gb.node("StridedSlice", "ss",
                 [x_ph,
                  gb.const("b", np.array(begin, np.int32)),
                  gb.const("e", np.array(end, np.int32)),
                  gb.const("s", np.array(strides, np.int32))],
                 T=f32, Index=gb.a_type(3),
                 begin_mask=gb.a_int(bm), end_mask=gb.a_int(em),
                 shrink_axis_mask=gb.a_int(sm),
                 ellipsis_mask=gb.a_int(0), new_axis_mask=gb.a_int(0))
    gb.mark_output("y", ss)
    return gb.build()


@pytest.mark.parametrize("begin,end,strides,bm,em,sm,ref", [
    # x[:, 0, :] — the CLS-token extraction of real BERT exports
    ([0, 0, 0], [0, 1, 0], [1, 1, 1], 0b101, 0b101, 0b010,
     lambda x: x[:, 0, :]),
    # x[:, 1:5, :]
    ([0, 1, 0], [0, 5, 0], [1, 1, 1], 0b101, 0b101, 0,
     lambda x: x[:, 1:5, :]),
    # x[:, ::2, 1:]  (begin_mask must NOT cover the begin=1 dim)
    ([0, 0, 1], [0, 0, 0], [1, 2, 1], 0b011, 0b111, 0,
     lambda x: x[:, ::2, 1:]),
    # negative begin: x[:, -3:, :]
    ([0, -3, 0], [0, 0, 0], [1, 1, 1], 0b101, 0b111, 0,
     lambda x: x[:, -3:, :]),
])
def test_strided_slice_family(tmp_path, begin, end, strides, bm, em, sm,
                              ref):
    sm_ = _ss_graph([-1, 7, 6], begin, end, strides, bm, em, sm)
    model = _load(tmp_path, sm_)
    assert [op.kind for op in model.plan.ops] == ["strided_copy"]
    x = np.random.default_rng(0).standard_normal((3, 7, 6)).astype(
        np.float32)
    got = model.predict({"x": x})["y"]
    want = ref(x)
    assert got.shape == want.shape, (got.shape, want.shape)
    np.testing.assert_allclose(got, want, rtol=1e-6)


def test_slice_op(tmp_path):
    gb = GraphBuilder()
    f32 = gb.a_type(1)
    x_ph = gb.placeholder("x", np.float32, [-1, 8, 5],
                          signature_name="x")
    sl = gb.node("Slice", "sl",
                 [x_ph,
                  gb.const("b", np.array([0, 2, 1], np.int32)),
                  gb.const("sz", np.array([-1, 4, 3], np.int32))],
                 T=f32, Index=gb.a_type(3))
    gb.mark_output("y", sl)
    model = _load(tmp_path, gb.build())
    x = np.random.default_rng(1).standard_normal((2, 8, 5)).astype(
        np.float32)
    got = model.predict({"x": x})["y"]
    np.testing.assert_allclose(got, x[:, 2:6, 1:4], rtol=1e-6)


def test_cast_and_argmax(tmp_path):
    gb = GraphBuilder()
    f32 = gb.a_type(1)
    i32 = gb.a_type(3)
    ids = gb.placeholder("ids", np.int32, [-1, 6], signature_name="ids")
    casted = gb.node("Cast", "c", [ids], SrcT=i32, DstT=f32)
    sq = gb.node("Square", "sq", [casted], T=f32)
    am = gb.node("ArgMax", "am",
                 [sq, gb.const("ax", np.int32(-1))],
                 T=f32, output_type=i32)
    gb.mark_output("idx", am)
    gb.mark_output("sq", sq)
    model = _load(tmp_path, gb.build())
    x = np.array([[1, -5, 3, 0, 2, -1], [0, 0, 7, -7, 1, 2]], np.int32)
    out = model.predict({"ids": x})
    np.testing.assert_allclose(out["sq"], (x.astype(np.float32)) ** 2)
    # |-5| and |±7|: argmax of squares; -7 comes after 7 -> first wins
    np.testing.assert_array_equal(out["idx"], [1, 2])
    assert out["idx"].dtype == np.int32


# -- Pack / Unpack / Concat / LeakyRelu / Pow --------------------------------

def test_pack_unpack_concat(tmp_path):
    gb = GraphBuilder()
    f32 = gb.a_type(1)
    a = gb.placeholder("a", np.float32, [-1, 4], signature_name="a")
    b = gb.placeholder("b", np.float32, [-1, 4], signature_name="b")
    pk = gb.node("Pack", "pk", [a, b], N=gb.a_int(2), T=f32,
                 axis=gb.a_int(1))                       # [B, 2, 4]
    up = gb.node("Unpack", "up", [pk], num=gb.a_int(2), T=f32,
                 axis=gb.a_int(1))
    cc = gb.node("ConcatV2", "cc",
                 ["up:0", "up:1", gb.const("ax", np.int32(1))],
                 N=gb.a_int(2), T=f32)                   # [B, 8]
    gb.mark_output("packed", pk)
    gb.mark_output("cat", cc)
    model = _load(tmp_path, gb.build())
    x = np.arange(8, dtype=np.float32).reshape(2, 4)
    y = -x
    out = model.predict({"a": x, "b": y})
    np.testing.assert_allclose(out["packed"],
                               np.stack([x, y], axis=1))
    np.testing.assert_allclose(out["cat"],
                               np.concatenate([x, y], axis=1))


def test_leaky_relu_and_pow(tmp_path):
    gb = GraphBuilder()
    f32 = gb.a_type(1)
    x_ph = gb.placeholder("x", np.float32, [-1, 6], signature_name="x")
    lr = gb.node("LeakyRelu", "lr", [x_ph], T=f32,
                 alpha=gb.a_float(0.1))
    p2 = gb.node("Pow", "p2", [x_ph, gb.const("e2", np.float32(2.0))],
                 T=f32)
    gb.mark_output("lr", lr)
    gb.mark_output("p2", p2)
    model = _load(tmp_path, gb.build())
    x = np.linspace(-3, 3, 12, dtype=np.float32).reshape(2, 6)
    out = model.predict({"x": x})
    np.testing.assert_allclose(out["lr"], np.where(x > 0, x, 0.1 * x),
                               rtol=1e-6)
    np.testing.assert_allclose(out["p2"], x * x, rtol=1e-6)


@pytest.mark.parametrize("mode,stride", [("avg", 1), ("avg", 2),
                                         ("max", 2)])
def test_pool_same_padding_tf_semantics(tmp_path, mode, stride):
    """SAME-padded pooling: TF's AvgPool divides by the VALID cell
    count only (excludes padding) — torch.nn.AvgPool2d with
    count_include_pad=False emulates that; MaxPool is unaffected.
    The GPU kernels already count valid cells (k_pool); this pins the
    CPU reference to the same semantics."""
    torch = pytest.importorskip("torch")
    rng = np.random.default_rng(7)
    H = W = 7
    C = 4
    k = 3
    gb = GraphBuilder()
    f32 = gb.a_type(1)
    x_ph = gb.placeholder("x", np.float32, [-1, H, W, C],
                          signature_name="x")
    op = "AvgPool" if mode == "avg" else "MaxPool"
    pl = gb.node(op, "pl", [x_ph], T=f32,
                 ksize=gb.a_ints([1, k, k, 1]),
                 strides=gb.a_ints([1, stride, stride, 1]),
                 padding=gb.a_str("SAME"),
                 data_format=gb.a_str("NHWC"))
    gb.mark_output("y", pl)
    model = _load(tmp_path, gb.build())

    x = rng.standard_normal((2, H, W, C)).astype(np.float32)
    got = model.predict({"x": x})["y"]

    xt = torch.from_numpy(x).permute(0, 3, 1, 2)    # NCHW
    # TF SAME padding for H=7,k=3: pad 1 on each side (stride 1) or
    # asymmetric (stride 2: Ho=4, pad_h=2 -> 1+1)
    Ho = -(-H // stride)
    ph = max((Ho - 1) * stride + k - H, 0)
    pt_, pb_ = ph // 2, ph - ph // 2
    if mode == "avg":
        # emulate asymmetric TF padding with symmetric torch pad when
        # equal; here ph is even for these cases -> pt_ == pb_
        assert pt_ == pb_
        pool = torch.nn.AvgPool2d(k, stride=stride, padding=pt_,
                                  count_include_pad=False)
        want = pool(xt)
    else:
        pool = torch.nn.MaxPool2d(k, stride=stride, padding=pt_)
        want = pool(xt)
    want = want.permute(0, 2, 3, 1).numpy()
    assert got.shape == want.shape, (got.shape, want.shape)
    np.testing.assert_allclose(got, want, rtol=1e-5, atol=1e-6)
