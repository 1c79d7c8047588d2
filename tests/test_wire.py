"""Cross-validate the hand-rolled wire codec against python-protobuf.

Builds the same message schemas dynamically with google.protobuf
(descriptor_pool + message_factory) and checks both directions:
our encode -> their decode, their encode -> our decode.
"""
import numpy as np
import pytest

from tfservingcache_amd.wire import messages as m
from tfservingcache_amd.wire import tensor as wt


# ---------------------------------------------------------------------------
# Dynamic reference schema via google.protobuf
# ---------------------------------------------------------------------------
@pytest.fixture(scope="module")
def ref():
    from google.protobuf import descriptor_pb2, descriptor_pool, message_factory

    pool = descriptor_pool.DescriptorPool()
    f = descriptor_pb2.FileDescriptorProto()
    f.name = "tfsc_ref.proto"
    f.package = "tensorflow.serving"
    f.syntax = "proto3"

    T = descriptor_pb2.FieldDescriptorProto

    def msg(name):
        return f.message_type.add(name=name)

    def field(msg_, name, number, ftype, label=T.LABEL_OPTIONAL, type_name=None):
        fd = msg_.field.add(name=name, number=number, type=ftype, label=label)
        if type_name:
            fd.type_name = type_name
        return fd

    iv = msg("Int64Value")
    field(iv, "value", 1, T.TYPE_INT64)

    dim = msg("Dim")
    field(dim, "size", 1, T.TYPE_INT64)
    field(dim, "name", 2, T.TYPE_STRING)

    shape = msg("TensorShapeProto")
    field(shape, "dim", 2, T.TYPE_MESSAGE, T.LABEL_REPEATED,
          ".tensorflow.serving.Dim")
    field(shape, "unknown_rank", 3, T.TYPE_BOOL)

    tp = msg("TensorProto")
    field(tp, "dtype", 1, T.TYPE_INT32)
    field(tp, "tensor_shape", 2, T.TYPE_MESSAGE,
          type_name=".tensorflow.serving.TensorShapeProto")
    field(tp, "version_number", 3, T.TYPE_INT32)
    field(tp, "tensor_content", 4, T.TYPE_BYTES)
    field(tp, "float_val", 5, T.TYPE_FLOAT, T.LABEL_REPEATED)
    field(tp, "double_val", 6, T.TYPE_DOUBLE, T.LABEL_REPEATED)
    field(tp, "int_val", 7, T.TYPE_INT32, T.LABEL_REPEATED)
    field(tp, "string_val", 8, T.TYPE_BYTES, T.LABEL_REPEATED)
    field(tp, "int64_val", 10, T.TYPE_INT64, T.LABEL_REPEATED)
    field(tp, "bool_val", 11, T.TYPE_BOOL, T.LABEL_REPEATED)
    field(tp, "half_val", 13, T.TYPE_INT32, T.LABEL_REPEATED)

    spec = msg("ModelSpec")
    field(spec, "name", 1, T.TYPE_STRING)
    field(spec, "version", 2, T.TYPE_MESSAGE,
          type_name=".tensorflow.serving.Int64Value")
    field(spec, "signature_name", 3, T.TYPE_STRING)

    # PredictRequest with a proper map<string, TensorProto>
    preq = msg("PredictRequest")
    entry = preq.nested_type.add(name="InputsEntry")
    entry.options.map_entry = True
    field(entry, "key", 1, T.TYPE_STRING)
    field(entry, "value", 2, T.TYPE_MESSAGE,
          type_name=".tensorflow.serving.TensorProto")
    field(preq, "model_spec", 1, T.TYPE_MESSAGE,
          type_name=".tensorflow.serving.ModelSpec")
    field(preq, "inputs", 2, T.TYPE_MESSAGE, T.LABEL_REPEATED,
          ".tensorflow.serving.PredictRequest.InputsEntry")
    field(preq, "output_filter", 3, T.TYPE_STRING, T.LABEL_REPEATED)

    fd = pool.Add(f)
    get = lambda n: message_factory.GetMessageClass(  # noqa: E731
        fd.message_types_by_name[n])
    return {n: get(n) for n in
            ["Int64Value", "TensorShapeProto", "TensorProto", "ModelSpec",
             "PredictRequest"]}


def make_predict_request():
    arr = np.arange(12, dtype=np.float32).reshape(3, 4)
    return m.PredictRequest(
        model_spec=m.ModelSpec(name="half_plus_two",
                               version=m.Int64Value(value=123),
                               signature_name="serving_default"),
        inputs={"x": wt.numpy_to_tensorproto(arr)},
        output_filter=["y"],
    )


def test_ours_decodes_in_reference(ref):
    data = make_predict_request().encode()
    theirs = ref["PredictRequest"]()
    theirs.ParseFromString(data)
    assert theirs.model_spec.name == "half_plus_two"
    assert theirs.model_spec.version.value == 123
    assert theirs.model_spec.signature_name == "serving_default"
    assert list(theirs.output_filter) == ["y"]
    t = theirs.inputs["x"]
    assert t.dtype == m.DT_FLOAT
    assert [d.size for d in t.tensor_shape.dim] == [3, 4]
    assert np.frombuffer(t.tensor_content, dtype=np.float32).tolist() == \
        list(range(12))


def test_reference_decodes_in_ours(ref):
    theirs = ref["PredictRequest"]()
    theirs.model_spec.name = "m"
    theirs.model_spec.version.value = 7
    t = theirs.inputs["inp"]
    t.dtype = m.DT_FLOAT
    t.tensor_shape.dim.add(size=3)
    t.float_val.extend([1.0, 2.0, 5.0])
    data = theirs.SerializeToString()

    ours = m.PredictRequest.decode(data)
    assert ours.model_spec.name == "m"
    assert ours.model_spec.version.value == 7
    arr = wt.tensorproto_to_numpy(ours.inputs["inp"])
    assert arr.tolist() == [1.0, 2.0, 5.0]


def test_roundtrip_bytes_stable():
    req = make_predict_request()
    data = req.encode()
    again = m.PredictRequest.decode(data).encode()
    assert data == again


def test_unknown_fields_preserved(ref):
    theirs = ref["TensorProto"]()
    theirs.dtype = m.DT_FLOAT
    theirs.version_number = 9  # field we model
    theirs.float_val.extend([1.5])
    data = theirs.SerializeToString()

    # decode with a reduced message that does not know version_number
    class Reduced(m.TensorProto.__class__("Tmp", (m.Message,), dict(FIELDS=[
            ("dtype", 1, "enum"),
            ("float_val", 5, "float", dict(repeated=True, packed=True)),
    ]))):
        pass

    red = Reduced.decode(data)
    out = red.encode()
    theirs2 = ref["TensorProto"]()
    theirs2.ParseFromString(out)
    assert theirs2.version_number == 9
    assert list(theirs2.float_val) == [1.5]


def test_negative_ints_roundtrip(ref):
    tp = m.TensorProto(dtype=m.DT_INT32, int_val=[-1, -2147483648, 3])
    theirs = ref["TensorProto"]()
    theirs.ParseFromString(tp.encode())
    assert list(theirs.int_val) == [-1, -2147483648, 3]
    back = m.TensorProto.decode(theirs.SerializeToString())
    assert back.int_val == [-1, -2147483648, 3]


def test_tensor_splat_and_dtypes():
    tp = m.TensorProto(dtype=m.DT_FLOAT,
                       tensor_shape=m.TensorShapeProto.of([4]),
                       float_val=[2.0])
    assert wt.tensorproto_to_numpy(tp).tolist() == [2.0] * 4

    for dt, np_dt in [(m.DT_INT64, np.int64), (m.DT_DOUBLE, np.float64),
                      (m.DT_BOOL, np.bool_), (m.DT_INT32, np.int32)]:
        arr = np.array([0, 1, 1, 0]).astype(np_dt)
        rt = wt.tensorproto_to_numpy(m.TensorProto.decode(
            wt.numpy_to_tensorproto(arr).encode()))
        assert rt.dtype == arr.dtype
        np.testing.assert_array_equal(rt, arr)


def test_model_server_config_roundtrip():
    cfg = m.ModelServerConfig(model_config_list=m.ModelConfigList(config=[
        m.ModelConfig(
            name="resnet", base_path="/models/resnet",
            model_platform="tensorflow",
            model_version_policy=m.ServableVersionPolicy(
                specific=m.ServableVersionPolicySpecific(versions=[1, 3])),
        )]))
    req = m.ReloadConfigRequest(config=cfg)
    back = m.ReloadConfigRequest.decode(req.encode())
    mc = back.config.model_config_list.config[0]
    assert mc.name == "resnet"
    assert mc.model_platform == "tensorflow"
    assert mc.model_version_policy.specific.versions == [1, 3]
