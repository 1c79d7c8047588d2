"""Property-based robustness for the hand-rolled protobuf codec:
round-trips over randomized messages, resilience of the partial peek
and the full decoder against arbitrary byte garbage (servers parse
untrusted request bytes), and tensor codec round-trips."""
import numpy as np
import pytest
from hypothesis import given, settings, strategies as st

import os
# TFSC_HYP_SCALE=N multiplies every max_examples (deep fuzz runs)
_SCALE = float(os.environ.get("TFSC_HYP_SCALE", "1"))


def _ex(n):
    return max(1, int(n * _SCALE))

from tfservingcache_amd.wire import messages as m
from tfservingcache_amd.wire.pb import Message
from tfservingcache_amd.wire.tensor import (numpy_to_tensorproto,
                                            tensorproto_to_numpy)

names = st.text(
    alphabet=st.characters(min_codepoint=32, max_codepoint=126),
    max_size=40)


@settings(deadline=None, max_examples=_ex(150))
@given(name=names, version=st.integers(0, 2 ** 62),
       label=names, sig=names)
def test_model_spec_round_trip(name, version, label, sig):
    spec = m.ModelSpec(name=name, version=m.Int64Value(value=version),
                       signature_name=sig, version_label=label)
    got = m.ModelSpec.decode(spec.encode())
    assert got.name == name
    assert got.version_value() == version
    assert got.version_label == label
    assert got.signature_name == sig


@settings(deadline=None, max_examples=_ex(100))
@given(data=st.binary(max_size=400))
def test_decoders_never_crash_on_garbage(data):
    """Arbitrary bytes: decode/peek either succeed or raise a clean
    ValueError/IndexError — never hang or segfault."""
    for op in (m.PredictRequest.decode, m.peek_model_spec,
               m.GetModelStatusRequest.decode):
        try:
            op(data)
        except (ValueError, IndexError, KeyError):
            pass


@settings(deadline=None, max_examples=_ex(60))
@given(shape=st.lists(st.integers(1, 6), min_size=0, max_size=4),
       dtype=st.sampled_from([np.float32, np.int32, np.int64,
                              np.float64]),
       splat=st.booleans())
def test_tensor_round_trip(shape, dtype, splat):
    rng = np.random.default_rng(0)
    if np.issubdtype(dtype, np.floating):
        arr = rng.standard_normal(shape).astype(dtype)
    else:
        arr = rng.integers(-1000, 1000, size=shape).astype(dtype)
    tp = numpy_to_tensorproto(arr)
    out = tensorproto_to_numpy(m.TensorProto.decode(tp.encode()))
    np.testing.assert_array_equal(out, arr)


@settings(deadline=None, max_examples=_ex(80))
@given(inputs=st.dictionaries(
    names.filter(bool),
    st.lists(st.floats(-1e6, 1e6, width=32), min_size=1, max_size=8),
    min_size=0, max_size=4),
    name=names)
def test_predict_request_round_trip(inputs, name):
    req = m.PredictRequest(
        model_spec=m.ModelSpec(name=name),
        inputs={k: numpy_to_tensorproto(np.array(v, np.float32))
                for k, v in inputs.items()})
    data = req.encode()
    got = m.PredictRequest.decode(data)
    assert set(got.inputs) == set(inputs)
    for k, v in inputs.items():
        np.testing.assert_allclose(
            tensorproto_to_numpy(got.inputs[k]),
            np.array(v, np.float32))
    peek = m.peek_model_spec(data)
    assert peek.name == name


@settings(deadline=None, max_examples=_ex(60))
@given(data=st.binary(max_size=200))
def test_unknown_fields_preserved(data):
    """A message with unknown trailing fields re-encodes them verbatim
    (the proxy tier forwards messages it only partially understands)."""
    spec = m.ModelSpec(name="m").encode()
    # append a syntactically valid unknown field (tag 1000, bytes)
    from tfservingcache_amd.wire.pb import write_tag, write_varint
    buf = bytearray(spec)
    write_tag(buf, 1000, 2)
    write_varint(buf, len(data))
    buf += data
    decoded = m.ModelSpec.decode(bytes(buf))
    assert decoded.name == "m"
    re = decoded.encode()
    # unknown payload bytes survive the round trip
    assert bytes(data) in bytes(re)


@settings(deadline=None, max_examples=_ex(80))
@given(body=st.recursive(
    st.one_of(st.none(), st.booleans(), st.integers(-10, 10),
              st.floats(-100, 100, allow_nan=False), st.text(max_size=8)),
    lambda children: st.one_of(
        st.lists(children, max_size=4),
        st.dictionaries(st.text(max_size=8), children, max_size=4)),
    max_leaves=12))
def test_rest_codec_never_crashes_on_garbage(body):
    """Arbitrary JSON-shaped bodies: parse either succeeds or raises
    RestCodecError — the REST handler's 400 path — never an unhandled
    TypeError deep in numpy."""
    from tfservingcache_amd.tfservingproxy.json_codec import (
        RestCodecError, parse_predict_body)
    if not isinstance(body, dict):
        return
    try:
        parse_predict_body(body)
    except RestCodecError:
        pass


@settings(deadline=None, max_examples=_ex(60))
@given(rows=st.integers(1, 5), cols=st.integers(1, 4),
       columnar=st.booleans())
def test_rest_codec_round_trip(rows, cols, columnar):
    from tfservingcache_amd.tfservingproxy.json_codec import (
        parse_predict_body, render_predict_response)
    rng = np.random.default_rng(rows * 10 + cols)
    arr = rng.standard_normal((rows, cols)).astype(np.float32)
    if columnar:
        body = {"inputs": {"x": arr.tolist()}}
    else:
        body = {"instances": [{"x": row.tolist()} for row in arr]}
    feeds, fmt, _sig = parse_predict_body(body)
    np.testing.assert_allclose(np.asarray(feeds["x"], np.float32)
                               .reshape(arr.shape), arr, rtol=1e-6)
    out = render_predict_response({"y": arr}, fmt)
    assert ("outputs" if columnar else "predictions") in out
