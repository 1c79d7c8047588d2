"""Hypothesis property tests over the pure-CPU subsystems.

- Consistent-hash ring: minimal-disruption on member removal and the
  distinct-replica contract of get_n (the reference leans on
  stathat's consistent.GetN for both — cluster.go:116-130).
- LRUCache: model-checked against an independent in-test mirror of the
  byte-budget + MRU-order semantics (lrucache.go:43-101).
- Planner/CPU executor: randomized MLP dimensions end-to-end vs a
  plain numpy forward pass.
"""
import os
import tempfile

import numpy as np
import pytest
from hypothesis import given, settings, strategies as st

# TFSC_HYP_SCALE=N multiplies every max_examples (deep fuzz runs)
_SCALE = float(os.environ.get("TFSC_HYP_SCALE", "1"))


def _ex(n):
    return max(1, int(n * _SCALE))

from tfservingcache_amd.cachemanager.lrucache import LRUCache, Model
from tfservingcache_amd.taskhandler.ring import ConsistentHashRing

# ---------------------------------------------------------------------------
# ring properties
# ---------------------------------------------------------------------------

_member_st = st.text(alphabet="abcdefgh0123", min_size=1, max_size=8).map(
    lambda s: f"host-{s}:8093:8100:gpu0")
_members_st = st.lists(_member_st, min_size=2, max_size=12, unique=True)
_keys_st = st.lists(st.text(alphabet="mnopq/0123456789#", min_size=1,
                            max_size=16),
                    min_size=1, max_size=30)


@given(members=_members_st, keys=_keys_st, data=st.data())
@settings(max_examples=_ex(60), deadline=None)
def test_ring_minimal_disruption_on_removal(members, keys, data):
    """Removing one member must not remap keys it did not own: the
    surviving vnodes keep their relative order, so every key whose
    primary owner was a survivor keeps that owner."""
    ring = ConsistentHashRing()
    ring.set_members(members)
    before = {k: ring.get(k) for k in keys}
    removed = data.draw(st.sampled_from(members))
    ring.set_members([mem for mem in members if mem != removed])
    for k in keys:
        if before[k] != removed:
            assert ring.get(k) == before[k]
        else:
            assert ring.get(k) != removed


@given(members=_members_st, keys=_keys_st,
       n=st.integers(min_value=1, max_value=15))
@settings(max_examples=_ex(60), deadline=None)
def test_ring_get_n_distinct_and_prefix_stable(members, keys, n):
    """get_n returns min(n, members) DISTINCT members, every one a real
    member, and get_n(k, a) is a prefix of get_n(k, b) for a <= b (so
    growing replicasPerModel only ADDS replica slots)."""
    ring = ConsistentHashRing()
    ring.set_members(members)
    for k in keys:
        owners = ring.get_n(k, n)
        assert len(owners) == min(n, len(members))
        assert len(set(owners)) == len(owners)
        assert all(o in members for o in owners)
        for a in range(1, len(owners) + 1):
            assert ring.get_n(k, a) == owners[:a]


@given(members=_members_st, keys=_keys_st)
@settings(max_examples=_ex(40), deadline=None)
def test_ring_assignment_is_deterministic(members, keys):
    """Two independently-seeded rings over the same member set agree on
    every key (routers need no coordination — taskhandler.go:84-93)."""
    r1, r2 = ConsistentHashRing(), ConsistentHashRing()
    r1.set_members(members)
    r2.set_members(list(reversed(members)))
    for k in keys:
        assert r1.get_n(k, 3) == r2.get_n(k, 3)


# ---------------------------------------------------------------------------
# LRU cache model check
# ---------------------------------------------------------------------------

_op_st = st.one_of(
    st.tuples(st.just("put"), st.integers(0, 9), st.integers(1, 50)),
    st.tuples(st.just("get"), st.integers(0, 9), st.just(0)),
    st.tuples(st.just("remove"), st.integers(0, 9), st.just(0)),
)


@given(ops=st.lists(_op_st, min_size=1, max_size=60),
       cap=st.integers(min_value=10, max_value=120))
@settings(max_examples=_ex(80), deadline=None)
def test_lru_matches_reference_model(ops, cap):
    """Replay a random op sequence against LRUCache and an independent
    mirror of the spec: byte budget enforced by tail eviction, front =
    MRU, replace-on-put, current_size always the sum of live entries."""
    with tempfile.TemporaryDirectory() as td:
        cache = LRUCache(td, cap)
        mirror = []                      # [(name, size)] front = MRU

        def msize():
            return sum(s for _, s in mirror)

        for kind, idx, size in ops:
            name = f"m{idx}"
            if kind == "put":
                mirror[:] = [e for e in mirror if e[0] != name]
                while mirror and msize() + size > cap:
                    mirror.pop()         # evict LRU tail
                mirror.insert(0, (name, size))
                cache.put(Model(name=name, version=1,
                                path=os.path.join(name, "1"),
                                size_on_disk=size))
            elif kind == "get":
                got = cache.get(name, 1)
                hit = any(e[0] == name for e in mirror)
                assert (got is not None) == hit
                if hit:
                    e = next(e for e in mirror if e[0] == name)
                    mirror.remove(e)
                    mirror.insert(0, e)
                    assert got.size_on_disk == e[1]
            else:
                removed = cache.remove(name, 1)
                assert removed == any(e[0] == name for e in mirror)
                mirror[:] = [e for e in mirror if e[0] != name]

            assert cache.current_size == msize()
            assert [(e.name, e.size_on_disk)
                    for e in cache.list_models()] == mirror
            # budget: only a single oversized entry may exceed the cap
            assert cache.current_size <= cap or len(mirror) == 1


# ---------------------------------------------------------------------------
# StridedSlice / Concat / Pack index math over random shapes
# ---------------------------------------------------------------------------

@st.composite
def _slice_case(draw):
    """A random [-1, H, W] shape plus per-dim slice specs that are
    expressible as numpy basic indexing (positive strides; dim 0 is the
    untouched batch dim, as the lowering requires)."""
    h = draw(st.integers(2, 8))
    w = draw(st.integers(2, 8))
    specs = []
    for size in (h, w):
        kind = draw(st.sampled_from(["full", "range", "shrink"]))
        if kind == "full":
            specs.append(("full", 0, 0, draw(st.sampled_from([1, 2, 3]))))
        elif kind == "shrink":
            j = draw(st.integers(0, size - 1))
            specs.append(("shrink", j, j + 1, 1))
        else:
            b = draw(st.integers(-size, size - 1))
            canon = b if b >= 0 else b + size
            e = draw(st.integers(canon + 1, size))
            specs.append(("range", b, e, draw(st.sampled_from([1, 2, 3]))))
    # at least one sliced dim, else the planner aliases instead
    if all(k == "full" and s == 1 for k, _, _, s in specs):
        specs[0] = ("shrink", 0, 1, 1)
    return h, w, specs


@given(case=_slice_case(), batch=st.integers(1, 4),
       seed=st.integers(0, 999))
@settings(max_examples=_ex(25), deadline=None)
def test_strided_slice_random_specs(case, batch, seed):
    from tfservingcache_amd.engine.model import load_model_from_dir
    from tfservingcache_amd.engine.savedmodel import (GraphBuilder,
                                                      write_saved_model)
    h, w, specs = case
    begin, end, strides = [0], [0], [1]
    bm, em, shm = 1, 1, 0
    np_index = [slice(None)]
    for i, (kind, b, e, s) in enumerate(specs, start=1):
        begin.append(b)
        end.append(e)
        strides.append(s)
        if kind == "full":
            bm |= 1 << i
            em |= 1 << i
            np_index.append(slice(None, None, s))
        elif kind == "shrink":
            shm |= 1 << i
            np_index.append(b)
        else:
            np_index.append(slice(b, e, s))

    gb = GraphBuilder()
    f32 = gb.a_type(1)
    x_ph = gb.placeholder("x", np.float32, [-1, h, w],
                          signature_name="x")
    ss = gb.node("StridedSlice", "ss",
                 [x_ph,
                  gb.const("b", np.array(begin, np.int32)),
                  gb.const("e", np.array(end, np.int32)),
                  gb.const("s", np.array(strides, np.int32))],
                 T=f32, Index=gb.a_type(3),
                 begin_mask=gb.a_int(bm), end_mask=gb.a_int(em),
                 shrink_axis_mask=gb.a_int(shm),
                 ellipsis_mask=gb.a_int(0), new_axis_mask=gb.a_int(0))
    gb.mark_output("y", ss)

    with tempfile.TemporaryDirectory() as td:
        d = os.path.join(td, "m", "1")
        write_saved_model(gb.build(), d)
        model = load_model_from_dir(d, "m", 1)
        x = np.random.default_rng(seed).standard_normal(
            (batch, h, w)).astype(np.float32)
        got = model.predict({"x": x})["y"]
        want = x[tuple(np_index)]
        assert got.shape == want.shape, (got.shape, want.shape)
        np.testing.assert_allclose(got, want, rtol=1e-6)


@given(h=st.integers(1, 6), w=st.integers(1, 6), batch=st.integers(1, 4),
       axis=st.integers(1, 2), pack_axis=st.integers(1, 3),
       seed=st.integers(0, 999))
@settings(max_examples=_ex(25), deadline=None)
def test_concat_pack_random_shapes(h, w, batch, axis, pack_axis, seed):
    """Concat of [x, 2x] along a random non-batch axis and Pack along a
    random axis vs numpy concatenate/stack."""
    from tfservingcache_amd.engine.model import load_model_from_dir
    from tfservingcache_amd.engine.savedmodel import (GraphBuilder,
                                                      write_saved_model)
    gb = GraphBuilder()
    f32 = gb.a_type(1)
    x_ph = gb.placeholder("x", np.float32, [-1, h, w],
                          signature_name="x")
    x2 = gb.node("Mul", "x2", [x_ph, gb.const("two", np.float32(2.0))],
                 T=f32)
    cc = gb.node("ConcatV2", "cc",
                 [x_ph, x2, gb.const("ax", np.int32(axis))],
                 N=gb.a_int(2), T=f32)
    pk = gb.node("Pack", "pk", [x_ph, x2], N=gb.a_int(2), T=f32,
                 axis=gb.a_int(pack_axis))
    gb.mark_output("cat", cc)
    gb.mark_output("packed", pk)

    with tempfile.TemporaryDirectory() as td:
        d = os.path.join(td, "m", "1")
        write_saved_model(gb.build(), d)
        model = load_model_from_dir(d, "m", 1)
        x = np.random.default_rng(seed).standard_normal(
            (batch, h, w)).astype(np.float32)
        out = model.predict({"x": x})
        np.testing.assert_allclose(
            out["cat"], np.concatenate([x, 2 * x], axis=axis), rtol=1e-6)
        np.testing.assert_allclose(
            out["packed"], np.stack([x, 2 * x], axis=pack_axis),
            rtol=1e-6)


# ---------------------------------------------------------------------------
# C++ dense-JSON parser (native REST fast path) vs Python json
# ---------------------------------------------------------------------------

def _probe():
    pytest.importorskip("torch")
    try:
        from tfservingcache_amd.engine import _tfsc_engine as ext
    except Exception:                    # noqa: BLE001
        pytest.skip("engine extension not built")
    return ext._rest_parse_probe


@st.composite
def _dense_array(draw):
    shape = draw(st.lists(st.integers(1, 5), min_size=1, max_size=4))
    n = int(np.prod(shape))
    vals = draw(st.lists(
        st.floats(allow_nan=False, allow_infinity=False, width=32),
        min_size=n, max_size=n))
    return np.array(vals, dtype=np.float32).reshape(shape)


@given(x=_dense_array(), pretty=st.booleans())
@settings(max_examples=_ex(60), deadline=None)
def test_cpp_json_parser_random_floats(x, pretty):
    """The hand-written C++ JSON number/nesting parser agrees with
    Python json + numpy on arbitrary float32 payloads, compact or
    whitespace-heavy."""
    import json
    probe = _probe()
    body = json.dumps(x.tolist(), indent=1 if pretty else None).encode()
    dims, vals = probe(body, False)
    assert list(dims) == list(x.shape)
    want = np.array(json.loads(body), dtype=np.float32).reshape(-1)
    np.testing.assert_allclose(np.array(vals, dtype=np.float32), want,
                               rtol=2e-6, atol=1e-37)


@given(shape=st.lists(st.integers(1, 5), min_size=1, max_size=3),
       seed=st.integers(0, 2 ** 31 - 1))
@settings(max_examples=_ex(40), deadline=None)
def test_cpp_json_parser_random_ints(shape, seed):
    import json
    probe = _probe()
    rng = np.random.default_rng(seed)
    x = rng.integers(-2 ** 31, 2 ** 31, size=shape, dtype=np.int64) \
        .astype(np.int32)
    dims, vals = probe(json.dumps(x.tolist()).encode(), True)
    assert list(dims) == list(x.shape)
    assert list(vals) == x.reshape(-1).tolist()


# ---------------------------------------------------------------------------
# Einsum trailing-K family over random equations/shapes
# ---------------------------------------------------------------------------

@st.composite
def _einsum_case(draw):
    """Random equation in the supported Dense-layer family: lhs =
    prefix+K, rhs = K+suffix, out = prefix+suffix."""
    n_pre = draw(st.integers(1, 3))
    nk = draw(st.integers(1, 2))
    n_suf = draw(st.integers(0, 2))
    letters = "abcdefg"
    pre = letters[:n_pre]
    kk = letters[n_pre:n_pre + nk]
    suf = letters[n_pre + nk:n_pre + nk + n_suf]
    eq = f"{pre + kk},{kk + suf}->{pre + suf}"
    pre_dims = [draw(st.integers(1, 4)) for _ in range(n_pre)]
    k_dims = [draw(st.integers(1, 4)) for _ in range(nk)]
    suf_dims = [draw(st.integers(1, 4)) for _ in range(n_suf)]
    return eq, pre_dims, k_dims, suf_dims


@given(case=_einsum_case(), seed=st.integers(0, 999))
@settings(max_examples=_ex(30), deadline=None)
def test_einsum_random_equations(case, seed):
    from tfservingcache_amd.engine.model import load_model_from_dir
    from tfservingcache_amd.engine.savedmodel import (GraphBuilder,
                                                      write_saved_model)
    eq, pre_dims, k_dims, suf_dims = case
    rng = np.random.default_rng(seed)
    w = (rng.standard_normal(tuple(k_dims + suf_dims)) * 0.3).astype(
        np.float32)
    gb = GraphBuilder()
    f32 = gb.a_type(1)
    ashape = [-1] + pre_dims[1:] + k_dims
    x_ph = gb.placeholder("x", np.float32, ashape, signature_name="x")
    e = gb.node("Einsum", "ein", [x_ph, gb.const("w", w)], T=f32,
                equation=gb.a_str(eq), N=gb.a_ints([2]))
    gb.mark_output("y", e)

    with tempfile.TemporaryDirectory() as td:
        d = os.path.join(td, "m", "1")
        write_saved_model(gb.build(), d)
        model = load_model_from_dir(d, "m", 1)
        x = (rng.standard_normal(tuple(pre_dims + k_dims)) * 0.5) \
            .astype(np.float32)
        got = model.predict({"x": x})["y"]
        want = np.einsum(eq, x, w)
        assert got.shape == want.shape, (eq, got.shape, want.shape)
        np.testing.assert_allclose(got, want, rtol=2e-4, atol=2e-5)


# ---------------------------------------------------------------------------
# cross-codec: Python protobuf encoder -> C++ spec peek
# ---------------------------------------------------------------------------

@given(name=st.text(min_size=0, max_size=40),
       version=st.one_of(st.none(),
                         st.integers(0, 2 ** 62),
                         st.sampled_from([0, 1, 127, 128, 16383, 16384,
                                          2 ** 31 - 1, 2 ** 31, 2 ** 62])),
       label=st.text(min_size=0, max_size=20),
       rows=st.integers(1, 3))
@settings(max_examples=_ex(60), deadline=None)
def test_cpp_peek_spec_matches_python_encoder(name, version, label, rows):
    """Two independent codec implementations must agree: requests built
    by the Python wire encoder are peeked identically by the C++
    fastpath parser (names/labels arbitrary unicode, versions across
    varint-width boundaries, with a tensor payload present)."""
    pytest.importorskip("torch")
    try:
        from tfservingcache_amd.engine import _tfsc_engine as ext
    except Exception:                    # noqa: BLE001
        pytest.skip("engine extension not built")
    from tfservingcache_amd.wire import messages as m
    from tfservingcache_amd.wire.tensor import numpy_to_tensorproto

    spec = m.ModelSpec(
        name=name,
        version=(m.Int64Value(value=version)
                 if version is not None else None),
        version_label=label)
    req = m.PredictRequest(
        model_spec=spec,
        inputs={"x": numpy_to_tensorproto(
            np.zeros((rows, 4), dtype=np.float32))})
    got_name, got_version, got_label = ext.peek_spec(req.encode())
    assert got_name == name
    assert got_label == label
    if version is None or version == 0:
        # version 0 encodes as an empty Int64Value submessage and means
        # UNSET — the reference has the same quirk (its gRPC director
        # formats GetVersion().GetValue(), so 0 == no version,
        # tfservingproxy.go:248)
        assert got_version is None
    else:
        assert got_version == version


# ---------------------------------------------------------------------------
# planner/executor randomized shapes
# ---------------------------------------------------------------------------

@pytest.mark.slow
@given(d_in=st.integers(1, 40), d_hidden=st.integers(1, 48),
       d_out=st.integers(2, 24), batch=st.integers(1, 5),
       seed=st.integers(0, 2 ** 16))
@settings(max_examples=_ex(15), deadline=None)
def test_mlp_random_dims_match_numpy(d_in, d_hidden, d_out, batch, seed):
    """End-to-end plan+execute over arbitrary (non-tile-aligned) MLP
    dims must match a plain numpy forward of the same weights."""
    from tfservingcache_amd.engine.model import load_model_from_dir
    from tfservingcache_amd.engine.savedmodel import write_saved_model
    from tfservingcache_amd.models import build_mlp

    with tempfile.TemporaryDirectory() as td:
        d = os.path.join(td, "m", "1")
        write_saved_model(build_mlp(d_in, d_hidden, d_out, seed=seed), d)
        model = load_model_from_dir(d, "m", 1)

        rng = np.random.default_rng(seed)
        w1 = rng.standard_normal((d_in, d_hidden), dtype=np.float32) * 0.3
        b1 = rng.standard_normal(d_hidden, dtype=np.float32) * 0.1
        w2 = rng.standard_normal((d_hidden, d_out), dtype=np.float32) * 0.3
        b2 = rng.standard_normal(d_out, dtype=np.float32) * 0.1

        x = rng.standard_normal((batch, d_in), dtype=np.float32)
        h = np.maximum(x @ w1 + b1, 0.0)
        logits = h @ w2 + b2
        e = np.exp(logits - logits.max(axis=-1, keepdims=True))
        want = e / e.sum(axis=-1, keepdims=True)

        out = model.predict({"x": x})
        got = next(iter(out.values()))
        np.testing.assert_allclose(got, want, rtol=2e-4, atol=2e-5)


# ---------------------------------------------------------------------------
# config surface drift guard
# ---------------------------------------------------------------------------

def test_every_config_key_read_in_code_is_documented():
    """Every dotted config key the package reads must appear in
    config.yaml.example (possibly as a commented-out block) — guards
    the README/config parity the reference promises its users
    (README.md:27-68)."""
    import pathlib
    import re
    root = pathlib.Path(__file__).resolve().parent.parent
    keys = set()
    for p in (root / "tfservingcache_amd").rglob("*.py"):
        for mm in re.finditer(
                r'get_(?:string|int|float|bool|dict|list)\(\s*f?"([^"]+)"',
                p.read_text()):
            if "{" not in mm.group(1):        # skip f-string templates
                keys.add(mm.group(1))
    doc = (root / "config.yaml.example").read_text().lower()
    missing = []
    for key in sorted(keys):
        for seg in key.split("."):
            if seg.lower() not in doc:
                missing.append(key)
                break
    assert not missing, f"undocumented config keys: {missing}"
    assert len(keys) >= 50      # the surface should not silently shrink


# ---------------------------------------------------------------------------
# DepthwiseConv2dNative over random geometry
# ---------------------------------------------------------------------------

@given(h=st.integers(3, 12), w=st.integers(3, 12),
       c=st.sampled_from([1, 3, 4, 8]), k=st.sampled_from([1, 3, 5]),
       stride=st.integers(1, 3),
       padding=st.sampled_from(["SAME", "VALID"]),
       batch=st.integers(1, 3), seed=st.integers(0, 999))
@settings(max_examples=_ex(25), deadline=None)
def test_depthwise_random_geometry(h, w, c, k, stride, padding, batch,
                                   seed):
    """DepthwiseConv2dNative lowering over random spatial sizes,
    kernel sizes, strides and both paddings vs direct loops."""
    from tests.test_op_coverage import _naive_depthwise
    from tfservingcache_amd.engine.model import load_model_from_dir
    from tfservingcache_amd.engine.savedmodel import (GraphBuilder,
                                                      write_saved_model)
    if padding == "VALID" and (h < k or w < k):
        return                       # empty output; TF rejects too
    rng = np.random.default_rng(seed)
    wgt = (rng.standard_normal((k, k, c, 1)) * 0.3).astype(np.float32)
    gb = GraphBuilder()
    f32 = gb.a_type(1)
    x_ph = gb.placeholder("input", np.float32, [-1, h, w, c],
                          signature_name="input")
    d = gb.node("DepthwiseConv2dNative", "dw",
                [x_ph, gb.const("w", wgt)], T=f32,
                strides=gb.a_ints([1, stride, stride, 1]),
                padding=gb.a_str(padding),
                data_format=gb.a_str("NHWC"))
    gb.mark_output("y", d)

    with tempfile.TemporaryDirectory() as td:
        vdir = os.path.join(td, "m", "1")
        write_saved_model(gb.build(), vdir)
        model = load_model_from_dir(vdir, "m", 1)
        x = (rng.standard_normal((batch, h, w, c)) * 0.5).astype(
            np.float32)
        got = model.predict({"input": x})["y"]
        want = _naive_depthwise(x, wgt, stride, padding)
        assert got.shape == want.shape, (got.shape, want.shape)
        np.testing.assert_allclose(got, want, rtol=1e-4, atol=1e-5)
