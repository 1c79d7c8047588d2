"""TensorProto <-> numpy conversion.

Matches TF semantics: dense numeric tensors round-trip through
`tensor_content` (native little-endian bytes) when produced by us; incoming
tensors may instead carry the typed `*_val` repeated fields (possibly
shorter than the shape's element count, in which case the last value is
broadcast — TF's "splat" rule).
"""
from __future__ import annotations

import numpy as np

from . import messages as m

_NP_BFLOAT16 = np.dtype("uint16")  # bfloat16 carried as raw uint16 words

DTYPE_TO_NP = {
    m.DT_FLOAT: np.dtype("float32"),
    m.DT_DOUBLE: np.dtype("float64"),
    m.DT_INT32: np.dtype("int32"),
    m.DT_UINT8: np.dtype("uint8"),
    m.DT_INT16: np.dtype("int16"),
    m.DT_INT8: np.dtype("int8"),
    m.DT_INT64: np.dtype("int64"),
    m.DT_BOOL: np.dtype("bool"),
    m.DT_UINT16: np.dtype("uint16"),
    m.DT_UINT32: np.dtype("uint32"),
    m.DT_UINT64: np.dtype("uint64"),
    m.DT_HALF: np.dtype("float16"),
    m.DT_BFLOAT16: _NP_BFLOAT16,
}

NP_TO_DTYPE = {
    np.dtype("float32"): m.DT_FLOAT,
    np.dtype("float64"): m.DT_DOUBLE,
    np.dtype("int32"): m.DT_INT32,
    np.dtype("uint8"): m.DT_UINT8,
    np.dtype("int16"): m.DT_INT16,
    np.dtype("int8"): m.DT_INT8,
    np.dtype("int64"): m.DT_INT64,
    np.dtype("bool"): m.DT_BOOL,
    np.dtype("uint16"): m.DT_UINT16,
    np.dtype("uint32"): m.DT_UINT32,
    np.dtype("uint64"): m.DT_UINT64,
    np.dtype("float16"): m.DT_HALF,
}

_VAL_FIELD = {
    m.DT_FLOAT: "float_val",
    m.DT_DOUBLE: "double_val",
    m.DT_INT32: "int_val",
    m.DT_UINT8: "int_val",
    m.DT_INT16: "int_val",
    m.DT_INT8: "int_val",
    m.DT_INT64: "int64_val",
    m.DT_BOOL: "bool_val",
    m.DT_UINT16: "int_val",
    m.DT_UINT32: "uint32_val",
    m.DT_UINT64: "uint64_val",
    m.DT_HALF: "half_val",
    m.DT_BFLOAT16: "half_val",
    m.DT_STRING: "string_val",
}


class TensorCodecError(ValueError):
    pass


def tensorproto_to_numpy(tp: m.TensorProto) -> np.ndarray:
    """Decode a TensorProto into a numpy array (DT_BFLOAT16 -> uint16 words)."""
    if tp.dtype == m.DT_STRING:
        shape = tp.tensor_shape.sizes() if tp.tensor_shape else [len(tp.string_val)]
        if any(d < 0 for d in shape):
            raise TensorCodecError(f"negative dim in tensor shape {shape}")
        n = int(np.prod(shape)) if shape else 1
        # dims are attacker-declared: never allocate past the actual
        # payload (TF requires |string_val| == prod(dims), splat aside)
        if n > max(len(tp.string_val), 1):
            raise TensorCodecError(
                f"string tensor dims {shape} exceed {len(tp.string_val)} "
                "values")
        arr = np.empty(n, dtype=object)
        for i, s in enumerate(tp.string_val[:n]):
            arr[i] = s
        return arr.reshape(shape)
    np_dtype = DTYPE_TO_NP.get(tp.dtype)
    if np_dtype is None:
        raise TensorCodecError(f"unsupported dtype {tp.dtype}")
    shape = tp.tensor_shape.sizes() if tp.tensor_shape is not None else None
    if shape is not None and any(d < 0 for d in shape):
        # negative dims are signature wildcards, not valid in a request
        # tensor (numpy reshape would silently treat them as -1)
        raise TensorCodecError(f"negative dim in tensor shape {shape}")
    if tp.tensor_content:
        arr = np.frombuffer(tp.tensor_content, dtype=np_dtype)
        if shape is not None:
            arr = arr.reshape(shape)
        return arr
    vals = getattr(tp, _VAL_FIELD[tp.dtype])
    if tp.dtype == m.DT_HALF:
        arr = np.array(vals, dtype=np.uint16).view(np.float16)
    elif tp.dtype == m.DT_BFLOAT16:
        arr = np.array(vals, dtype=np.uint16)
    else:
        arr = np.array(vals, dtype=np_dtype)
    if shape is not None:
        n = int(np.prod(shape)) if shape else 1
        if arr.size == n:
            arr = arr.reshape(shape)
        elif arr.size and arr.size < n:
            # TF splat rule: repeat the last value. n is attacker-
            # declared — bound the expansion (256M elements ~ 1 GiB f32)
            if n > 1 << 28:
                raise TensorCodecError(
                    f"splat to {n} elements exceeds the 2^28 cap")
            arr = np.concatenate([arr, np.full(n - arr.size, arr[-1],
                                               dtype=arr.dtype)]).reshape(shape)
        elif arr.size == 0 and n == 0:
            arr = arr.reshape(shape)
        else:
            raise TensorCodecError(
                f"value count {arr.size} exceeds shape {shape}")
    return arr


def numpy_to_tensorproto(arr: np.ndarray, dtype: int | None = None) -> m.TensorProto:
    """Encode a numpy array as a TensorProto using tensor_content."""
    if arr.dtype == object or (dtype == m.DT_STRING):
        flat = arr.reshape(-1)
        return m.TensorProto(
            dtype=m.DT_STRING,
            tensor_shape=m.TensorShapeProto.of(arr.shape),
            string_val=[v if isinstance(v, bytes) else str(v).encode()
                        for v in flat],
        )
    if dtype is None:
        dtype = NP_TO_DTYPE.get(arr.dtype)
        if dtype is None:
            raise TensorCodecError(f"unsupported numpy dtype {arr.dtype}")
    arr = np.ascontiguousarray(arr)
    return m.TensorProto(
        dtype=dtype,
        tensor_shape=m.TensorShapeProto.of(arr.shape),
        tensor_content=arr.tobytes(),
    )
