"""TF Serving REST JSON <-> tensors.

Implements the TF Serving REST API's predict body formats
(https://www.tensorflow.org/tfx/serving/api_rest — the wire contract the
reference proxies through unchanged):
  * row format:      {"instances": [...]}           -> {"predictions": [...]}
  * columnar format: {"inputs": ... | {name: ...}}  -> {"outputs": ...}
  * base64 binary strings {"b64": "..."} inside values.
"""
from __future__ import annotations

import base64
from typing import Dict, List, Tuple

import numpy as np


class RestCodecError(ValueError):
    pass


def _decode_b64(value):
    if isinstance(value, dict):
        if set(value.keys()) == {"b64"}:
            return base64.b64decode(value["b64"])
        return {k: _decode_b64(v) for k, v in value.items()}
    if isinstance(value, list):
        return [_decode_b64(v) for v in value]
    return value


def _to_array(value) -> np.ndarray:
    value = _decode_b64(value)
    arr = np.asarray(value)
    if arr.dtype == np.float64:
        arr = arr.astype(np.float32)
    elif arr.dtype == np.int64:
        arr = arr.astype(np.int32)
    elif arr.dtype.kind in ("U", "S"):
        # JSON strings / b64 bytes -> DT_STRING object array (decoded at
        # this tier, rejected with a clean error by the native engine;
        # proxied untouched at the routing tier)
        flat = arr.reshape(-1)
        out = np.empty(flat.shape, dtype=object)
        for i, v in enumerate(flat):
            out[i] = v.encode() if isinstance(v, str) else bytes(v)
        return out.reshape(arr.shape)
    elif arr.dtype == object:
        flat = arr.reshape(-1)
        if all(isinstance(v, (str, bytes)) for v in flat):
            out = np.empty(flat.shape, dtype=object)
            for i, v in enumerate(flat):
                out[i] = v.encode() if isinstance(v, str) else v
            return out.reshape(arr.shape)
        raise RestCodecError("ragged inputs unsupported")
    return arr


def parse_predict_body(body: dict) -> Tuple[Dict[str, np.ndarray], str, str]:
    """Returns (inputs_by_name, format, signature_name). Single-input
    models may use the anonymous form (the caller maps '' to the sole
    signature input)."""
    signature = body.get("signature_name", "")
    if "instances" in body and "inputs" in body:
        raise RestCodecError('specify only one of "instances" or "inputs"')
    if "instances" in body:
        rows = body["instances"]
        if not isinstance(rows, list):
            raise RestCodecError('"instances" must be a list')
        if rows and isinstance(rows[0], dict) and "b64" not in rows[0]:
            names = rows[0].keys()
            cols: Dict[str, list] = {n: [] for n in names}
            for r in rows:
                if not isinstance(r, dict) or r.keys() != names:
                    raise RestCodecError("inconsistent instance objects")
                for n in names:
                    cols[n].append(r[n])
            return ({n: _to_array(v) for n, v in cols.items()},
                    "row", signature)
        return ({"": _to_array(rows)}, "row", signature)
    if "inputs" in body:
        val = body["inputs"]
        if isinstance(val, dict) and "b64" not in val:
            return ({n: _to_array(v) for n, v in val.items()},
                    "col", signature)
        return ({"": _to_array(val)}, "col", signature)
    raise RestCodecError('missing "instances" or "inputs" key')


def _jsonable(arr: np.ndarray):
    if isinstance(arr, np.ndarray):
        if arr.dtype.kind == "f":
            arr = arr.astype(np.float64)
        return arr.tolist()
    return arr


def render_predict_response(outputs: Dict[str, np.ndarray],
                            fmt: str) -> dict:
    if fmt == "col":
        if len(outputs) == 1:
            return {"outputs": _jsonable(next(iter(outputs.values())))}
        return {"outputs": {k: _jsonable(v) for k, v in outputs.items()}}
    # row format: predictions[i] is row i across outputs
    if len(outputs) == 1:
        vals = next(iter(outputs.values()))
        return {"predictions": _jsonable(vals)}
    names = list(outputs)
    n_rows = {len(v) for v in outputs.values()}
    if len(n_rows) != 1:
        # outputs disagree on batch — fall back to named object
        return {"predictions": {k: _jsonable(v) for k, v in outputs.items()}}
    preds: List[dict] = []
    for i in range(n_rows.pop()):
        preds.append({n: _jsonable(outputs[n][i]) for n in names})
    return {"predictions": preds}
