"""CPU reference executor — numpy fp32 interpretation of a Plan.

Serves three purposes:
  * serving fallback when no GPU is present (tests, config #1 half_plus_two
    on CPU — BASELINE.json configs[0]);
  * ground truth for the HIP engine's numerics tests (tests compare each
    CDNA4 kernel against this fp32 path);
  * documentation of each plan op's exact semantics.
"""
from __future__ import annotations

from typing import Dict, List, Optional

import numpy as np

from .planner import Plan, PlanOp


def _act(x: np.ndarray, act: str) -> np.ndarray:
    if act == "none":
        return x
    if act == "relu":
        return np.maximum(x, 0.0)
    if act == "relu6":
        return np.clip(x, 0.0, 6.0)
    if act == "tanh":
        return np.tanh(x)
    if act == "sigmoid":
        return 1.0 / (1.0 + np.exp(-x))
    if act == "gelu":
        from scipy.special import erf
        return 0.5 * x * (1.0 + erf(x / np.sqrt(2.0)))
    raise ValueError(f"unknown act {act}")


_UNARY = {
    "relu": lambda x: np.maximum(x, 0.0),
    "relu6": lambda x: np.clip(x, 0.0, 6.0),
    "tanh": np.tanh,
    "sigmoid": lambda x: 1.0 / (1.0 + np.exp(-x)),
    "erf": None,  # filled lazily from scipy
    "sqrt": np.sqrt,
    "rsqrt": lambda x: 1.0 / np.sqrt(x),
    "exp": np.exp,
    "neg": np.negative,
    "square": np.square,
}

_BINARY = {
    "add": np.add, "sub": np.subtract, "mul": np.multiply,
    "div": np.divide, "max": np.maximum, "min": np.minimum,
    "sqdiff": lambda a, b: (a - b) ** 2,
}


class CpuExecutor:
    def __init__(self, plan: Plan):
        self.plan = plan

    def run(self, feeds: Dict[int, np.ndarray], batch: int,
            fetch: Optional[List[int]] = None) -> Dict[int, np.ndarray]:
        plan = self.plan
        vals: Dict[int, np.ndarray] = {}
        for t in plan.tensors:
            if t.kind == "weight" and t.weight is not None:
                vals[t.idx] = t.weight
        for idx, arr in feeds.items():
            if hasattr(arr, "materialize"):
                arr = arr.materialize()
            want = np.int32 if plan.tensors[idx].dtype == "i32" else np.float32
            vals[idx] = np.asarray(arr, dtype=want)
        # resolve aliases of fed tensors lazily below via _get

        def _get(i: int) -> np.ndarray:
            t = plan.tensors[i]
            if i in vals:
                return vals[i]
            if t.alias_of is not None:
                src = _get(t.alias_of)
                shape = plan.resolve_shape(t.shape, batch)
                vals[i] = src.reshape(shape)
                return vals[i]
            raise KeyError(f"tensor {i} ({t.name}) has no value")

        for op in plan.ops:
            self._run_op(op, _get, vals, batch)

        wanted = fetch if fetch is not None else list(plan.sig_outputs.values())
        return {i: _get(i) for i in wanted}

    # -- op semantics ------------------------------------------------------
    def _run_op(self, op: PlanOp, _get, vals, batch: int) -> None:
        k = op.kind
        p = op.params
        if k == "eltwise":
            fn = p["fn"]
            if len(op.inputs) == 1:
                f = _UNARY[fn]
                if fn == "erf":
                    from scipy.special import erf as _erf
                    f = _erf
                vals[op.outputs[0]] = f(_get(op.inputs[0])).astype(np.float32)
            else:
                a, b = _get(op.inputs[0]), _get(op.inputs[1])
                vals[op.outputs[0]] = _BINARY[fn](a, b).astype(np.float32)
        elif k == "gemm":
            a = _get(op.inputs[0])
            w = _get(op.inputs[1])
            if p.get("trans_a"):
                a = a.T
            if p.get("trans_b"):
                w = w.T
            y = a @ w
            ni = 2
            if p.get("has_bias"):
                y = y + _get(op.inputs[ni])
                ni += 1
            if p.get("residual"):
                y = y + _get(op.inputs[ni])
            y = _act(y, p.get("act", "none")).astype(np.float32)
            out_shape = self.plan.resolve_shape(
                self.plan.tensors[op.outputs[0]].shape, batch)
            vals[op.outputs[0]] = y.reshape(out_shape)
        elif k == "batched_gemm":
            a, b = _get(op.inputs[0]), _get(op.inputs[1])
            if p.get("trans_a"):
                a = np.swapaxes(a, -1, -2)
            if p.get("trans_b"):
                b = np.swapaxes(b, -1, -2)
            vals[op.outputs[0]] = (a @ b).astype(np.float32)
        elif k == "conv2d":
            self._conv2d(op, _get, vals)
        elif k == "depthwise_conv":
            self._depthwise(op, _get, vals)
        elif k == "pool":
            self._pool(op, _get, vals)
        elif k == "global_mean":
            x = _get(op.inputs[0])
            y = x.mean(axis=(1, 2), keepdims=p.get("keep", False))
            vals[op.outputs[0]] = y.astype(np.float32)
        elif k == "reduce_mean_mid":
            x = _get(op.inputs[0])
            y = x.mean(axis=1, keepdims=p.get("keep", False))
            vals[op.outputs[0]] = y.astype(np.float32)
        elif k == "reduce_mean_last":
            x = _get(op.inputs[0])
            y = x.mean(axis=-1, keepdims=p.get("keep", False))
            vals[op.outputs[0]] = y.astype(np.float32)
        elif k == "softmax":
            x = _get(op.inputs[0])
            mx = x.max(axis=-1, keepdims=True)
            e = np.exp(x - mx)
            vals[op.outputs[0]] = (e / e.sum(axis=-1, keepdims=True)).astype(
                np.float32)
        elif k == "layernorm":
            x = _get(op.inputs[0])
            gamma = _get(op.inputs[1])
            beta = _get(op.inputs[2])
            mean = x.mean(axis=-1, keepdims=True)
            var = ((x - mean) ** 2).mean(axis=-1, keepdims=True)
            y = (x - mean) / np.sqrt(var + p["eps"]) * gamma + beta
            vals[op.outputs[0]] = y.astype(np.float32)
        elif k == "bn_act":
            x = _get(op.inputs[0])
            scale = _get(op.inputs[1])
            shift = _get(op.inputs[2])
            vals[op.outputs[0]] = _act(x * scale + shift,
                                       p.get("act", "none")).astype(np.float32)
        elif k == "pad":
            x = _get(op.inputs[0])
            pads = [(int(lo), int(hi)) for lo, hi in p["pads"]]
            vals[op.outputs[0]] = np.pad(x, pads).astype(np.float32)
        elif k == "transpose":
            x = _get(op.inputs[0])
            vals[op.outputs[0]] = np.transpose(x, p["perm"]).copy()
        elif k == "gather":
            table = _get(op.inputs[0])
            idx = _get(op.inputs[1]).astype(np.int64)
            vals[op.outputs[0]] = table[idx].astype(np.float32)
        elif k == "attention":
            q = _get(op.inputs[0])      # [B, S, H, D]
            kk_ = _get(op.inputs[1])
            v = _get(op.inputs[2])
            scale = p["scale"]
            scores = np.einsum("bqhd,bkhd->bhqk", q, kk_) * scale
            mx = scores.max(-1, keepdims=True)
            e = np.exp(scores - mx)
            probs = e / e.sum(-1, keepdims=True)
            ctx = np.einsum("bhqk,bkhd->bqhd", probs, v)
            vals[op.outputs[0]] = ctx.astype(np.float32)
        elif k == "concat":
            parts = [_get(i) for i in op.inputs]
            vals[op.outputs[0]] = np.concatenate(parts, axis=p["axis"])
        elif k == "strided_copy":
            x = _get(op.inputs[0])
            sl = []
            for d in range(x.ndim):
                b = p["starts"][d]
                st = p["steps"][d]
                if p["shrink"][d]:
                    sl.append(b)
                else:
                    sl.append(slice(b, None, st))
            y = x[tuple(sl)]
            out_shape = self.plan.resolve_shape(
                self.plan.tensors[op.outputs[0]].shape, batch)
            vals[op.outputs[0]] = np.ascontiguousarray(
                y)[tuple(slice(0, dd) for dd in out_shape)].copy()
        elif k == "cast":
            x = _get(op.inputs[0])
            if p["mode"] == "i2f":
                vals[op.outputs[0]] = x.astype(np.float32)
            else:
                vals[op.outputs[0]] = x.astype(np.int32)
        elif k == "argmax_last":
            x = _get(op.inputs[0])
            vals[op.outputs[0]] = x.argmax(axis=-1).astype(np.int32)
        else:
            raise ValueError(f"unknown plan op {k}")

    def _conv2d(self, op: PlanOp, _get, vals) -> None:
        x = _get(op.inputs[0])          # [N,H,W,C]
        w = _get(op.inputs[1])          # [R,S,C,K]
        b = _get(op.inputs[2])          # [K]
        p = op.params
        sh, sw = p["stride"]
        pt, pb, pl, pr = p["pads"]
        R, S, C, K = w.shape
        N = x.shape[0]
        Ho, Wo = p["out_hw"]
        xp = np.pad(x, ((0, 0), (pt, pb), (pl, pr), (0, 0)))
        # im2col
        cols = np.empty((N, Ho, Wo, R * S * C), dtype=np.float32)
        for r in range(R):
            for s in range(S):
                patch = xp[:, r:r + sh * Ho:sh, s:s + sw * Wo:sw, :]
                cols[..., (r * S + s) * C:(r * S + s + 1) * C] = patch
        y = cols.reshape(-1, R * S * C) @ w.reshape(-1, K)
        y = y.reshape(N, Ho, Wo, K) + b
        ni = 3
        if p.get("residual"):
            y = y + _get(op.inputs[ni])
        vals[op.outputs[0]] = _act(y, p.get("act", "none")).astype(np.float32)

    def _depthwise(self, op: PlanOp, _get, vals) -> None:
        x = _get(op.inputs[0])          # [N,H,W,C]
        w = _get(op.inputs[1])          # [R,S,C]
        b = _get(op.inputs[2])          # [C]
        p = op.params
        sh, sw = p["stride"]
        pt, pb, pl, pr = p["pads"]
        R, S, C = w.shape
        N = x.shape[0]
        Ho, Wo = p["out_hw"]
        xp = np.pad(x, ((0, 0), (pt, pb), (pl, pr), (0, 0)))
        y = np.zeros((N, Ho, Wo, C), dtype=np.float32)
        for r in range(R):
            for s_ in range(S):
                patch = xp[:, r:r + sh * Ho:sh, s_:s_ + sw * Wo:sw, :]
                y += patch * w[r, s_]
        y = y + b
        vals[op.outputs[0]] = _act(y, p.get("act", "none")).astype(
            np.float32)

    def _pool(self, op: PlanOp, _get, vals) -> None:
        x = _get(op.inputs[0])
        p = op.params
        kh, kw = p["ksize"]
        sh, sw = p["stride"]
        pt, pb, pl, pr = p["pads"]
        Ho, Wo = p["out_hw"]
        if p["mode"] == "max":
            xp = np.pad(x, ((0, 0), (pt, pb), (pl, pr), (0, 0)),
                        constant_values=-np.inf)
        else:
            xp = np.pad(x, ((0, 0), (pt, pb), (pl, pr), (0, 0)))
        N, _, _, C = x.shape
        out = np.empty((N, Ho, Wo, C), dtype=np.float32)
        stack = np.empty((kh * kw, N, Ho, Wo, C), dtype=np.float32)
        for i in range(kh):
            for j in range(kw):
                stack[i * kw + j] = xp[:, i:i + sh * Ho:sh, j:j + sw * Wo:sw, :]
        if p["mode"] == "max":
            out = stack.max(axis=0)
        else:
            # TF AvgPool excludes padded cells from the divisor (the
            # GPU kernels count valid cells the same way)
            ones = np.pad(np.ones((1, x.shape[1], x.shape[2], 1),
                                  dtype=np.float32),
                          ((0, 0), (pt, pb), (pl, pr), (0, 0)))
            cnt = np.zeros((1, Ho, Wo, 1), dtype=np.float32)
            for i in range(kh):
                for j in range(kw):
                    cnt += ones[:, i:i + sh * Ho:sh, j:j + sw * Wo:sw, :]
            out = stack.sum(axis=0) / np.maximum(cnt, 1.0)
        vals[op.outputs[0]] = out
