"""GraphDef -> execution plan compiler.

Turns a frozen SavedModel GraphDef (savedmodel.py) into a flat op plan
the executors run — the CPU reference executor (executor_cpu.py, numpy
fp32) and the CDNA4 HIP engine (csrc/, bf16 MFMA). This replaces the graph
execution the reference outsourced to tensorflow_model_server (SURVEY.md
§2.4 table, row 1).

Design:
  * the batch dimension is symbolic ("B"); shapes resolve per batch bucket
    at execution-context build time;
  * inference-mode fusion happens here, once per model load:
      - Conv2D + BiasAdd + FusedBatchNorm* + (residual Add) + Relu
        -> one conv op with BN folded into weights/bias,
      - MatMul + BiasAdd + {Relu,Tanh,Sigmoid,GELU-pattern} -> one gemm op,
      - LayerNormalization primitive pattern -> layernorm op,
      - Identity/Reshape aliasing;
  * weights come out as fp32 numpy arrays; the GPU engine converts/lays
    them out at pool-load time.
"""
from __future__ import annotations

import math
from dataclasses import dataclass, field
from typing import Dict, List, Optional, Sequence, Tuple

import numpy as np

from ..wire import graph as g
from ..wire import messages as m
from ..wire.tensor import tensorproto_to_numpy

# Symbolic batch encoding: a NEGATIVE dim -k means k*B (B = runtime batch).
# So B itself is -1, and a [B,128,768] tensor reshaped to [-1,768] gets the
# dim -128 (= 128*B). Positive dims are concrete.
B = -1
Dim = int
Shape = Tuple[Dim, ...]


def is_sym(d: Dim) -> bool:
    return d < 0


def resolve_dim(d: Dim, batch: int) -> int:
    return (-d) * batch if d < 0 else d


class PlanError(Exception):
    pass


@dataclass
class PlanTensor:
    idx: int
    shape: Shape
    dtype: str                      # 'f32' | 'i32' (plan level; engine may bf16)
    kind: str                       # 'input' | 'weight' | 'activation'
    name: str = ""                  # graph tensor name ("node:0")
    weight: Optional[np.ndarray] = None
    alias_of: Optional[int] = None  # set for reshape/identity views


@dataclass
class PlanOp:
    kind: str
    inputs: List[int]
    outputs: List[int]
    params: dict = field(default_factory=dict)


@dataclass
class Plan:
    tensors: List[PlanTensor]
    ops: List[PlanOp]
    # signature alias -> tensor idx
    sig_inputs: Dict[str, int]
    sig_outputs: Dict[str, int]
    # graph tensor name -> idx (for SessionRun feeds/fetches)
    by_name: Dict[str, int]
    signature_def: Optional[m.SignatureDef] = None

    def weight_bytes(self) -> int:
        return sum(t.weight.nbytes for t in self.tensors
                   if t.kind == "weight" and t.weight is not None)

    def resolve_shape(self, shape: Shape, batch: int) -> Tuple[int, ...]:
        return tuple(resolve_dim(d, batch) for d in shape)


# ---------------------------------------------------------------------------
# helpers
# ---------------------------------------------------------------------------

def _tensor_name(ref: str) -> str:
    """Normalize a graph input ref: strip ^control and default :0."""
    if ref.startswith("^"):
        return ""
    if ":" not in ref:
        return ref + ":0"
    return ref


def _node_of(ref: str) -> str:
    return ref.split(":")[0].lstrip("^")


def _attr_i(node: g.NodeDef, key: str, default=0):
    a = node.attr.get(key)
    return a.i if a is not None else default


def _attr_s(node: g.NodeDef, key: str, default="") -> str:
    a = node.attr.get(key)
    return a.s.decode() if a is not None and a.s else default


def _attr_ints(node: g.NodeDef, key: str) -> List[int]:
    a = node.attr.get(key)
    return list(a.list.i) if a is not None and a.list is not None else []


def _attr_f(node: g.NodeDef, key: str, default=0.0) -> float:
    a = node.attr.get(key)
    return a.f if a is not None else default


def _attr_b(node: g.NodeDef, key: str, default=False) -> bool:
    a = node.attr.get(key)
    return a.b if a is not None else default


def numel(shape: Sequence[Dim], batch: int = 1) -> int:
    n = 1
    for d in shape:
        n *= resolve_dim(d, batch)
    return n


ACT_NONE, ACT_RELU, ACT_TANH, ACT_SIGMOID, ACT_GELU, ACT_RELU6 = \
    "none", "relu", "tanh", "sigmoid", "gelu", "relu6"

_ELTWISE_UNARY = {"Relu": "relu", "Relu6": "relu6", "Tanh": "tanh",
                  "Sigmoid": "sigmoid",
                  "Erf": "erf", "Sqrt": "sqrt", "Rsqrt": "rsqrt",
                  "Exp": "exp", "Neg": "neg", "Square": "square"}
_ELTWISE_BINARY = {"Add": "add", "AddV2": "add", "Sub": "sub", "Mul": "mul",
                   "RealDiv": "div", "Maximum": "max", "Minimum": "min",
                   "SquaredDifference": "sqdiff"}


class _Lowerer:
    """One-pass topological lowering with peephole fusion."""

    def __init__(self, graph_def: g.GraphDef,
                 signature: m.SignatureDef,
                 variables: Optional[Dict[str, np.ndarray]] = None):
        self.nodes: Dict[str, g.NodeDef] = {}
        self.order: List[g.NodeDef] = []
        for nd in graph_def.node:
            self.nodes[nd.name] = nd
            self.order.append(nd)
        self.signature = signature
        self.variables = variables or {}
        if self.variables:
            # non-frozen SavedModels carry a save/restore subgraph
            # (SaveV2/RestoreV2/Assign/string Consts) that never executes
            # at serving time — lower only ancestors of the signature
            # outputs (+ inputs), like TF Serving's session pruning
            keep = set()
            stack = [_node_of(ti.name)
                     for ti in signature.outputs.values()]
            stack += [_node_of(ti.name)
                      for ti in signature.inputs.values()]
            while stack:
                name = stack.pop()
                if name in keep or name not in self.nodes:
                    continue
                keep.add(name)
                for ref in self.nodes[name].input:
                    stack.append(_node_of(ref.lstrip("^")))
            self.order = [nd for nd in self.order if nd.name in keep]
        self.consumers: Dict[str, List[g.NodeDef]] = {}
        for nd in self.order:
            for ref in nd.input:
                if ref.startswith("^"):
                    continue
                self.consumers.setdefault(_tensor_name(ref), []).append(nd)

        self.tensors: List[PlanTensor] = []
        self.ops: List[PlanOp] = []
        self.by_name: Dict[str, int] = {}
        self.consts: Dict[str, np.ndarray] = {}
        self.fused: set = set()      # node names absorbed into a fused op
        # final-node-name -> emit callback for pre-pass-matched patterns
        self.pattern_emit: Dict[str, tuple] = {}

    # -- tensor bookkeeping ----------------------------------------------
    def new_tensor(self, shape: Shape, dtype: str, kind: str, name: str = "",
                   weight: Optional[np.ndarray] = None,
                   alias_of: Optional[int] = None) -> int:
        idx = len(self.tensors)
        t = PlanTensor(idx=idx, shape=tuple(shape), dtype=dtype, kind=kind,
                       name=name, weight=weight, alias_of=alias_of)
        self.tensors.append(t)
        if name:
            self.by_name[name] = idx
        return idx

    def tid(self, ref: str) -> int:
        name = _tensor_name(ref)
        if name not in self.by_name:
            raise PlanError(f"tensor {name} not lowered yet")
        return self.by_name[name]

    def shape_of(self, ref: str) -> Shape:
        return self.tensors[self.tid(ref)].shape

    def const_value(self, ref: str) -> Optional[np.ndarray]:
        node = _node_of(ref)
        v = self.consts.get(node)
        if v is not None:
            return v
        # lazy read straight off the NodeDef (pre-pass matching runs before
        # Const nodes are lowered); follow Identity chains
        nd = self.nodes.get(node)
        seen = 0
        while nd is not None and nd.op in ("Identity", "ReadVariableOp") \
                and seen < 8:
            nd = self.nodes.get(_node_of(nd.input[0]))
            seen += 1
        if nd is not None and nd.op == "Const":
            arr = tensorproto_to_numpy(nd.attr["value"].tensor)
            self.consts[node] = arr
            return arr
        if nd is not None and nd.op in ("VariableV2", "Variable",
                                        "VarHandleOp"):
            arr = self.variables.get(nd.name)
            if arr is not None:
                self.consts[node] = arr
            return arr
        return None

    def weight_of(self, ref: str) -> np.ndarray:
        v = self.const_value(ref)
        if v is None:
            raise PlanError(f"{ref} is not a constant")
        return v

    def sole_consumer(self, node: g.NodeDef, out: int = 0) -> Optional[g.NodeDef]:
        cons = self.consumers.get(f"{node.name}:{out}", [])
        if len(cons) == 1:
            return cons[0]
        return None

    # -- main --------------------------------------------------------------
    def run(self) -> Plan:
        sig_input_nodes = {}
        for alias, ti in self.signature.inputs.items():
            sig_input_nodes[_node_of(ti.name)] = (alias, ti)

        # pre-pass: match multi-node patterns whose constituents precede
        # their root in topological order (LayerNorm)
        for nd in self.order:
            if nd.op == "Rsqrt" and nd.name not in self.fused:
                self.match_layernorm(nd)

        for nd in self.order:
            if nd.name in self.pattern_emit:
                emit, args = self.pattern_emit.pop(nd.name)
                emit(*args)
                continue
            if nd.name in self.fused:
                continue
            self.lower_node(nd, sig_input_nodes)

        sig_inputs, sig_outputs = {}, {}
        for alias, ti in self.signature.inputs.items():
            name = _tensor_name(ti.name)
            if name in self.by_name:
                sig_inputs[alias] = self.by_name[name]
        for alias, ti in self.signature.outputs.items():
            name = _tensor_name(ti.name)
            if name not in self.by_name:
                raise PlanError(f"signature output {name} not produced")
            sig_outputs[alias] = self.by_name[name]

        return Plan(tensors=self.tensors, ops=self.ops,
                    sig_inputs=sig_inputs, sig_outputs=sig_outputs,
                    by_name=self.by_name, signature_def=self.signature)

    # -- op lowering -------------------------------------------------------
    def lower_node(self, nd: g.NodeDef, sig_input_nodes) -> None:
        op = nd.op
        out = f"{nd.name}:0"

        if op in ("NoOp",):
            return
        if op == "Placeholder" or op == "PlaceholderV2":
            shape_attr = nd.attr.get("shape")
            dims: List[Dim] = []
            if shape_attr is not None and shape_attr.shape is not None:
                for i, d in enumerate(shape_attr.shape.dim):
                    dims.append(B if d.size in (-1, 0) and i == 0
                                else (B if d.size == -1 else d.size))
            dt_attr = nd.attr.get("dtype")
            dt = dt_attr.type if dt_attr is not None else m.DT_FLOAT
            dtype = "i32" if dt in (m.DT_INT32, m.DT_INT64) else "f32"
            self.new_tensor(tuple(dims), dtype, "input", out)
            return
        if op == "Const":
            tp = nd.attr.get("value")
            arr = tensorproto_to_numpy(tp.tensor)
            self.consts[nd.name] = arr
            dtype = "i32" if arr.dtype in (np.int32, np.int64) else "f32"
            if arr.dtype == np.float64:
                arr = arr.astype(np.float32)
            self.new_tensor(tuple(arr.shape), dtype, "weight", out,
                            weight=np.asarray(arr))
            return
        if op in ("VariableV2", "Variable", "VarHandleOp"):
            arr = self.variables.get(nd.name)
            if arr is None:
                raise PlanError(
                    f"variable {nd.name} not found in the checkpoint "
                    f"bundle ({len(self.variables)} tensors loaded)")
            if arr.dtype == np.float64:
                arr = arr.astype(np.float32)
            self.consts[nd.name] = arr
            dtype = "i32" if arr.dtype in (np.int32, np.int64) else "f32"
            self.new_tensor(tuple(arr.shape), dtype, "weight", out,
                            weight=np.asarray(arr))
            return
        if op in ("ReadVariableOp",):
            src = self.tid(nd.input[0])
            root = self.tensors[src]
            self.new_tensor(root.shape, root.dtype, root.kind, out,
                            weight=root.weight, alias_of=src)
            cv = self.const_value(nd.input[0])
            if cv is not None:
                self.consts[nd.name] = cv
            return
        if op in ("Identity", "StopGradient", "PreventGradient", "Snapshot"):
            src = self.tid(nd.input[0])
            root = self.tensors[src]
            self.new_tensor(root.shape, root.dtype, root.kind, out,
                            weight=root.weight, alias_of=src)
            # keep const chain visible
            cv = self.const_value(nd.input[0])
            if cv is not None:
                self.consts[nd.name] = cv
            return
        if op == "Reshape":
            self.lower_reshape(nd, out)
            return
        if op in ("Squeeze", "ExpandDims"):
            self.lower_squeeze_expand(nd, out)
            return
        if op == "Conv2D":
            self.lower_conv(nd, out)
            return
        if op == "DepthwiseConv2dNative":
            self.lower_depthwise(nd, out)
            return
        if op == "Einsum":
            self.lower_einsum(nd, out)
            return
        if op in ("StridedSlice", "Slice"):
            self.lower_strided_slice(nd, out)
            return
        if op == "Cast":
            self.lower_cast(nd, out)
            return
        if op == "ArgMax":
            self.lower_argmax(nd, out)
            return
        if op == "MatMul":
            self.lower_matmul(nd, out)
            return
        if op == "BatchMatMulV2" or op == "BatchMatMul":
            self.lower_batched_matmul(nd, out)
            return
        if op == "BiasAdd":
            # unfused BiasAdd (producer wasn't conv/matmul)
            if _attr_s(nd, "data_format", "NHWC") != "NHWC":
                raise PlanError("only NHWC BiasAdd supported")
            x = self.tid(nd.input[0])
            b_id = self.tid(nd.input[1])
            shape = self.tensors[x].shape
            y = self.new_tensor(shape, "f32", "activation", out)
            self.ops.append(PlanOp("eltwise", [x, b_id], [y], {"fn": "add"}))
            return
        if op in ("MaxPool", "AvgPool"):
            self.lower_pool(nd, out)
            return
        if op == "Mean":
            self.lower_mean(nd, out)
            return
        if op == "Softmax":
            x = self.tid(nd.input[0])
            shape = self.tensors[x].shape
            y = self.new_tensor(shape, "f32", "activation", out)
            self.ops.append(PlanOp("softmax", [x], [y], {}))
            return
        if op == "Pad" or op == "PadV2":
            self.lower_pad(nd, out)
            return
        if op == "Transpose":
            self.lower_transpose(nd, out)
            return
        if op == "GatherV2" or op == "Gather":
            self.lower_gather(nd, out)
            return
        if op == "FusedBatchNorm" or op == "FusedBatchNormV3":
            self.lower_batchnorm_standalone(nd, out)
            return
        if op in _ELTWISE_UNARY:
            x = self.tid(nd.input[0])
            shape = self.tensors[x].shape
            y = self.new_tensor(shape, "f32", "activation", out)
            self.ops.append(PlanOp("eltwise", [x], [y],
                                   {"fn": _ELTWISE_UNARY[op]}))
            return
        if op in _ELTWISE_BINARY:
            a = self.tid(nd.input[0])
            b_ = self.tid(nd.input[1])
            sa, sb = self.tensors[a].shape, self.tensors[b_].shape
            shape = _broadcast(sa, sb)
            y = self.new_tensor(shape, "f32", "activation", out)
            self.ops.append(PlanOp("eltwise", [a, b_], [y],
                                   {"fn": _ELTWISE_BINARY[op]}))
            return
        if op == "ConcatV2" or op == "Concat":
            self.lower_concat(nd, out)
            return
        if op == "Pack":
            self.lower_pack(nd, out)
            return
        if op == "Unpack":
            self.lower_unpack(nd)
            return
        if op == "LeakyRelu":
            x = self.tid(nd.input[0])
            shape = self.tensors[x].shape
            alpha = _attr_f(nd, "alpha", 0.2)
            a_id = self.new_tensor((), "f32", "weight",
                                   f"{nd.name}/alpha",
                                   weight=np.float32(alpha))
            sc = self.new_tensor(shape, "f32", "activation",
                                 f"{nd.name}/scaled:0")
            self.ops.append(PlanOp("eltwise", [x, a_id], [sc],
                                   {"fn": "mul"}))
            y = self.new_tensor(shape, "f32", "activation", out)
            self.ops.append(PlanOp("eltwise", [x, sc], [y],
                                   {"fn": "max"}))
            return
        if op == "Pow":
            x = self.tid(nd.input[0])
            e = self.const_value(nd.input[1])
            ev = float(np.asarray(e).reshape(-1)[0]) if e is not None \
                and np.asarray(e).size == 1 else None
            shape = self.tensors[x].shape
            if ev == 2.0:
                y = self.new_tensor(shape, "f32", "activation", out)
                self.ops.append(PlanOp("eltwise", [x], [y],
                                       {"fn": "square"}))
                return
            if ev == 0.5:
                y = self.new_tensor(shape, "f32", "activation", out)
                self.ops.append(PlanOp("eltwise", [x], [y],
                                       {"fn": "sqrt"}))
                return
            if ev == 1.0:
                root = self.tensors[x]
                self.new_tensor(root.shape, root.dtype, root.kind, out,
                                alias_of=x)
                return
            raise PlanError(f"Pow exponent {ev} unsupported")
        if op == "Shape":
            # produce a const from the (possibly symbolic) shape
            shape = self.shape_of(nd.input[0])
            vals = np.array([(-1 if is_sym(d) else d) for d in shape],
                            dtype=np.int32)
            self.consts[nd.name] = vals
            self.new_tensor((len(vals),), "i32", "weight", out, weight=vals)
            return
        raise PlanError(f"unsupported op {op} (node {nd.name})")

    # -- individual lowerings ---------------------------------------------
    def lower_reshape(self, nd: g.NodeDef, out: str) -> None:
        x = self.tid(nd.input[0])
        xt = self.tensors[x]
        target = self.const_value(nd.input[1])
        if target is None:
            raise PlanError("dynamic Reshape target unsupported")
        has_batch = any(is_sym(d) for d in xt.shape)
        # total element count of the input, as coeff (× B if has_batch)
        total = 1
        for d in xt.shape:
            total *= -d if is_sym(d) else d
        dims: List[Dim] = [int(v) for v in target]
        n_wild = sum(1 for d in dims if d == -1)
        if n_wild > 1:
            raise PlanError("Reshape with multiple -1 dims")
        if n_wild == 1:
            fixed = 1
            for d in dims:
                if d != -1:
                    fixed *= d
            coeff = total // max(fixed, 1)
            # the wildcard absorbs the batch dependence when the input has it
            dims[dims.index(-1)] = -coeff if has_batch else coeff
        elif has_batch:
            raise PlanError("Reshape of batched tensor needs a -1 dim")
        self.new_tensor(tuple(dims), xt.dtype, xt.kind, out,
                        weight=(xt.weight.reshape(dims)
                                if xt.weight is not None and
                                not any(is_sym(d) for d in dims) else None),
                        alias_of=(x if xt.kind != "weight" else None))

    def lower_squeeze_expand(self, nd: g.NodeDef, out: str) -> None:
        x = self.tid(nd.input[0])
        xt = self.tensors[x]
        if nd.op == "Squeeze":
            axes = _attr_ints(nd, "squeeze_dims") or _attr_ints(nd, "axis")
            shape = [d for i, d in enumerate(xt.shape)
                     if not (i in axes or (not axes and d == 1))]
        else:
            axis_v = self.const_value(nd.input[1])
            axis = (int(np.asarray(axis_v).reshape(-1)[0])
                    if axis_v is not None else 0)
            shape = list(xt.shape)
            if axis < 0:
                axis += len(shape) + 1
            shape.insert(axis, 1)
        self.new_tensor(tuple(shape), xt.dtype, xt.kind, out,
                        alias_of=(x if xt.kind != "weight" else None))

    def _absorb_act_chain(self, node: g.NodeDef):
        """Follow sole-consumer chain from a conv/matmul-ish node, absorbing
        BiasAdd / BN / residual-Add / activation. Returns
        (bias, bn, residual_ref, act, final_node)."""
        bias = None
        bn = None          # (scale, offset, mean, var, eps)
        residual = None
        act = ACT_NONE
        cur = node
        while True:
            nxt = self.sole_consumer(cur)
            if nxt is None:
                break
            if nxt.op == "BiasAdd" and bias is None and bn is None and \
                    residual is None and act == ACT_NONE and \
                    _attr_s(nxt, "data_format", "NHWC") == "NHWC" and \
                    self.const_value(nxt.input[1]) is not None:
                bias = np.asarray(self.weight_of(nxt.input[1]), dtype=np.float32)
            elif nxt.op in ("FusedBatchNorm", "FusedBatchNormV3") and \
                    bn is None and residual is None and act == ACT_NONE and \
                    not _attr_b(nxt, "is_training", False) and \
                    _attr_s(nxt, "data_format", "NHWC") == "NHWC" and \
                    all(self.const_value(r) is not None for r in nxt.input[1:5]):
                bn = tuple(np.asarray(self.weight_of(r), dtype=np.float32)
                           for r in nxt.input[1:5]) + \
                    (_attr_f(nxt, "epsilon", 1e-3),)
            elif nxt.op in ("Add", "AddV2") and residual is None and \
                    act == ACT_NONE:
                other = [r for r in nxt.input
                         if _node_of(r) != cur.name]
                if len(other) != 1 or self.const_value(other[0]) is not None:
                    break
                # residual operand must already be lowered
                if _tensor_name(other[0]) not in self.by_name:
                    break
                residual = other[0]
            elif nxt.op == "Relu" and act == ACT_NONE:
                act = ACT_RELU
            elif nxt.op == "Relu6" and act == ACT_NONE:
                act = ACT_RELU6
            elif nxt.op == "Tanh" and act == ACT_NONE and residual is None:
                act = ACT_TANH
            elif nxt.op == "Sigmoid" and act == ACT_NONE and residual is None:
                act = ACT_SIGMOID
            else:
                break
            self.fused.add(nxt.name)
            cur = nxt
            if act != ACT_NONE:
                break
        return bias, bn, residual, act, cur

    def lower_conv(self, nd: g.NodeDef, out: str) -> None:
        x_ref, w_ref = nd.input[0], nd.input[1]
        x = self.tid(x_ref)
        if _attr_s(nd, "data_format", "NHWC") != "NHWC":
            raise PlanError("only NHWC Conv2D supported")
        w = np.asarray(self.weight_of(w_ref), dtype=np.float32)  # [R,S,Cin,K]
        dil = _attr_ints(nd, "dilations") or [1, 1, 1, 1]
        if any(d != 1 for d in dil):
            raise PlanError(f"dilated Conv2D unsupported ({dil})")
        strides = _attr_ints(nd, "strides") or [1, 1, 1, 1]
        padding = _attr_s(nd, "padding", "SAME")
        xs = self.tensors[x].shape                      # [B,H,W,C]
        R, S, Cin, K = w.shape
        H, W = int(xs[1]), int(xs[2])
        sh, sw = strides[1], strides[2]
        if padding == "SAME":
            Ho, Wo = math.ceil(H / sh), math.ceil(W / sw)
            pad_h = max((Ho - 1) * sh + R - H, 0)
            pad_w = max((Wo - 1) * sw + S - W, 0)
            pads = (pad_h // 2, pad_h - pad_h // 2,
                    pad_w // 2, pad_w - pad_w // 2)
        elif padding == "VALID":
            Ho, Wo = (H - R) // sh + 1, (W - S) // sw + 1
            pads = (0, 0, 0, 0)
        elif padding == "EXPLICIT":
            ep = _attr_ints(nd, "explicit_paddings")
            pads = (ep[2], ep[3], ep[4], ep[5])
            Ho = (H + pads[0] + pads[1] - R) // sh + 1
            Wo = (W + pads[2] + pads[3] - S) // sw + 1
        else:
            raise PlanError(f"padding {padding}")

        bias, bn, residual, act, final = self._absorb_act_chain(nd)
        if bn is not None:
            scale, offset, mean, var, eps = bn
            g_ = scale / np.sqrt(var + eps)
            w = w * g_.reshape(1, 1, 1, K)
            base = bias if bias is not None else np.zeros(K, np.float32)
            bias = (base - mean) * g_ + offset
        if bias is None:
            bias = np.zeros(K, np.float32)
        w_id = self.new_tensor(tuple(w.shape), "f32", "weight",
                               f"{nd.name}/fused_w", weight=w)
        b_id = self.new_tensor((K,), "f32", "weight",
                               f"{nd.name}/fused_b", weight=bias)
        out_name = f"{final.name}:0"
        y = self.new_tensor((B, Ho, Wo, K), "f32", "activation", out_name)
        inputs = [x, w_id, b_id]
        params = {"stride": (sh, sw), "pads": pads, "act": act,
                  "rsck": (R, S, Cin, K), "hw": (H, W), "out_hw": (Ho, Wo)}
        if residual is not None:
            inputs.append(self.tid(residual))
            params["residual"] = True
        self.ops.append(PlanOp("conv2d", inputs, [y], params))
        if out_name != out:
            self.by_name[out] = y   # conv node's own tensor name -> fused out

    def lower_depthwise(self, nd: g.NodeDef, out: str) -> None:
        """DepthwiseConv2dNative (MobileNet-class), depth_multiplier==1.
        Per-channel R x S taps — no MFMA shape, lowered to the dedicated
        memory-bound kernel (csrc/ops/ops_memory.hip depthwise). BN and
        Relu/Relu6 consumers fold into the kernel like Conv2D."""
        x_ref, w_ref = nd.input[0], nd.input[1]
        x = self.tid(x_ref)
        if _attr_s(nd, "data_format", "NHWC") != "NHWC":
            raise PlanError("only NHWC DepthwiseConv2dNative supported")
        w = np.asarray(self.weight_of(w_ref), dtype=np.float32)  # [R,S,C,M]
        R, S, Cin, M = w.shape
        if M != 1:
            raise PlanError(
                f"depth_multiplier {M} unsupported (only 1)")
        dil = _attr_ints(nd, "dilations") or [1, 1, 1, 1]
        if any(d != 1 for d in dil):
            raise PlanError(f"dilated depthwise conv unsupported ({dil})")
        strides = _attr_ints(nd, "strides") or [1, 1, 1, 1]
        padding = _attr_s(nd, "padding", "SAME")
        xs = self.tensors[x].shape
        H, W = int(xs[1]), int(xs[2])
        sh, sw = strides[1], strides[2]
        if padding == "SAME":
            Ho, Wo = math.ceil(H / sh), math.ceil(W / sw)
            pad_h = max((Ho - 1) * sh + R - H, 0)
            pad_w = max((Wo - 1) * sw + S - W, 0)
            pads = (pad_h // 2, pad_h - pad_h // 2,
                    pad_w // 2, pad_w - pad_w // 2)
        elif padding == "VALID":
            Ho, Wo = (H - R) // sh + 1, (W - S) // sw + 1
            pads = (0, 0, 0, 0)
        else:
            raise PlanError(f"depthwise padding {padding}")

        bias, bn, residual, act, final = self._absorb_act_chain(nd)
        if bn is not None:
            scale, offset, mean, var, eps = bn
            g_ = scale / np.sqrt(var + eps)
            w = w * g_.reshape(1, 1, Cin, 1)
            base = bias if bias is not None else np.zeros(Cin, np.float32)
            bias = (base - mean) * g_ + offset
        if bias is None:
            bias = np.zeros(Cin, np.float32)
        w_id = self.new_tensor((R, S, Cin), "f32", "weight",
                               f"{nd.name}/fused_w",
                               weight=w.reshape(R, S, Cin))
        b_id = self.new_tensor((Cin,), "f32", "weight",
                               f"{nd.name}/fused_b", weight=bias)
        out_name = f"{final.name}:0"
        dw_act = act if residual is None else ACT_NONE
        y = self.new_tensor((B, Ho, Wo, Cin), "f32", "activation",
                            out_name if residual is None
                            else f"{nd.name}/dw:0")
        self.ops.append(PlanOp("depthwise_conv", [x, w_id, b_id], [y],
                               {"stride": (sh, sw), "pads": pads,
                                "act": dw_act, "rsc": (R, S, Cin),
                                "hw": (H, W), "out_hw": (Ho, Wo)}))
        if residual is not None:
            # rare shape (MobileNet residuals connect 1x1 convs): emit
            # the absorbed add/act as separate eltwise ops
            y2 = self.new_tensor((B, Ho, Wo, Cin), "f32", "activation",
                                 out_name if act == ACT_NONE
                                 else f"{nd.name}/res:0")
            self.ops.append(PlanOp("eltwise", [y, self.tid(residual)],
                                   [y2], {"fn": "add"}))
            y = y2
            if act != ACT_NONE:
                y3 = self.new_tensor((B, Ho, Wo, Cin), "f32",
                                     "activation", out_name)
                self.ops.append(PlanOp("eltwise", [y], [y3], {"fn": act}))
                y = y3
        if out_name != out:
            self.by_name[out] = y

    def lower_strided_slice(self, nd: g.NodeDef, out: str) -> None:
        """StridedSlice / Slice with constant begin/end/strides (the
        common export shapes: CLS-token extraction x[:, 0], windowing,
        channel slices). Positive strides only; no ellipsis/new-axis
        masks; the symbolic batch dim must be taken whole. Lowered to a
        strided-copy (the transpose kernel's arbitrary-stride gather
        with a baked-in element offset)."""
        x = self.tid(nd.input[0])
        xt = self.tensors[x]
        xs = list(xt.shape)
        rank = len(xs)
        begin = self.const_value(nd.input[1])
        if nd.op == "Slice":
            sizes = self.const_value(nd.input[2])
            if begin is None or sizes is None:
                raise PlanError("dynamic Slice unsupported")
            begin = [int(b) for b in begin]
            end = []
            for d in range(len(begin)):
                sz = int(sizes[d])
                if sz == -1:
                    end.append(None)
                else:
                    end.append(begin[d] + sz)
            strides = [1] * len(begin)
            bm = em = sm = 0
        else:
            end_v = self.const_value(nd.input[2])
            str_v = self.const_value(nd.input[3])
            if begin is None or end_v is None or str_v is None:
                raise PlanError("dynamic StridedSlice unsupported")
            if _attr_i(nd, "ellipsis_mask", 0) or \
                    _attr_i(nd, "new_axis_mask", 0):
                raise PlanError("StridedSlice ellipsis/new-axis masks "
                                "unsupported")
            begin = [int(b) for b in begin]
            end = [int(e) for e in end_v]
            strides = [int(s_) for s_ in str_v]
            bm = _attr_i(nd, "begin_mask", 0)
            em = _attr_i(nd, "end_mask", 0)
            sm = _attr_i(nd, "shrink_axis_mask", 0)
        if xt.dtype != "f32":
            raise PlanError("StridedSlice on non-f32 unsupported")

        starts, steps, shrink, out_dims = [], [], [], []
        for d in range(rank):
            if d < len(begin):
                b = begin[d]
                e = end[d] if d < len(end) else None
                st = strides[d] if d < len(strides) else 1
                if st <= 0:
                    raise PlanError("negative StridedSlice stride")
                if (bm >> d) & 1:
                    b = 0
                if nd.op == "StridedSlice" and (em >> d) & 1:
                    e = None
                sh = bool((sm >> d) & 1)
            else:
                b, e, st, sh = 0, None, 1, False
            if is_sym(xs[d]):
                if b != 0 or e is not None or st != 1 or sh:
                    raise PlanError("slicing the batch dim unsupported")
                starts.append(0)
                steps.append(1)
                shrink.append(False)
                out_dims.append(xs[d])
                continue
            size = int(xs[d])
            if b < 0:
                b += size
            b = max(0, min(b, size))
            if sh:
                e = b + 1
            elif e is None:
                e = size
            else:
                if e < 0:
                    e += size
                e = max(0, min(e, size))
            starts.append(b)
            steps.append(st)
            shrink.append(sh)
            if not sh:
                out_dims.append(max(0, (e - b + st - 1) // st))
        y = self.new_tensor(tuple(out_dims), "f32", "activation", out)
        self.ops.append(PlanOp("strided_copy", [x], [y],
                               {"starts": starts, "steps": steps,
                                "shrink": shrink}))

    def lower_cast(self, nd: g.NodeDef, out: str) -> None:
        x = self.tid(nd.input[0])
        xt = self.tensors[x]
        dt_attr = nd.attr.get("DstT")
        dst_t = dt_attr.type if dt_attr is not None else m.DT_FLOAT
        dst = "i32" if dst_t in (m.DT_INT32, m.DT_INT64) else "f32"
        cv = self.const_value(nd.input[0])
        if cv is not None:
            # constant fold
            arr = np.asarray(cv)
            arr = arr.astype(np.int32 if dst == "i32" else np.float32)
            self.consts[nd.name] = arr
            self.new_tensor(tuple(arr.shape), dst, "weight", out,
                            weight=arr)
            return
        if xt.dtype == dst:
            self.new_tensor(xt.shape, dst, xt.kind, out, alias_of=x)
            return
        y = self.new_tensor(xt.shape, dst, "activation", out)
        mode = "i2f" if xt.dtype == "i32" else "f2i"
        self.ops.append(PlanOp("cast", [x], [y], {"mode": mode}))

    def lower_argmax(self, nd: g.NodeDef, out: str) -> None:
        """ArgMax over the LAST axis (classification heads). Output is
        int32 (TF defaults to int64 on the wire; int32 carries the same
        indices for any realistic class count)."""
        x = self.tid(nd.input[0])
        xs = self.tensors[x].shape
        axis_v = self.const_value(nd.input[1]) if len(nd.input) > 1 \
            else np.int32(-1)
        axis = int(np.asarray(axis_v).reshape(-1)[0])
        if axis < 0:
            axis += len(xs)
        if axis != len(xs) - 1:
            raise PlanError("ArgMax only over the last axis")
        y = self.new_tensor(tuple(xs[:-1]), "i32", "activation", out)
        self.ops.append(PlanOp("argmax_last", [x], [y], {}))

    def lower_einsum(self, nd: g.NodeDef, out: str) -> None:
        """Einsum (transformer SavedModel exports). Supported family:
        2-operand contractions where the contracted indices are the
        TRAILING indices of the lhs and the LEADING indices of the rhs
        (in the same order), the rhs is a constant weight, and the
        output is lhs-prefix + rhs-suffix — i.e. every Dense-layer
        einsum ("ij,jk->ik", "abc,cd->abd", "abc,cde->abde",
        "abcd,cde->abe", ...). Lowered to the MFMA GEMM on a reshaped
        view. Anything else raises PlanError (loudly unsupported)."""
        eq = _attr_s(nd, "equation", "").replace(" ", "")
        if len(nd.input) != 2 or "->" not in eq or "." in eq:
            raise PlanError(f"unsupported einsum {eq!r}")
        lhs_rhs, outs = eq.split("->", 1)
        if "," not in lhs_rhs:
            raise PlanError(f"unsupported einsum {eq!r}")
        lhs, rhs = lhs_rhs.split(",", 1)
        if len(set(lhs)) != len(lhs) or len(set(rhs)) != len(rhs):
            raise PlanError(f"repeated index in einsum {eq!r}")
        kchars = [ch for ch in lhs if ch in rhs and ch not in outs]
        nk = len(kchars)
        if nk == 0 or lhs[-nk:] != rhs[:nk] or \
                outs != lhs[:-nk] + rhs[nk:]:
            raise PlanError(f"einsum {eq!r} is not a trailing-K "
                            "contraction (unsupported)")
        a = self.tid(nd.input[0])
        sa = self.tensors[a].shape
        w = self.const_value(nd.input[1])
        if w is None:
            raise PlanError(
                f"einsum {eq!r}: rhs must be a constant weight")
        w = np.asarray(w, dtype=np.float32)
        kflat = int(np.prod(w.shape[:nk]))
        suffix = tuple(int(d) for d in w.shape[nk:])
        nflat = int(np.prod(suffix)) if suffix else 1
        w2 = w.reshape(kflat, nflat)
        # reshape lhs to [..., kflat] via an alias (K dims contiguous)
        a2 = a
        if nk > 1:
            a2 = self.new_tensor(tuple(sa[:-nk]) + (kflat,), "f32",
                                 "activation", f"{nd.name}/a2d",
                                 alias_of=a)
        act = ACT_NONE
        bias = None
        residual = None
        final = nd
        if len(suffix) <= 1:
            bias, bn, residual, act, final = self._absorb_act_chain(nd)
            if bn is not None:
                raise PlanError("BN after Einsum unsupported")
            if act == ACT_NONE and residual is None:
                gl = self.try_match_gelu(final)
                if gl is not None:
                    act = ACT_GELU
                    final = gl
        w_id = self.new_tensor((kflat, nflat), "f32", "weight",
                               f"{nd.name}/w2d", weight=w2)
        out_name = f"{final.name}:0"
        y = self.new_tensor(tuple(sa[:-nk]) + suffix, "f32",
                            "activation", out_name)
        inputs = [a2, w_id]
        params = {"trans_a": False, "trans_b": False, "act": act}
        if bias is not None:
            bias_id = self.new_tensor(bias.shape, "f32", "weight",
                                      f"{nd.name}/fused_bias",
                                      weight=bias)
            inputs.append(bias_id)
            params["has_bias"] = True
        if residual is not None:
            inputs.append(self.tid(residual))
            params["residual"] = True
        self.ops.append(PlanOp("gemm", inputs, [y], params))
        if out_name != out:
            self.by_name[out] = y

    def lower_matmul(self, nd: g.NodeDef, out: str) -> None:
        a_ref, b_ref = nd.input[0], nd.input[1]
        a = self.tid(a_ref)
        ta = _attr_b(nd, "transpose_a", False)
        tb = _attr_b(nd, "transpose_b", False)
        b_t = self.tensors[self.tid(b_ref)]
        sa = self.tensors[a].shape
        Mdim = sa[1] if ta else sa[0]
        Kdim = sa[0] if ta else sa[1]
        N = b_t.shape[0] if tb else b_t.shape[1]

        bias, bn, residual, act, final = self._absorb_act_chain(nd)
        # GELU pattern: matmul -> [bias] -> the erf-gelu subgraph
        if act == ACT_NONE and residual is None:
            gl = self.try_match_gelu(final)
            if gl is not None:
                act = ACT_GELU
                final = gl
        b_id = self.tid(b_ref)
        out_name = f"{final.name}:0"
        y = self.new_tensor((Mdim, N), "f32", "activation", out_name)
        inputs = [a, b_id]
        params = {"trans_a": ta, "trans_b": tb, "act": act}
        if bn is not None:
            raise PlanError("BN after MatMul unsupported")
        if bias is not None:
            bias_id = self.new_tensor(bias.shape, "f32", "weight",
                                      f"{nd.name}/fused_bias", weight=bias)
            inputs.append(bias_id)
            params["has_bias"] = True
        if residual is not None:
            inputs.append(self.tid(residual))
            params["residual"] = True
        self.ops.append(PlanOp("gemm", inputs, [y], params))
        if out_name != out:
            self.by_name[out] = y

    def try_match_gelu(self, nd: g.NodeDef) -> Optional[g.NodeDef]:
        """Matches x*0.5*(1+erf(x*rsqrt2)) rooted at nd (the producer of x).

        Canonical frozen-graph decomposition:
            m1 = Mul(x, 0.70710678)   erf = Erf(m1)
            a1 = AddV2(erf, 1.0)      m2 = Mul(x, 0.5)
            y  = Mul(m2, a1)
        Tolerates operand order swaps.
        """
        x_name = f"{nd.name}:0"
        cons = self.consumers.get(x_name, [])
        if len(cons) != 2:
            return None
        def mul_const(n):
            if n.op != "Mul":
                return None
            for r in n.input:
                cv = self.const_value(r)
                if cv is not None and cv.size == 1:
                    return float(np.asarray(cv).reshape(-1)[0])
            return None
        c_map = {}
        for c in cons:
            v = mul_const(c)
            if v is None:
                return None
            c_map[round(v, 4)] = c
        m1 = c_map.get(round(0.70710678, 4))
        m2 = c_map.get(0.5)
        if m1 is None or m2 is None:
            return None
        erf = self.sole_consumer(m1)
        if erf is None or erf.op != "Erf":
            return None
        a1 = self.sole_consumer(erf)
        if a1 is None or a1.op not in ("Add", "AddV2"):
            return None
        one = [self.const_value(r) for r in a1.input]
        if not any(v is not None and v.size == 1 and float(np.asarray(v).reshape(-1)[0]) == 1.0
                   for v in one):
            return None
        y = self.sole_consumer(a1)
        y2 = self.sole_consumer(m2)
        if y is None or y is not y2 or y.op != "Mul":
            return None
        for n in (m1, m2, erf, a1, y):
            self.fused.add(n.name)
        return y

    def match_layernorm(self, rsqrt: g.NodeDef) -> bool:
        """Match the canonical LayerNorm primitive pattern rooted at
        Rsqrt(AddV2(Mean(SquaredDifference(x, Mean(x))), eps)).

        Runs as a PRE-pass (constituent Mean/SquaredDifference nodes precede
        the Rsqrt in topological order); on a match all constituents are
        marked fused and a deferred emit is registered at the final AddV2."""
        addeps = self.nodes.get(_node_of(rsqrt.input[0]))
        if addeps is None or addeps.op not in ("Add", "AddV2"):
            return False
        eps = None
        var_node = None
        for r in addeps.input:
            cv = self.const_value(r)
            if cv is not None and cv.size == 1:
                eps = float(np.asarray(cv).reshape(-1)[0])
            else:
                var_node = self.nodes.get(_node_of(r))
        if eps is None or var_node is None or var_node.op != "Mean":
            return False
        sqd = self.nodes.get(_node_of(var_node.input[0]))
        if sqd is None or sqd.op != "SquaredDifference":
            return False
        mean_node = None
        x_ref = None
        for r in sqd.input:
            n_ = self.nodes.get(_node_of(r))
            if n_ is not None and n_.op == "Mean":
                mean_node = n_
            else:
                x_ref = r
        if mean_node is None or x_ref is None:
            return False
        if _tensor_name(mean_node.input[0]) != _tensor_name(x_ref):
            return False
        # downstream: sub = Sub(x, mean); mul = Mul(sub, rsqrt);
        # y = AddV2(Mul(mul, gamma), beta)
        mul1 = self.sole_consumer(rsqrt)
        if mul1 is None or mul1.op != "Mul":
            return False
        sub = None
        for r in mul1.input:
            n_ = self.nodes.get(_node_of(r))
            if n_ is not None and n_.op == "Sub":
                sub = n_
        if sub is None:
            return False
        if _tensor_name(sub.input[0]) != _tensor_name(x_ref) or \
                _node_of(sub.input[1]) != mean_node.name:
            return False
        mul_g = self.sole_consumer(mul1)
        if mul_g is None or mul_g.op != "Mul":
            return False
        gamma = None
        for r in mul_g.input:
            cv = self.const_value(r)
            if cv is not None and cv.ndim == 1:
                gamma = cv.astype(np.float32)
        if gamma is None:
            return False
        add_b = self.sole_consumer(mul_g)
        if add_b is None or add_b.op not in ("Add", "AddV2"):
            return False
        beta = None
        for r in add_b.input:
            cv = self.const_value(r)
            if cv is not None and cv.ndim == 1:
                beta = cv.astype(np.float32)
        if beta is None:
            return False

        members = {rsqrt, addeps, var_node, sqd, mean_node, sub, mul1, mul_g,
                   add_b}
        member_names = {n_.name for n_ in members}
        # intermediates must not leak outside the pattern
        for n_ in members - {add_b}:
            for c in self.consumers.get(f"{n_.name}:0", []):
                if c.name not in member_names:
                    return False
        for n_ in members:
            self.fused.add(n_.name)

        def emit(x_ref, gamma, beta, eps, out_node):
            x = self.tid(x_ref)
            shape = self.tensors[x].shape
            g_id = self.new_tensor(gamma.shape, "f32", "weight",
                                   f"{out_node}/ln_gamma", weight=gamma)
            b_id = self.new_tensor(beta.shape, "f32", "weight",
                                   f"{out_node}/ln_beta", weight=beta)
            y = self.new_tensor(shape, "f32", "activation", f"{out_node}:0")
            self.ops.append(PlanOp("layernorm", [x, g_id, b_id], [y],
                                   {"eps": eps}))

        self.pattern_emit[add_b.name] = (
            emit, (x_ref, gamma, beta, eps, add_b.name))
        return True

    def lower_batched_matmul(self, nd: g.NodeDef, out: str) -> None:
        a = self.tid(nd.input[0])
        b_ = self.tid(nd.input[1])
        ta = _attr_b(nd, "adj_x", False)
        tb = _attr_b(nd, "adj_y", False)
        sa, sb = self.tensors[a].shape, self.tensors[b_].shape
        Mdim = sa[-1] if ta else sa[-2]
        N = sb[-2] if tb else sb[-1]
        batch = _broadcast(sa[:-2], sb[:-2])
        shape = tuple(batch) + (Mdim, N)
        y = self.new_tensor(shape, "f32", "activation", out)
        self.ops.append(PlanOp("batched_gemm", [a, b_], [y],
                               {"trans_a": ta, "trans_b": tb}))

    def lower_pool(self, nd: g.NodeDef, out: str) -> None:
        if _attr_s(nd, "data_format", "NHWC") != "NHWC":
            raise PlanError(f"only NHWC {nd.op} supported")
        x = self.tid(nd.input[0])
        ks = _attr_ints(nd, "ksize")
        st = _attr_ints(nd, "strides")
        padding = _attr_s(nd, "padding", "VALID")
        xs = self.tensors[x].shape
        H, W, C = int(xs[1]), int(xs[2]), int(xs[3])
        kh, kw, sh, sw = ks[1], ks[2], st[1], st[2]
        if padding == "SAME":
            Ho, Wo = math.ceil(H / sh), math.ceil(W / sw)
            pad_h = max((Ho - 1) * sh + kh - H, 0)
            pad_w = max((Wo - 1) * sw + kw - W, 0)
            pads = (pad_h // 2, pad_h - pad_h // 2,
                    pad_w // 2, pad_w - pad_w // 2)
        else:
            Ho, Wo = (H - kh) // sh + 1, (W - kw) // sw + 1
            pads = (0, 0, 0, 0)
        y = self.new_tensor((B, Ho, Wo, C), "f32", "activation", out)
        self.ops.append(PlanOp("pool", [x], [y], {
            "mode": "max" if nd.op == "MaxPool" else "avg",
            "ksize": (kh, kw), "stride": (sh, sw), "pads": pads,
            "hw": (H, W), "out_hw": (Ho, Wo)}))

    def lower_mean(self, nd: g.NodeDef, out: str) -> None:
        x = self.tid(nd.input[0])
        axes_v = self.const_value(nd.input[1])
        if axes_v is None:
            raise PlanError("dynamic Mean axes unsupported")
        axes = sorted(int(a) for a in np.atleast_1d(axes_v))
        keep = _attr_b(nd, "keep_dims", False)
        xs = self.tensors[x].shape
        if axes == [1, 2] and len(xs) == 4:
            shape = (xs[0], 1, 1, xs[3]) if keep else (xs[0], xs[3])
            y = self.new_tensor(shape, "f32", "activation", out)
            self.ops.append(PlanOp("global_mean", [x], [y], {"keep": keep}))
            return
        if axes in ([len(xs) - 1], [-1]):
            shape = tuple(list(xs[:-1]) + ([1] if keep else []))
            y = self.new_tensor(shape, "f32", "activation", out)
            self.ops.append(PlanOp("reduce_mean_last", [x], [y],
                                   {"keep": keep}))
            return
        if axes == [1] and len(xs) == 3:
            shape = (xs[0], 1, xs[2]) if keep else (xs[0], xs[2])
            y = self.new_tensor(shape, "f32", "activation", out)
            self.ops.append(PlanOp("reduce_mean_mid", [x], [y],
                                   {"keep": keep}))
            return
        raise PlanError(f"Mean over axes {axes} unsupported")

    def lower_pad(self, nd: g.NodeDef, out: str) -> None:
        x = self.tid(nd.input[0])
        pads_v = self.const_value(nd.input[1])
        if pads_v is None:
            raise PlanError("dynamic Pad unsupported")
        if len(nd.input) > 2:           # PadV2 constant_values input
            cv = self.const_value(nd.input[2])
            if cv is None or float(np.asarray(cv).reshape(-1)[0]) != 0.0:
                raise PlanError("PadV2 with non-zero constant unsupported")
        pads = pads_v.astype(int)
        xs = list(self.tensors[x].shape)
        shape = []
        for i, d in enumerate(xs):
            lo, hi = int(pads[i][0]), int(pads[i][1])
            if is_sym(d) and (lo or hi):
                raise PlanError("cannot pad the batch dimension")
            shape.append(d if is_sym(d) else int(d) + lo + hi)
        y = self.new_tensor(tuple(shape), "f32", "activation", out)
        self.ops.append(PlanOp("pad", [x], [y],
                               {"pads": pads.tolist()}))

    def lower_transpose(self, nd: g.NodeDef, out: str) -> None:
        x = self.tid(nd.input[0])
        perm_v = self.const_value(nd.input[1])
        if perm_v is None:
            raise PlanError("dynamic Transpose unsupported")
        perm = [int(p) for p in perm_v]
        xs = self.tensors[x].shape
        shape = tuple(xs[p] for p in perm)
        y = self.new_tensor(shape, "f32", "activation", out)
        self.ops.append(PlanOp("transpose", [x], [y], {"perm": perm}))

    def lower_gather(self, nd: g.NodeDef, out: str) -> None:
        if len(nd.input) > 2:           # GatherV2 axis input
            axis_v = self.const_value(nd.input[2])
            if axis_v is None or int(np.asarray(axis_v).reshape(-1)[0]) != 0:
                raise PlanError("only axis-0 Gather supported")
        if _attr_i(nd, "batch_dims", 0) != 0:
            raise PlanError("batched Gather unsupported")
        table = self.tid(nd.input[0])
        idx = self.tid(nd.input[1])
        ts = self.tensors[table].shape
        is_ = self.tensors[idx].shape
        shape = tuple(is_) + tuple(ts[1:])
        y = self.new_tensor(shape, "f32", "activation", out)
        self.ops.append(PlanOp("gather", [table, idx], [y], {}))

    def lower_batchnorm_standalone(self, nd: g.NodeDef, out: str) -> None:
        """FusedBatchNorm whose producer isn't a conv — lower to the fused
        scale/shift (+ optional downstream relu) kernel."""
        if _attr_s(nd, "data_format", "NHWC") != "NHWC":
            raise PlanError("only NHWC FusedBatchNorm supported")
        x = self.tid(nd.input[0])
        scale, offset, mean, var = (np.asarray(self.weight_of(r),
                                               dtype=np.float32)
                                    for r in nd.input[1:5])
        eps = _attr_f(nd, "epsilon", 1e-3)
        g_ = scale / np.sqrt(var + eps)
        b_ = offset - mean * g_
        act = ACT_NONE
        final = nd
        nxt = self.sole_consumer(nd)
        if nxt is not None and nxt.op == "Relu":
            act = ACT_RELU
            self.fused.add(nxt.name)
            final = nxt
        g_id = self.new_tensor(g_.shape, "f32", "weight",
                               f"{nd.name}/bn_scale", weight=g_)
        b_id = self.new_tensor(b_.shape, "f32", "weight",
                               f"{nd.name}/bn_shift", weight=b_)
        shape = self.tensors[x].shape
        out_name = f"{final.name}:0"
        y = self.new_tensor(shape, "f32", "activation", out_name)
        self.ops.append(PlanOp("bn_act", [x, g_id, b_id], [y], {"act": act}))
        if out_name != out:
            self.by_name[out] = y

    def lower_pack(self, nd: g.NodeDef, out: str) -> None:
        """Pack (tf.stack): expand each input with a new unit axis
        (reshape alias) and concat along it — reuses the validated
        concat path."""
        axis = _attr_i(nd, "axis", 0)
        ids = [self.tid(r) for r in nd.input]
        shapes = [self.tensors[i].shape for i in ids]
        rank = len(shapes[0]) + 1
        if axis < 0:
            axis += rank
        exp_ids = []
        for i in ids:
            sh = list(self.tensors[i].shape)
            sh.insert(axis, 1)
            exp_ids.append(self.new_tensor(
                tuple(sh), "f32", "activation",
                f"{nd.name}/exp{len(exp_ids)}", alias_of=i))
        out_shape = list(shapes[0])
        out_shape.insert(axis, len(ids))
        y = self.new_tensor(tuple(out_shape), "f32", "activation", out)
        self.ops.append(PlanOp("concat", exp_ids, [y], {"axis": axis}))

    def lower_unpack(self, nd: g.NodeDef) -> None:
        """Unpack (tf.unstack): one strided-copy per output slice."""
        x = self.tid(nd.input[0])
        xs = list(self.tensors[x].shape)
        axis = _attr_i(nd, "axis", 0)
        if axis < 0:
            axis += len(xs)
        if is_sym(xs[axis]):
            raise PlanError("Unpack along the batch dim unsupported")
        num = _attr_i(nd, "num", int(xs[axis]))
        for j in range(num):
            starts = [0] * len(xs)
            steps = [1] * len(xs)
            shrink = [False] * len(xs)
            starts[axis] = j
            shrink[axis] = True
            out_dims = [d for q, d in enumerate(xs) if q != axis]
            y = self.new_tensor(tuple(out_dims), "f32", "activation",
                                f"{nd.name}:{j}")
            self.ops.append(PlanOp("strided_copy", [x], [y],
                                   {"starts": starts, "steps": steps,
                                    "shrink": shrink}))

    def lower_concat(self, nd: g.NodeDef, out: str) -> None:
        axis_v = self.const_value(nd.input[-1])
        if axis_v is None:
            raise PlanError("dynamic ConcatV2 axis unsupported")
        axis = int(np.asarray(axis_v).reshape(-1)[0])
        ids = [self.tid(r) for r in nd.input[:-1]]
        shapes = [self.tensors[i].shape for i in ids]
        if axis < 0:
            axis += len(shapes[0])
        total = sum(int(s[axis]) for s in shapes)
        shape = list(shapes[0])
        shape[axis] = total
        y = self.new_tensor(tuple(shape), "f32", "activation", out)
        self.ops.append(PlanOp("concat", ids, [y], {"axis": axis}))


def _broadcast(sa: Sequence[Dim], sb: Sequence[Dim]) -> Shape:
    la, lb = list(sa), list(sb)
    out: List[Dim] = []
    while la or lb:
        a = la.pop() if la else 1
        b_ = lb.pop() if lb else 1
        if a == b_:
            out.append(a)
        elif a == 1:
            out.append(b_)
        elif b_ == 1:
            out.append(a)
        else:
            raise PlanError(f"cannot broadcast {sa} vs {sb}")
    return tuple(reversed(out))


def fuse_attention(plan: Plan) -> Plan:
    """Post-pass: collapse the lowered multi-head-attention op chain

        transpose(q4d) , transpose(k4d) , transpose(v4d)   [perm 0,2,1,3]
        batched_gemm(qh, kh, trans_b=True) -> eltwise mul(scalar)
        -> softmax -> batched_gemm(probs, vh) -> transpose [perm 0,2,1,3]

    into one `attention` op operating on the PRE-transpose [B,S,H,D]
    views (flash-style fused kernel on GPU; no S x S materialization,
    no transposes). Fused only when head_dim == 64 (the kernel's
    constraint); otherwise left as-is."""
    ops = plan.ops
    produced_by: Dict[int, int] = {}
    for oi, op in enumerate(ops):
        for o in op.outputs:
            produced_by[o] = oi
    consumers: Dict[int, List[int]] = {}
    for oi, op in enumerate(ops):
        for i in op.inputs:
            consumers.setdefault(i, []).append(oi)

    def sole_consumer(tensor_idx):
        c = consumers.get(tensor_idx, [])
        return c[0] if len(c) == 1 else None

    def producer(tensor_idx):
        oi = produced_by.get(tensor_idx)
        return ops[oi] if oi is not None else None

    to_remove: set = set()
    replacements: Dict[int, PlanOp] = {}

    for oi, op in enumerate(ops):
        if op.kind != "batched_gemm" or not op.params.get("trans_b"):
            continue
        tq, tk = producer(op.inputs[0]), producer(op.inputs[1])
        if tq is None or tk is None or tq.kind != "transpose" or \
                tk.kind != "transpose":
            continue
        if tq.params.get("perm") != [0, 2, 1, 3] or \
                tk.params.get("perm") != [0, 2, 1, 3]:
            continue
        mul_oi = sole_consumer(op.outputs[0])
        if mul_oi is None or ops[mul_oi].kind != "eltwise" or \
                ops[mul_oi].params.get("fn") != "mul":
            continue
        mul_op = ops[mul_oi]
        scale_idx = mul_op.inputs[1] if mul_op.inputs[0] == op.outputs[0] \
            else mul_op.inputs[0]
        scale_t = plan.tensors[scale_idx]
        if scale_t.weight is None or scale_t.weight.size != 1:
            continue
        scale = float(np.asarray(scale_t.weight).reshape(-1)[0])
        sm_oi = sole_consumer(mul_op.outputs[0])
        if sm_oi is None or ops[sm_oi].kind != "softmax":
            continue
        pv_oi = sole_consumer(ops[sm_oi].outputs[0])
        if pv_oi is None or ops[pv_oi].kind != "batched_gemm" or \
                ops[pv_oi].params.get("trans_b"):
            continue
        pv = ops[pv_oi]
        tv = producer(pv.inputs[1])
        if tv is None or tv.kind != "transpose" or \
                tv.params.get("perm") != [0, 2, 1, 3]:
            continue
        tc_oi = sole_consumer(pv.outputs[0])
        if tc_oi is None or ops[tc_oi].kind != "transpose" or \
                ops[tc_oi].params.get("perm") != [0, 2, 1, 3]:
            continue
        tc = ops[tc_oi]
        q4d, k4d, v4d = tq.inputs[0], tk.inputs[0], tv.inputs[0]
        shape = plan.tensors[q4d].shape        # [B, S, H, D]
        if len(shape) != 4 or shape[3] != 64:
            continue
        for rm in (produced_by[tq.outputs[0]], produced_by[tk.outputs[0]],
                   produced_by[tv.outputs[0]], oi, mul_oi, sm_oi, pv_oi,
                   tc_oi):
            to_remove.add(rm)
        replacements[tc_oi] = PlanOp(
            "attention", [q4d, k4d, v4d], [tc.outputs[0]],
            {"scale": scale, "seq": shape[1], "heads": shape[2],
             "head_dim": shape[3]})

    if not to_remove:
        return plan
    new_ops: List[PlanOp] = []
    for oi, op in enumerate(ops):
        if oi in replacements:
            new_ops.append(replacements[oi])
        elif oi not in to_remove:
            new_ops.append(op)
    plan.ops = new_ops
    return plan


def compile_graph(graph_def: g.GraphDef,
                  signature: m.SignatureDef,
                  variables: Optional[Dict[str, np.ndarray]] = None
                  ) -> Plan:
    """variables: checkpoint tensors (engine/tensor_bundle.py) for
    non-frozen SavedModels — VariableV2/VarHandleOp nodes resolve to
    these by node name, exactly as TF Serving's restore would."""
    return fuse_attention(
        _Lowerer(graph_def, signature, variables).run())
