"""SavedModel reading/writing (frozen-graph form).

The serving engine consumes standard `saved_model.pb` files whose weights
are Const nodes (freeze_graph output). A writer is included because this
environment has no TensorFlow to generate fixtures: the model builders in
tfservingcache_amd/models/ emit GraphDefs through GraphBuilder and
write_saved_model, producing byte-valid SavedModel protos that any TF
tooling could read.

Replaces the role TF Serving's SavedModel loader plays behind the
reference (the reference delegates all of this to tensorflow_model_server;
SURVEY.md §2.4).
"""
from __future__ import annotations

import os
from typing import Dict, List, Optional, Sequence, Tuple

import numpy as np

from ..wire import graph as g
from ..wire import messages as m
from ..wire.tensor import NP_TO_DTYPE, numpy_to_tensorproto

SAVED_MODEL_FILENAME = "saved_model.pb"


class GraphBuilder:
    """Builds a frozen GraphDef + serving signature."""

    def __init__(self):
        self.nodes: List[g.NodeDef] = []
        self._names = set()
        self.inputs: Dict[str, Tuple[str, int, Sequence[int]]] = {}
        self.outputs: Dict[str, str] = {}

    def _unique(self, name: str) -> str:
        if name not in self._names:
            self._names.add(name)
            return name
        i = 1
        while f"{name}_{i}" in self._names:
            i += 1
        self._names.add(f"{name}_{i}")
        return f"{name}_{i}"

    def node(self, op: str, name: str, inputs: Sequence[str] = (),
             **attrs) -> str:
        name = self._unique(name)
        nd = g.NodeDef(name=name, op=op, input=list(inputs))
        for k, v in attrs.items():
            nd.attr[k] = v
        self.nodes.append(nd)
        return name

    # -- attr helpers ------------------------------------------------------
    @staticmethod
    def a_type(dtype: int) -> g.AttrValue:
        return g.AttrValue(type=dtype)

    @staticmethod
    def a_int(i: int) -> g.AttrValue:
        return g.AttrValue(i=i)

    @staticmethod
    def a_float(f: float) -> g.AttrValue:
        return g.AttrValue(f=f)

    @staticmethod
    def a_bool(b: bool) -> g.AttrValue:
        return g.AttrValue(b=b)

    @staticmethod
    def a_str(s: str) -> g.AttrValue:
        return g.AttrValue(s=s.encode())

    @staticmethod
    def a_shape(dims: Sequence[int]) -> g.AttrValue:
        return g.AttrValue(shape=m.TensorShapeProto.of(dims))

    @staticmethod
    def a_ints(vals: Sequence[int]) -> g.AttrValue:
        return g.AttrValue(list=g.AttrListValue(i=list(vals)))

    # -- common nodes ------------------------------------------------------
    def placeholder(self, name: str, dtype_np, shape: Sequence[int],
                    signature_name: Optional[str] = None) -> str:
        dt = NP_TO_DTYPE[np.dtype(dtype_np)]
        n = self.node("Placeholder", name, dtype=self.a_type(dt),
                      shape=self.a_shape(shape))
        self.inputs[signature_name or name] = (n, dt, list(shape))
        return n

    def const(self, name: str, value: np.ndarray) -> str:
        value = np.asarray(value)
        dt = NP_TO_DTYPE[value.dtype]
        tp = numpy_to_tensorproto(value)
        return self.node("Const", name, dtype=self.a_type(dt),
                         value=g.AttrValue(tensor=tp))

    def mark_output(self, signature_name: str, tensor_name: str) -> None:
        self.outputs[signature_name] = tensor_name

    # -- assembly ----------------------------------------------------------
    def build(self, method_name: str = g.PREDICT_METHOD_NAME) -> g.SavedModel:
        sig = m.SignatureDef(method_name=method_name)
        for sig_name, (node_name, dt, shape) in self.inputs.items():
            sig.inputs[sig_name] = m.TensorInfo(
                name=node_name + ":0", dtype=dt,
                tensor_shape=m.TensorShapeProto.of(shape))
        for sig_name, tensor_name in self.outputs.items():
            if ":" not in tensor_name:
                tensor_name += ":0"
            sig.outputs[sig_name] = m.TensorInfo(name=tensor_name)
        meta = g.MetaGraphDef(
            meta_info_def=g.MetaInfoDef(tags=[g.SERVE_TAG],
                                        tensorflow_version="2.15.0-tfsc-amd"),
            graph_def=g.GraphDef(node=self.nodes,
                                 versions=g.VersionDef(producer=1987)),
        )
        meta.signature_def[g.DEFAULT_SERVING_SIGNATURE] = sig
        return g.SavedModel(saved_model_schema_version=1, meta_graphs=[meta])


def write_saved_model(saved_model: g.SavedModel, version_dir: str) -> None:
    os.makedirs(version_dir, exist_ok=True)
    os.makedirs(os.path.join(version_dir, "variables"), exist_ok=True)
    os.makedirs(os.path.join(version_dir, "assets"), exist_ok=True)
    with open(os.path.join(version_dir, SAVED_MODEL_FILENAME), "wb") as f:
        f.write(saved_model.encode())


class SavedModelError(Exception):
    pass


def read_saved_model(version_dir: str) -> Tuple[g.GraphDef, Dict[str, m.SignatureDef]]:
    """Returns (graph_def, signature_def map) of the `serve` meta graph."""
    path = os.path.join(version_dir, SAVED_MODEL_FILENAME)
    if not os.path.exists(path):
        raise SavedModelError(f"no {SAVED_MODEL_FILENAME} in {version_dir}")
    with open(path, "rb") as f:
        sm = g.SavedModel.decode(f.read())
    if not sm.meta_graphs:
        raise SavedModelError("SavedModel has no meta graphs")
    chosen = None
    for mg in sm.meta_graphs:
        tags = mg.meta_info_def.tags if mg.meta_info_def else []
        if g.SERVE_TAG in tags:
            chosen = mg
            break
    if chosen is None:
        chosen = sm.meta_graphs[0]
    if chosen.graph_def is None:
        raise SavedModelError("meta graph has no graph_def")
    return chosen.graph_def, dict(chosen.signature_def)
