"""SavedModel / GraphDef wire messages (tensorflow core protos).

Field numbers follow the public tensorflow protos
(core/framework/{graph,node_def,attr_value,op_def,function,versions}.proto,
core/protobuf/{saved_model,meta_graph,saver}.proto), cross-checked against
the reference's generated Go under
/root/reference/proto/tensorflow/core/{framework,protobuf}/.

We execute "frozen" SavedModels: weights live in Const nodes of the
GraphDef (the standard freeze_graph output form). The variables/ bundle
format is not parsed in this round.
"""
from __future__ import annotations

from .pb import Message
from .messages import (SignatureDef, TensorProto, TensorShapeProto)


class AttrListValue(Message):
    # attr_value.proto AttrValue.ListValue
    FIELDS = [
        ("s", 2, "bytes", dict(repeated=True)),
        ("i", 3, "int64", dict(repeated=True, packed=True)),
        ("f", 4, "float", dict(repeated=True, packed=True)),
        ("b", 5, "bool", dict(repeated=True, packed=True)),
        ("type", 6, "enum", dict(repeated=True, packed=True)),
        ("shape", 7, "message", dict(msg_cls=TensorShapeProto, repeated=True)),
        ("tensor", 8, "message", dict(msg_cls=TensorProto, repeated=True)),
    ]


class AttrValue(Message):
    FIELDS = [
        ("list", 1, "message", dict(msg_cls=AttrListValue)),
        ("s", 2, "bytes"),
        ("i", 3, "int64"),
        ("f", 4, "float"),
        ("b", 5, "bool"),
        ("type", 6, "enum"),
        ("shape", 7, "message", dict(msg_cls=TensorShapeProto)),
        ("tensor", 8, "message", dict(msg_cls=TensorProto)),
        ("placeholder", 9, "string"),
    ]


class NodeDef(Message):
    FIELDS = [
        ("name", 1, "string"),
        ("op", 2, "string"),
        ("input", 3, "string", dict(repeated=True)),
        ("device", 4, "string"),
        ("attr", 5, "map", dict(msg_cls=AttrValue, map_value="message")),
    ]


class VersionDef(Message):
    FIELDS = [
        ("producer", 1, "int32"),
        ("min_consumer", 2, "int32"),
    ]


class GraphDef(Message):
    FIELDS = [
        ("node", 1, "message", dict(msg_cls=NodeDef, repeated=True)),
        ("versions", 4, "message", dict(msg_cls=VersionDef)),
    ]


class MetaInfoDef(Message):
    # meta_graph.proto MetaGraphDef.MetaInfoDef (subset)
    FIELDS = [
        ("meta_graph_version", 1, "string"),
        ("tags", 4, "string", dict(repeated=True)),
        ("tensorflow_version", 5, "string"),
    ]


class MetaGraphDef(Message):
    FIELDS = [
        ("meta_info_def", 1, "message", dict(msg_cls=MetaInfoDef)),
        ("graph_def", 2, "message", dict(msg_cls=GraphDef)),
        ("signature_def", 5, "map",
         dict(msg_cls=SignatureDef, map_value="message")),
    ]


class SavedModel(Message):
    FIELDS = [
        ("saved_model_schema_version", 1, "int64"),
        ("meta_graphs", 2, "message", dict(msg_cls=MetaGraphDef, repeated=True)),
    ]


SERVE_TAG = "serve"
DEFAULT_SERVING_SIGNATURE = "serving_default"
PREDICT_METHOD_NAME = "tensorflow/serving/predict"
CLASSIFY_METHOD_NAME = "tensorflow/serving/classify"
REGRESS_METHOD_NAME = "tensorflow/serving/regress"
