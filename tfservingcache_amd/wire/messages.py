"""TensorFlow / TF Serving wire messages (hand-declared, wire-compatible).

Field numbers mirror the public tensorflow + tensorflow_serving protos; the
reference repo carries the same definitions as generated Go
(/root/reference/proto/tensorflow/serving/*.pb.go,
 /root/reference/proto/tensorflow/core/framework/*.pb.go) — the numbers
below were cross-checked against those files and are the wire contract
TF-Serving clients speak.
"""
from __future__ import annotations

from .pb import Message

# --------------------------------------------------------------------------
# tensorflow.DataType (core/framework/types.proto)
# --------------------------------------------------------------------------
DT_INVALID = 0
DT_FLOAT = 1
DT_DOUBLE = 2
DT_INT32 = 3
DT_UINT8 = 4
DT_INT16 = 5
DT_INT8 = 6
DT_STRING = 7
DT_COMPLEX64 = 8
DT_INT64 = 9
DT_BOOL = 10
DT_QINT8 = 11
DT_QUINT8 = 12
DT_QINT32 = 13
DT_BFLOAT16 = 14
DT_QINT16 = 15
DT_QUINT16 = 16
DT_UINT16 = 17
DT_COMPLEX128 = 18
DT_HALF = 19
DT_RESOURCE = 20
DT_VARIANT = 21
DT_UINT32 = 22
DT_UINT64 = 23

DTYPE_NAMES = {
    DT_FLOAT: "DT_FLOAT", DT_DOUBLE: "DT_DOUBLE", DT_INT32: "DT_INT32",
    DT_UINT8: "DT_UINT8", DT_INT16: "DT_INT16", DT_INT8: "DT_INT8",
    DT_STRING: "DT_STRING", DT_INT64: "DT_INT64", DT_BOOL: "DT_BOOL",
    DT_BFLOAT16: "DT_BFLOAT16", DT_UINT16: "DT_UINT16", DT_HALF: "DT_HALF",
    DT_UINT32: "DT_UINT32", DT_UINT64: "DT_UINT64",
}

# tensorflow.error.Code (subset of canonical gRPC codes)
ERROR_OK = 0
ERROR_CANCELLED = 1
ERROR_UNKNOWN = 2
ERROR_INVALID_ARGUMENT = 3
ERROR_NOT_FOUND = 5
ERROR_UNAVAILABLE = 14

# tensorflow.serving.ModelVersionStatus.State
STATE_UNKNOWN = 0
STATE_START = 10
STATE_LOADING = 20
STATE_AVAILABLE = 30
STATE_UNLOADING = 40
STATE_END = 50

STATE_NAMES = {
    STATE_UNKNOWN: "UNKNOWN", STATE_START: "START", STATE_LOADING: "LOADING",
    STATE_AVAILABLE: "AVAILABLE", STATE_UNLOADING: "UNLOADING",
    STATE_END: "END",
}


# --------------------------------------------------------------------------
# google.protobuf well-known wrappers
# --------------------------------------------------------------------------
class Int64Value(Message):
    FIELDS = [("value", 1, "int64")]


class Any(Message):
    FIELDS = [("type_url", 1, "string"), ("value", 2, "bytes")]


# --------------------------------------------------------------------------
# core/framework: tensor shape + tensor
# --------------------------------------------------------------------------
class TensorShapeDim(Message):
    FIELDS = [("size", 1, "int64"), ("name", 2, "string")]


class TensorShapeProto(Message):
    FIELDS = [
        ("dim", 2, "message", dict(msg_cls=TensorShapeDim, repeated=True)),
        ("unknown_rank", 3, "bool"),
    ]

    @classmethod
    def of(cls, dims):
        return cls(dim=[TensorShapeDim(size=int(d)) for d in dims])

    def sizes(self):
        return [d.size for d in self.dim]


class TensorProto(Message):
    FIELDS = [
        ("dtype", 1, "enum"),
        ("tensor_shape", 2, "message", dict(msg_cls=TensorShapeProto)),
        ("version_number", 3, "int32"),
        ("tensor_content", 4, "bytes"),
        ("float_val", 5, "float", dict(repeated=True, packed=True)),
        ("double_val", 6, "double", dict(repeated=True, packed=True)),
        ("int_val", 7, "int32", dict(repeated=True, packed=True)),
        ("string_val", 8, "bytes", dict(repeated=True)),
        ("scomplex_val", 9, "float", dict(repeated=True, packed=True)),
        ("int64_val", 10, "int64", dict(repeated=True, packed=True)),
        ("bool_val", 11, "bool", dict(repeated=True, packed=True)),
        ("dcomplex_val", 12, "double", dict(repeated=True, packed=True)),
        ("half_val", 13, "int32", dict(repeated=True, packed=True)),
        ("uint32_val", 16, "uint32", dict(repeated=True, packed=True)),
        ("uint64_val", 17, "uint64", dict(repeated=True, packed=True)),
    ]


# --------------------------------------------------------------------------
# core/protobuf: named tensors (SessionService)
# --------------------------------------------------------------------------
class NamedTensorProto(Message):
    FIELDS = [
        ("name", 1, "string"),
        ("tensor", 2, "message", dict(msg_cls=TensorProto)),
    ]


# --------------------------------------------------------------------------
# tensorflow.serving: model spec + predict
# --------------------------------------------------------------------------
class ModelSpec(Message):
    FIELDS = [
        ("name", 1, "string"),
        ("version", 2, "message", dict(msg_cls=Int64Value)),
        ("signature_name", 3, "string"),
        ("version_label", 4, "string"),
    ]

    def version_value(self):
        return self.version.value if self.version is not None else 0


class PredictRequest(Message):
    FIELDS = [
        ("model_spec", 1, "message", dict(msg_cls=ModelSpec)),
        ("inputs", 2, "map", dict(msg_cls=TensorProto, map_value="message")),
        ("output_filter", 3, "string", dict(repeated=True)),
    ]


class PredictResponse(Message):
    FIELDS = [
        ("outputs", 1, "map", dict(msg_cls=TensorProto, map_value="message")),
        ("model_spec", 2, "message", dict(msg_cls=ModelSpec)),
    ]


# --------------------------------------------------------------------------
# tensorflow.serving: tf.Example-based Classify / Regress
# (core/example/example.proto + feature.proto)
# --------------------------------------------------------------------------
class BytesList(Message):
    FIELDS = [("value", 1, "bytes", dict(repeated=True))]


class FloatList(Message):
    FIELDS = [("value", 1, "float", dict(repeated=True, packed=True))]


class Int64List(Message):
    FIELDS = [("value", 1, "int64", dict(repeated=True, packed=True))]


class Feature(Message):
    FIELDS = [
        ("bytes_list", 1, "message", dict(msg_cls=BytesList)),
        ("float_list", 2, "message", dict(msg_cls=FloatList)),
        ("int64_list", 3, "message", dict(msg_cls=Int64List)),
    ]


class Features(Message):
    FIELDS = [("feature", 1, "map", dict(msg_cls=Feature, map_value="message"))]


class Example(Message):
    FIELDS = [("features", 1, "message", dict(msg_cls=Features))]


class ExampleList(Message):
    FIELDS = [("examples", 1, "message", dict(msg_cls=Example, repeated=True))]


class ExampleListWithContext(Message):
    FIELDS = [
        ("examples", 1, "message", dict(msg_cls=Example, repeated=True)),
        ("context", 2, "message", dict(msg_cls=Example)),
    ]


class Input(Message):
    FIELDS = [
        ("example_list", 1, "message", dict(msg_cls=ExampleList)),
        ("example_list_with_context", 2, "message",
         dict(msg_cls=ExampleListWithContext)),
    ]


class Class(Message):
    FIELDS = [("label", 1, "string"), ("score", 2, "float")]


class Classifications(Message):
    FIELDS = [("classes", 1, "message", dict(msg_cls=Class, repeated=True))]


class ClassificationResult(Message):
    FIELDS = [("classifications", 1, "message",
               dict(msg_cls=Classifications, repeated=True))]


class ClassificationRequest(Message):
    FIELDS = [
        ("model_spec", 1, "message", dict(msg_cls=ModelSpec)),
        ("input", 2, "message", dict(msg_cls=Input)),
    ]


class ClassificationResponse(Message):
    FIELDS = [
        ("result", 1, "message", dict(msg_cls=ClassificationResult)),
        ("model_spec", 2, "message", dict(msg_cls=ModelSpec)),
    ]


class Regression(Message):
    FIELDS = [("value", 1, "float")]


class RegressionResult(Message):
    FIELDS = [("regressions", 1, "message",
               dict(msg_cls=Regression, repeated=True))]


class RegressionRequest(Message):
    FIELDS = [
        ("model_spec", 1, "message", dict(msg_cls=ModelSpec)),
        ("input", 2, "message", dict(msg_cls=Input)),
    ]


class RegressionResponse(Message):
    FIELDS = [
        ("result", 1, "message", dict(msg_cls=RegressionResult)),
        ("model_spec", 2, "message", dict(msg_cls=ModelSpec)),
    ]


# --------------------------------------------------------------------------
# tensorflow.serving: model metadata (SignatureDef map)
# --------------------------------------------------------------------------
class TensorInfo(Message):
    FIELDS = [
        ("name", 1, "string"),
        ("dtype", 2, "enum"),
        ("tensor_shape", 3, "message", dict(msg_cls=TensorShapeProto)),
    ]


class SignatureDef(Message):
    FIELDS = [
        ("inputs", 1, "map", dict(msg_cls=TensorInfo, map_value="message")),
        ("outputs", 2, "map", dict(msg_cls=TensorInfo, map_value="message")),
        ("method_name", 3, "string"),
    ]


class SignatureDefMap(Message):
    FIELDS = [("signature_def", 1, "map",
               dict(msg_cls=SignatureDef, map_value="message"))]


class GetModelMetadataRequest(Message):
    FIELDS = [
        ("model_spec", 1, "message", dict(msg_cls=ModelSpec)),
        ("metadata_field", 2, "string", dict(repeated=True)),
    ]


class GetModelMetadataResponse(Message):
    FIELDS = [
        ("model_spec", 1, "message", dict(msg_cls=ModelSpec)),
        ("metadata", 2, "map", dict(msg_cls=Any, map_value="message")),
    ]


# --------------------------------------------------------------------------
# tensorflow.serving: model status + reload config (ModelService)
# --------------------------------------------------------------------------
class StatusProto(Message):
    FIELDS = [("error_code", 1, "enum"), ("error_message", 2, "string")]


class GetModelStatusRequest(Message):
    FIELDS = [("model_spec", 1, "message", dict(msg_cls=ModelSpec))]


class ModelVersionStatus(Message):
    FIELDS = [
        ("version", 1, "int64"),
        ("state", 2, "enum"),
        ("status", 3, "message", dict(msg_cls=StatusProto)),
    ]


class GetModelStatusResponse(Message):
    FIELDS = [("model_version_status", 1, "message",
               dict(msg_cls=ModelVersionStatus, repeated=True))]


class ServableVersionPolicyLatest(Message):
    FIELDS = [("num_versions", 1, "uint32")]


class ServableVersionPolicyAll(Message):
    FIELDS = []


class ServableVersionPolicySpecific(Message):
    FIELDS = [("versions", 1, "int64", dict(repeated=True, packed=True))]


class ServableVersionPolicy(Message):
    # oneof policy_choice (file_system_storage_path_source.proto)
    FIELDS = [
        ("latest", 100, "message", dict(msg_cls=ServableVersionPolicyLatest)),
        ("all", 101, "message", dict(msg_cls=ServableVersionPolicyAll)),
        ("specific", 102, "message",
         dict(msg_cls=ServableVersionPolicySpecific)),
    ]


class ModelConfig(Message):
    FIELDS = [
        ("name", 1, "string"),
        ("base_path", 2, "string"),
        ("model_platform", 4, "string"),
        ("model_version_policy", 7, "message",
         dict(msg_cls=ServableVersionPolicy)),
        ("version_labels", 8, "map", dict(map_value="int64")),
    ]


class ModelConfigList(Message):
    FIELDS = [("config", 1, "message", dict(msg_cls=ModelConfig, repeated=True))]


class ModelServerConfig(Message):
    FIELDS = [
        ("model_config_list", 1, "message", dict(msg_cls=ModelConfigList)),
        ("custom_model_config", 2, "message", dict(msg_cls=Any)),
    ]


class ReloadConfigRequest(Message):
    FIELDS = [("config", 1, "message", dict(msg_cls=ModelServerConfig))]


class ReloadConfigResponse(Message):
    FIELDS = [("status", 1, "message", dict(msg_cls=StatusProto))]


# --------------------------------------------------------------------------
# tensorflow.serving: SessionService (SessionRun)
# --------------------------------------------------------------------------
class SessionRunRequest(Message):
    FIELDS = [
        ("model_spec", 1, "message", dict(msg_cls=ModelSpec)),
        ("feed", 2, "message", dict(msg_cls=NamedTensorProto, repeated=True)),
        ("fetch", 3, "string", dict(repeated=True)),
        ("target", 4, "string", dict(repeated=True)),
    ]


class SessionRunResponse(Message):
    FIELDS = [
        ("tensor", 1, "message", dict(msg_cls=NamedTensorProto, repeated=True)),
        ("model_spec", 3, "message", dict(msg_cls=ModelSpec)),
    ]


def peek_model_spec(data: bytes) -> ModelSpec:
    """Fast partial parse: extract only the model_spec (field 1) of a
    PredictRequest without touching the tensor payloads — the routing
    hot path for the C++ fast predict."""
    from .pb import read_varint, skip_field
    pos, end = 0, len(data)
    while pos < end:
        tag, pos = read_varint(data, pos)
        if (tag >> 3) == 1 and (tag & 7) == 2:
            n, pos = read_varint(data, pos)
            return ModelSpec.decode(data, pos, pos + n)
        pos = skip_field(data, pos, tag & 7)
    return ModelSpec()


# --------------------------------------------------------------------------
# gRPC service/method names (the wire-level routing contract)
# --------------------------------------------------------------------------
PREDICTION_SERVICE = "tensorflow.serving.PredictionService"
MODEL_SERVICE = "tensorflow.serving.ModelService"
SESSION_SERVICE = "tensorflow.serving.SessionService"

PREDICTION_METHODS = {
    "Classify": (ClassificationRequest, ClassificationResponse),
    "Regress": (RegressionRequest, RegressionResponse),
    "Predict": (PredictRequest, PredictResponse),
    "GetModelMetadata": (GetModelMetadataRequest, GetModelMetadataResponse),
}

MODEL_METHODS = {
    "GetModelStatus": (GetModelStatusRequest, GetModelStatusResponse),
    "HandleReloadConfigRequest": (ReloadConfigRequest, ReloadConfigResponse),
}

SESSION_METHODS = {
    "SessionRun": (SessionRunRequest, SessionRunResponse),
}
