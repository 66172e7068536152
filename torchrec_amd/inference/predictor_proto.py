"""Wire-compatible predictor.proto messages, constructed at runtime.

The schema mirrors the reference gRPC service
(reference torchrec/inference/protos/predictor.proto: SparseFeatures,
FloatFeatures, PredictionRequest, PredictionResponse, service Predictor) so
clients built against the reference proto interoperate. Messages are built
from a FileDescriptorProto at import time — no protoc step.
"""

from __future__ import annotations

from google.protobuf import descriptor_pb2, descriptor_pool, message_factory

_POOL = descriptor_pool.DescriptorPool()


def _build_file() -> descriptor_pb2.FileDescriptorProto:
    f = descriptor_pb2.FileDescriptorProto()
    f.name = "predictor.proto"
    f.package = "predictor"
    f.syntax = "proto3"

    sparse = f.message_type.add()
    sparse.name = "SparseFeatures"
    for i, (name, typ) in enumerate(
        [("num_features", "int32"), ("lengths", "bytes"), ("values", "bytes"),
         ("weights", "bytes")],
        start=1,
    ):
        fld = sparse.field.add()
        fld.name = name
        fld.number = i
        fld.label = descriptor_pb2.FieldDescriptorProto.LABEL_OPTIONAL
        fld.type = (
            descriptor_pb2.FieldDescriptorProto.TYPE_INT32
            if typ == "int32"
            else descriptor_pb2.FieldDescriptorProto.TYPE_BYTES
        )

    floats = f.message_type.add()
    floats.name = "FloatFeatures"
    fld = floats.field.add()
    fld.name, fld.number = "num_features", 1
    fld.label = descriptor_pb2.FieldDescriptorProto.LABEL_OPTIONAL
    fld.type = descriptor_pb2.FieldDescriptorProto.TYPE_INT32
    fld = floats.field.add()
    fld.name, fld.number = "values", 2
    fld.label = descriptor_pb2.FieldDescriptorProto.LABEL_OPTIONAL
    fld.type = descriptor_pb2.FieldDescriptorProto.TYPE_BYTES

    req = f.message_type.add()
    req.name = "PredictionRequest"
    fld = req.field.add()
    fld.name, fld.number = "batch_size", 1
    fld.label = descriptor_pb2.FieldDescriptorProto.LABEL_OPTIONAL
    fld.type = descriptor_pb2.FieldDescriptorProto.TYPE_INT32
    for i, (name, tname) in enumerate(
        [("float_features", "FloatFeatures"),
         ("id_list_features", "SparseFeatures"),
         ("id_score_list_features", "SparseFeatures"),
         ("embedding_features", "FloatFeatures"),
         ("unary_features", "SparseFeatures")],
        start=2,
    ):
        fld = req.field.add()
        fld.name = name
        fld.number = i
        fld.label = descriptor_pb2.FieldDescriptorProto.LABEL_OPTIONAL
        fld.type = descriptor_pb2.FieldDescriptorProto.TYPE_MESSAGE
        fld.type_name = f".predictor.{tname}"

    vec = f.message_type.add()
    vec.name = "FloatVec"
    fld = vec.field.add()
    fld.name, fld.number = "data", 1
    fld.label = descriptor_pb2.FieldDescriptorProto.LABEL_REPEATED
    fld.type = descriptor_pb2.FieldDescriptorProto.TYPE_FLOAT

    resp = f.message_type.add()
    resp.name = "PredictionResponse"
    # map<string, FloatVec> lowers to a repeated nested MapEntry message
    entry = resp.nested_type.add()
    entry.name = "PredictionsEntry"
    entry.options.map_entry = True
    k = entry.field.add()
    k.name, k.number = "key", 1
    k.label = descriptor_pb2.FieldDescriptorProto.LABEL_OPTIONAL
    k.type = descriptor_pb2.FieldDescriptorProto.TYPE_STRING
    v = entry.field.add()
    v.name, v.number = "value", 2
    v.label = descriptor_pb2.FieldDescriptorProto.LABEL_OPTIONAL
    v.type = descriptor_pb2.FieldDescriptorProto.TYPE_MESSAGE
    v.type_name = ".predictor.FloatVec"
    fld = resp.field.add()
    fld.name, fld.number = "predictions", 1
    fld.label = descriptor_pb2.FieldDescriptorProto.LABEL_REPEATED
    fld.type = descriptor_pb2.FieldDescriptorProto.TYPE_MESSAGE
    fld.type_name = ".predictor.PredictionResponse.PredictionsEntry"

    svc = f.service.add()
    svc.name = "Predictor"
    m = svc.method.add()
    m.name = "Predict"
    m.input_type = ".predictor.PredictionRequest"
    m.output_type = ".predictor.PredictionResponse"
    return f


_fd = _POOL.Add(_build_file())


def _msg(name: str):
    return message_factory.GetMessageClass(_POOL.FindMessageTypeByName(name))


SparseFeatures = _msg("predictor.SparseFeatures")
FloatFeatures = _msg("predictor.FloatFeatures")
PredictionRequest = _msg("predictor.PredictionRequest")
FloatVec = _msg("predictor.FloatVec")
PredictionResponse = _msg("predictor.PredictionResponse")

PREDICT_METHOD = "/predictor.Predictor/Predict"
