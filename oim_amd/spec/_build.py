"""Runtime protobuf descriptor construction.

This image has no protoc / grpcio-tools, so instead of generated
``*_pb2.py`` modules we build ``FileDescriptorProto`` objects
programmatically and materialize message classes through
``google.protobuf.message_factory``.  The wire format depends only on
field numbers and types, so messages built this way are byte-compatible
with the reference's generated Go bindings (pkg/spec/oim/v0/oim.pb.go).

The tiny DSL below keeps the schema declarations readable; see
``oim_v0.py`` for the actual oim.v0 schema (authored from docs/spec.md).
"""

from __future__ import annotations

from typing import Dict, Optional, Sequence, Tuple

from google.protobuf import descriptor_pb2, descriptor_pool, message_factory

F = descriptor_pb2.FieldDescriptorProto

_TYPES = {
    "string": F.TYPE_STRING,
    "bytes": F.TYPE_BYTES,
    "bool": F.TYPE_BOOL,
    "int32": F.TYPE_INT32,
    "int64": F.TYPE_INT64,
    "uint32": F.TYPE_UINT32,
    "uint64": F.TYPE_UINT64,
    "double": F.TYPE_DOUBLE,
    "float": F.TYPE_FLOAT,
}


class Field:
    def __init__(
        self,
        name: str,
        number: int,
        type_: str,
        repeated: bool = False,
        oneof: Optional[str] = None,
    ):
        self.name = name
        self.number = number
        self.type = type_
        self.repeated = repeated
        self.oneof = oneof


class Message:
    def __init__(self, name: str, fields: Sequence[Field] = (), map_fields: Sequence[Tuple[str, int, str, str]] = ()):
        self.name = name
        self.fields = list(fields)
        # (name, number, key_type, value_type) triples for map<k,v> fields.
        self.map_fields = list(map_fields)


class Service:
    def __init__(self, name: str, methods: Sequence[Tuple[str, str, str]]):
        # methods: (method_name, request_message, response_message)
        self.name = name
        self.methods = list(methods)


def _add_field(msg_proto, field: Field, package: str, oneof_index: Optional[int]):
    fp = msg_proto.field.add()
    fp.name = field.name
    fp.number = field.number
    fp.label = F.LABEL_REPEATED if field.repeated else F.LABEL_OPTIONAL
    if field.type in _TYPES:
        fp.type = _TYPES[field.type]
    else:
        fp.type = F.TYPE_MESSAGE
        type_name = field.type
        if not type_name.startswith("."):
            type_name = f".{package}.{type_name}"
        fp.type_name = type_name
    if oneof_index is not None:
        fp.oneof_index = oneof_index


def build_file(
    *,
    name: str,
    package: str,
    messages: Sequence[Message],
    services: Sequence[Service] = (),
    dependencies: Sequence[str] = (),
    pool: Optional[descriptor_pool.DescriptorPool] = None,
) -> Dict[str, type]:
    """Register the schema and return {message_name: message_class}.

    `dependencies` lists imported .proto files already present in the
    pool (e.g. "google/protobuf/wrappers.proto"); fields may then use
    fully-qualified external type names like ".google.protobuf.BoolValue".
    """
    pool = pool or descriptor_pool.Default()
    file_proto = descriptor_pb2.FileDescriptorProto()
    file_proto.name = name
    file_proto.package = package
    file_proto.syntax = "proto3"
    for dep in dependencies:
        file_proto.dependency.append(dep)

    for message in messages:
        msg_proto = file_proto.message_type.add()
        msg_proto.name = message.name
        oneof_indices: Dict[str, int] = {}
        for field in message.fields:
            oneof_index = None
            if field.oneof is not None:
                if field.oneof not in oneof_indices:
                    oneof_indices[field.oneof] = len(msg_proto.oneof_decl)
                    msg_proto.oneof_decl.add().name = field.oneof
                oneof_index = oneof_indices[field.oneof]
            _add_field(msg_proto, field, package, oneof_index)
        for map_name, number, key_type, value_type in message.map_fields:
            # map<k,v> lowers to a repeated nested MapEntry message.
            entry = msg_proto.nested_type.add()
            entry.name = "".join(p.capitalize() for p in map_name.split("_")) + "Entry"
            entry.options.map_entry = True
            _add_field(entry, Field("key", 1, key_type), package, None)
            _add_field(entry, Field("value", 2, value_type), package, None)
            fp = msg_proto.field.add()
            fp.name = map_name
            fp.number = number
            fp.label = F.LABEL_REPEATED
            fp.type = F.TYPE_MESSAGE
            fp.type_name = f".{package}.{message.name}.{entry.name}"

    for service in services:
        svc_proto = file_proto.service.add()
        svc_proto.name = service.name
        for method_name, request, response in service.methods:
            method = svc_proto.method.add()
            method.name = method_name
            method.input_type = f".{package}.{request}"
            method.output_type = f".{package}.{response}"

    file_desc = pool.Add(file_proto)
    classes: Dict[str, type] = {}
    for message in messages:
        desc = pool.FindMessageTypeByName(f"{package}.{message.name}")
        classes[message.name] = message_factory.GetMessageClass(desc)
    return classes
