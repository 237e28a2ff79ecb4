"""etcd v3 KV API subset (etcdserverpb.KV + mvccpb.KeyValue), built at
runtime like the rest of oim_amd.spec.

Field numbers follow the public etcd v3 API (etcd-io/etcd
api/etcdserverpb/rpc.proto, api/mvccpb/kv.proto), so this client talks
to a real etcd cluster; tests use the in-process FakeEtcdServer below
(no etcd binary ships in this image)."""

from __future__ import annotations

import threading
from typing import Dict

import grpc

from ..spec._build import Field, Message, Service, build_file

_kv_classes = build_file(
    name="oim_amd/mvccpb.proto",
    package="mvccpb",
    messages=[
        Message(
            "KeyValue",
            [
                Field("key", 1, "bytes"),
                Field("create_revision", 2, "int64"),
                Field("mod_revision", 3, "int64"),
                Field("version", 4, "int64"),
                Field("value", 5, "bytes"),
                Field("lease", 6, "int64"),
            ],
        ),
    ],
)
KeyValue = _kv_classes["KeyValue"]

_rpc_classes = build_file(
    name="oim_amd/etcdserverpb.proto",
    package="etcdserverpb",
    dependencies=["oim_amd/mvccpb.proto"],
    messages=[
        Message(
            "ResponseHeader",
            [
                Field("cluster_id", 1, "uint64"),
                Field("member_id", 2, "uint64"),
                Field("revision", 3, "int64"),
                Field("raft_term", 4, "uint64"),
            ],
        ),
        Message(
            "RangeRequest",
            [
                Field("key", 1, "bytes"),
                Field("range_end", 2, "bytes"),
                Field("limit", 3, "int64"),
                Field("revision", 4, "int64"),
                Field("serializable", 7, "bool"),
                Field("keys_only", 8, "bool"),
                Field("count_only", 9, "bool"),
            ],
        ),
        Message(
            "RangeResponse",
            [
                Field("header", 1, "ResponseHeader"),
                Field("kvs", 2, ".mvccpb.KeyValue", repeated=True),
                Field("more", 3, "bool"),
                Field("count", 4, "int64"),
            ],
        ),
        Message(
            "PutRequest",
            [
                Field("key", 1, "bytes"),
                Field("value", 2, "bytes"),
                Field("lease", 3, "int64"),
                Field("prev_kv", 4, "bool"),
            ],
        ),
        Message(
            "PutResponse",
            [
                Field("header", 1, "ResponseHeader"),
                Field("prev_kv", 2, ".mvccpb.KeyValue"),
            ],
        ),
        Message(
            "DeleteRangeRequest",
            [
                Field("key", 1, "bytes"),
                Field("range_end", 2, "bytes"),
                Field("prev_kv", 3, "bool"),
            ],
        ),
        Message(
            "DeleteRangeResponse",
            [
                Field("header", 1, "ResponseHeader"),
                Field("deleted", 2, "int64"),
                Field("prev_kvs", 3, ".mvccpb.KeyValue", repeated=True),
            ],
        ),
    ],
    services=[
        Service(
            "KV",
            [
                ("Range", "RangeRequest", "RangeResponse"),
                ("Put", "PutRequest", "PutResponse"),
                ("DeleteRange", "DeleteRangeRequest", "DeleteRangeResponse"),
            ],
        ),
    ],
)

ResponseHeader = _rpc_classes["ResponseHeader"]
RangeRequest = _rpc_classes["RangeRequest"]
RangeResponse = _rpc_classes["RangeResponse"]
PutRequest = _rpc_classes["PutRequest"]
PutResponse = _rpc_classes["PutResponse"]
DeleteRangeRequest = _rpc_classes["DeleteRangeRequest"]
DeleteRangeResponse = _rpc_classes["DeleteRangeResponse"]

KV_SERVICE = "etcdserverpb.KV"


class KVStub:
    def __init__(self, channel: grpc.Channel):
        def method(name, reply_class):
            return channel.unary_unary(
                f"/{KV_SERVICE}/{name}",
                request_serializer=lambda m: m.SerializeToString(),
                response_deserializer=reply_class.FromString)

        self.Range = method("Range", RangeResponse)
        self.Put = method("Put", PutResponse)
        self.DeleteRange = method("DeleteRange", DeleteRangeResponse)


class FakeEtcdServicer:
    """In-memory etcd KV (tests; etcd itself is not in this image)."""

    def __init__(self):
        self._mutex = threading.Lock()
        self._data: Dict[bytes, bytes] = {}
        self._revision = 1

    def _header(self):
        return ResponseHeader(cluster_id=1, member_id=1,
                              revision=self._revision, raft_term=1)

    def Range(self, request, context):
        with self._mutex:
            response = RangeResponse(header=self._header())
            if request.range_end:
                selected = sorted(
                    (k, v) for k, v in self._data.items()
                    if request.key <= k < request.range_end)
            else:
                selected = [(request.key, self._data[request.key])] \
                    if request.key in self._data else []
            for key, value in selected:
                response.kvs.add(key=key, value=value, version=1,
                                 mod_revision=self._revision)
            response.count = len(selected)
            return response

    def Put(self, request, context):
        with self._mutex:
            self._revision += 1
            self._data[request.key] = request.value
            return PutResponse(header=self._header())

    def DeleteRange(self, request, context):
        with self._mutex:
            self._revision += 1
            deleted = 0
            if request.range_end:
                keys = [k for k in self._data
                        if request.key <= k < request.range_end]
            else:
                keys = [request.key] if request.key in self._data else []
            for key in keys:
                del self._data[key]
                deleted += 1
            return DeleteRangeResponse(header=self._header(), deleted=deleted)


def add_fake_etcd_to_server(servicer: FakeEtcdServicer, server: grpc.Server):
    def unary(fn, request_class):
        return grpc.unary_unary_rpc_method_handler(
            fn,
            request_deserializer=request_class.FromString,
            response_serializer=lambda m: m.SerializeToString())

    handlers = {
        "Range": unary(servicer.Range, RangeRequest),
        "Put": unary(servicer.Put, PutRequest),
        "DeleteRange": unary(servicer.DeleteRange, DeleteRangeRequest),
    }
    server.add_generic_rpc_handlers(
        (grpc.method_handlers_generic_handler(KV_SERVICE, handlers),))
