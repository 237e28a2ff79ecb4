"""etcd-backed RegistryDB (the reference's intended production backend,
README.md:132-139 — never implemented there; built fresh here behind
the 3-method RegistryDB interface, registry.go:31-41).

Keys live under an ``/oim/`` prefix so several registry replicas can
share one etcd cluster: any replica sees every controller's
``<id>/address``/``<id>/pci`` entries, which is what makes the 8
controllers of an MI355X node reachable through any registry instance
(BASELINE config 5)."""

from __future__ import annotations

import threading
from typing import List, Optional, Sequence, Tuple

import grpc

from ..common import join_registry_path, split_registry_path
from .db import RegistryDB, _has_prefix
from . import etcdpb

PREFIX = b"/oim/"


def _range_end(prefix: bytes) -> bytes:
    """The etcd convention for a prefix scan: prefix with its last byte
    incremented (etcd clientv3 GetPrefixRangeEnd)."""
    end = bytearray(prefix)
    for i in reversed(range(len(end))):
        if end[i] < 0xFF:
            end[i] += 1
            return bytes(end[: i + 1])
    return b"\0"


class EtcdRegistryDB(RegistryDB):
    def __init__(self, endpoints: Sequence[str], timeout: float = 10.0,
                 credentials: Optional[grpc.ChannelCredentials] = None):
        if not endpoints:
            raise ValueError("need at least one etcd endpoint")
        self.endpoints = list(endpoints)
        self.timeout = timeout
        self.credentials = credentials
        self._mutex = threading.Lock()
        self._channel: Optional[grpc.Channel] = None
        self._stub: Optional[etcdpb.KVStub] = None
        self._endpoint_index = 0

    def _kv(self) -> etcdpb.KVStub:
        with self._mutex:
            if self._stub is None:
                target = self.endpoints[self._endpoint_index % len(self.endpoints)]
                if self.credentials is not None:
                    self._channel = grpc.secure_channel(target, self.credentials)
                else:
                    self._channel = grpc.insecure_channel(target)
                self._stub = etcdpb.KVStub(self._channel)
            return self._stub

    def _failover(self):
        with self._mutex:
            if self._channel is not None:
                self._channel.close()
            self._channel = None
            self._stub = None
            self._endpoint_index += 1

    def _call(self, fn):
        try:
            return fn(self._kv())
        except grpc.RpcError:
            # one failover attempt to the next endpoint
            self._failover()
            return fn(self._kv())

    def store(self, elements: Sequence[str], value: str) -> None:
        key = PREFIX + join_registry_path(list(elements)).encode()
        if value == "":
            self._call(lambda kv: kv.DeleteRange(
                etcdpb.DeleteRangeRequest(key=key), timeout=self.timeout))
        else:
            self._call(lambda kv: kv.Put(
                etcdpb.PutRequest(key=key, value=value.encode()),
                timeout=self.timeout))

    def lookup(self, elements: Sequence[str]) -> Optional[str]:
        key = PREFIX + join_registry_path(list(elements)).encode()
        response = self._call(lambda kv: kv.Range(
            etcdpb.RangeRequest(key=key), timeout=self.timeout))
        if not response.kvs:
            return None
        return response.kvs[0].value.decode()

    def list(self, prefix: Sequence[str]) -> List[Tuple[List[str], str]]:
        response = self._call(lambda kv: kv.Range(
            etcdpb.RangeRequest(key=PREFIX, range_end=_range_end(PREFIX)),
            timeout=self.timeout))
        out = []
        for kv in response.kvs:
            path = kv.key[len(PREFIX):].decode()
            try:
                elements = split_registry_path(path)
            except ValueError:
                continue
            if _has_prefix(elements, prefix):
                out.append((elements, kv.value.decode()))
        out.sort()
        return out

    def close(self) -> None:
        with self._mutex:
            if self._channel is not None:
                self._channel.close()
                self._channel = None
                self._stub = None
