"""etcd v3 discovery backend (gRPC API, hand-rolled stubs).

Behavior mirrors the reference's etcd integration
(pkg/taskhandler/discovery/etcd/etcd.go):
  * key `/service/<name>/<uuid>` holding "host:rest:grpc[:slot]"
    (etcd.go:53, cluster.go:142-144);
  * every TTL/2: grant a fresh lease of TTL seconds and Put the key
    with it, so the key expires if the node dies (etcd.go:134-148);
  * membership from the key prefix, WATCHED (etcdserverpb.Watch bidi
    stream over the prefix, like the reference's clientv3 watch,
    etcd.go:150-170): changes push immediately; falls back to TTL/2
    polling while the watch stream is unavailable;
  * optional username/password auth (etcd.go:29-47).

No protoc in this environment, so the etcdserverpb messages used
(LeaseGrant / Put / Range / Watch / Authenticate) are declared on the
wire codec.
"""
from __future__ import annotations

import logging
import threading
import uuid
from typing import List, Optional

import grpc

from ...wire.pb import Message
from .base import DiscoveryService, ServingService

log = logging.getLogger("tfsc.discovery.etcd")


# -- etcdserverpb subset ----------------------------------------------------
class LeaseGrantRequest(Message):
    FIELDS = [("ttl", 1, "int64"), ("id", 2, "int64")]


class LeaseGrantResponse(Message):
    FIELDS = [("id", 2, "int64"), ("ttl", 3, "int64")]


class PutRequest(Message):
    FIELDS = [("key", 1, "bytes"), ("value", 2, "bytes"),
              ("lease", 3, "int64")]


class PutResponse(Message):
    FIELDS = []


class RangeRequest(Message):
    FIELDS = [("key", 1, "bytes"), ("range_end", 2, "bytes")]


class KeyValue(Message):
    FIELDS = [("key", 1, "bytes"), ("create_revision", 2, "int64"),
              ("mod_revision", 3, "int64"), ("version", 4, "int64"),
              ("value", 5, "bytes"), ("lease", 6, "int64")]


class RangeResponse(Message):
    FIELDS = [("kvs", 2, "message", dict(msg_cls=KeyValue, repeated=True)),
              ("more", 3, "bool"), ("count", 4, "int64")]


class WatchCreateRequest(Message):
    FIELDS = [("key", 1, "bytes"), ("range_end", 2, "bytes"),
              ("start_revision", 3, "int64")]


class WatchRequest(Message):
    FIELDS = [("create_request", 1, "message",
               dict(msg_cls=WatchCreateRequest))]


class Event(Message):
    FIELDS = [("type", 1, "int64"),
              ("kv", 2, "message", dict(msg_cls=KeyValue))]


class WatchResponse(Message):
    FIELDS = [("watch_id", 2, "int64"), ("created", 3, "bool"),
              ("canceled", 4, "bool"),
              ("events", 11, "message", dict(msg_cls=Event,
                                             repeated=True))]


class AuthenticateRequest(Message):
    FIELDS = [("name", 1, "string"), ("password", 2, "string")]


class AuthenticateResponse(Message):
    FIELDS = [("token", 2, "string")]


def _prefix_range_end(key: bytes) -> bytes:
    end = bytearray(key)
    for i in range(len(end) - 1, -1, -1):
        if end[i] < 0xFF:
            end[i] += 1
            return bytes(end[:i + 1])
    return b"\x00"


class EtcdDiscovery(DiscoveryService):
    def __init__(self, service_name: str, endpoints: List[str],
                 heartbeat_ttl: float = 5.0,
                 username: str = "", password: str = "",
                 allow_localhost: bool = True):
        super().__init__()
        self.name = service_name
        self.endpoints = endpoints or ["127.0.0.1:2379"]
        self.ttl = max(int(heartbeat_ttl), 1)
        self.username = username
        self.password = password
        self.allow_localhost = allow_localhost
        self.service_id = uuid.uuid4().hex
        self._stop = threading.Event()
        self._threads: List[threading.Thread] = []
        self._channel: Optional[grpc.Channel] = None
        self._token: Optional[str] = None
        self._value = b""

    # -- gRPC plumbing -----------------------------------------------------
    def _chan(self) -> grpc.Channel:
        if self._channel is None:
            self._channel = grpc.insecure_channel(self.endpoints[0])
        return self._channel

    def _call(self, method: str, req: Message, resp_cls):
        fn = self._chan().unary_unary(
            method, request_serializer=lambda r: r.encode(),
            response_deserializer=resp_cls.decode)
        metadata = [("token", self._token)] if self._token else None
        return fn(req, timeout=10, metadata=metadata)

    def _auth(self) -> None:
        if self.username:
            resp = self._call("/etcdserverpb.Auth/Authenticate",
                              AuthenticateRequest(name=self.username,
                                                  password=self.password),
                              AuthenticateResponse)
            self._token = resp.token

    # -- DiscoveryService --------------------------------------------------
    def _key(self) -> bytes:
        return f"/service/{self.name}/{self.service_id}".encode()

    def register(self, service: ServingService) -> None:
        self._value = service.serialize().encode()
        self._auth()
        self._heartbeat_once()
        t1 = threading.Thread(target=self._ttl_loop, daemon=True)
        t2 = threading.Thread(target=self._poll_loop, daemon=True)
        self._threads = [t1, t2]
        t1.start()
        t2.start()

    def unregister(self) -> None:
        self._stop.set()
        for t in self._threads:
            t.join(timeout=2)
        if self._channel is not None:
            self._channel.close()

    # -- internals ---------------------------------------------------------
    def _heartbeat_once(self) -> None:
        grant = self._call("/etcdserverpb.Lease/LeaseGrant",
                           LeaseGrantRequest(ttl=self.ttl),
                           LeaseGrantResponse)
        self._call("/etcdserverpb.KV/Put",
                   PutRequest(key=self._key(), value=self._value,
                              lease=grant.id), PutResponse)

    def _ttl_loop(self) -> None:
        while not self._stop.wait(self.ttl / 2):
            try:
                self._heartbeat_once()
            except grpc.RpcError:
                log.warning("etcd lease heartbeat failed", exc_info=True)

    def fetch_members(self) -> List[ServingService]:
        prefix = f"/service/{self.name}/".encode()
        resp = self._call("/etcdserverpb.KV/Range",
                          RangeRequest(key=prefix,
                                       range_end=_prefix_range_end(prefix)),
                          RangeResponse)
        members = []
        for kv in resp.kvs:
            try:
                members.append(ServingService.parse(
                    bytes(kv.value).decode()))
            except ValueError:
                continue
        return sorted(members, key=lambda s: s.serialize())

    def _poll_loop(self) -> None:
        """Watch-driven: open a Watch stream on the prefix; every
        created/event response triggers a Range re-fetch + notify. While
        the stream can't be established, degrade to TTL/2 polling."""
        last = None

        def refresh():
            nonlocal last
            members = self.fetch_members()
            if members != last:
                last = members
                self._notify(members)

        prefix = f"/service/{self.name}/".encode()
        while not self._stop.is_set():
            try:
                refresh()
            except grpc.RpcError:
                log.warning("etcd member fetch failed", exc_info=True)
                self._stop.wait(self.ttl / 2)
                continue
            try:
                stream = self._chan().stream_stream(
                    "/etcdserverpb.Watch/Watch",
                    request_serializer=lambda r: r.encode(),
                    response_deserializer=WatchResponse.decode)
                metadata = [("token", self._token)] if self._token \
                    else None

                def requests_iter():
                    yield WatchRequest(create_request=WatchCreateRequest(
                        key=prefix,
                        range_end=_prefix_range_end(prefix)))
                    self._stop.wait()          # hold the send side open

                for resp in stream(requests_iter(), metadata=metadata):
                    if self._stop.is_set():
                        return
                    if resp.canceled:
                        break
                    if resp.events or resp.created:
                        refresh()
            except grpc.RpcError:
                if self._stop.is_set():
                    return
                log.warning("etcd watch unavailable; polling",
                            exc_info=True)
                self._stop.wait(self.ttl / 2)
