"""Service discovery interface + in-memory/static/file backends.

Mirrors the reference's DiscoveryService contract
(pkg/taskhandler/cluster.go:25-30): RegisterService/UnregisterService +
a channel of full member-list updates. Here the "channel" is a callback
subscription (add_listener), which also removes the reference's unlocked
ListUpdatedChans maps.

Members are ServingService records serialized "host:restPort:grpcPort"
(cluster.go:142-164 format), optionally suffixed ":gpuN" for per-GPU
ring slots.
"""
from __future__ import annotations

import abc
import json
import logging
import os
import threading
import time
from dataclasses import dataclass
from typing import Callable, List, Optional

log = logging.getLogger("tfsc.discovery")


@dataclass(frozen=True)
class ServingService:
    host: str
    rest_port: int
    grpc_port: int
    slot: str = ""          # optional GPU-slot tag ("gpu0"...)

    def serialize(self) -> str:
        base = f"{self.host}:{self.rest_port}:{self.grpc_port}"
        return f"{base}:{self.slot}" if self.slot else base

    @classmethod
    def parse(cls, s: str) -> "ServingService":
        parts = s.split(":")
        if len(parts) < 3:
            raise ValueError(f"bad member string {s!r}")
        slot = parts[3] if len(parts) > 3 else ""
        return cls(parts[0], int(parts[1]), int(parts[2]), slot)

    @property
    def rest_addr(self) -> str:
        return f"{self.host}:{self.rest_port}"

    @property
    def grpc_addr(self) -> str:
        return f"{self.host}:{self.grpc_port}"


Listener = Callable[[List[ServingService]], None]


class DiscoveryService(abc.ABC):
    """Register this node; push full member lists to listeners."""

    def __init__(self):
        self._listeners: List[Listener] = []
        self._listeners_lock = threading.Lock()

    def add_listener(self, fn: Listener) -> None:
        with self._listeners_lock:
            self._listeners.append(fn)

    def _notify(self, members: List[ServingService]) -> None:
        with self._listeners_lock:
            listeners = list(self._listeners)
        for fn in listeners:
            try:
                fn(members)
            except Exception:       # noqa: BLE001
                log.exception("discovery listener failed")

    @abc.abstractmethod
    def register(self, service: ServingService) -> None: ...

    @abc.abstractmethod
    def unregister(self) -> None: ...


class StaticDiscovery(DiscoveryService):
    """Fixed peer list from config (serviceDiscovery.static.members)."""

    def __init__(self, members: List[str]):
        super().__init__()
        self._members = [ServingService.parse(m) for m in members]

    def register(self, service: ServingService) -> None:
        if service not in self._members:
            self._members.append(service)
        self._notify(list(self._members))

    def unregister(self) -> None:
        pass


class MockDiscovery(DiscoveryService):
    """Test double (the reference's DiscoveryServiceMock,
    cluster_test.go:12-49)."""

    def __init__(self):
        super().__init__()
        self.registered: Optional[ServingService] = None

    def register(self, service: ServingService) -> None:
        self.registered = service

    def unregister(self) -> None:
        self.registered = None

    def push(self, members: List[ServingService]) -> None:
        self._notify(members)

    def generate_members(self, n: int, base_port: int = 8000) -> None:
        self.push([ServingService(f"testhost_{i}", base_port + i,
                                  base_port + 1000 + i)
                   for i in range(n)])


class FileDiscovery(DiscoveryService):
    """Shared-directory membership with heartbeat files — a TTL-lease
    backend in the spirit of the reference's etcd lease registration
    (discovery/etcd/etcd.go:134-148) that works with any shared
    filesystem (and in multi-process tests). Each member writes
    <dir>/<id>.json every ttl/2; files older than 3*ttl are expired."""

    def __init__(self, directory: str, heartbeat_ttl: float = 5.0,
                 poll_interval: Optional[float] = None):
        super().__init__()
        self.dir = directory
        self.ttl = heartbeat_ttl
        self.poll = poll_interval or max(heartbeat_ttl / 2, 0.2)
        self._service: Optional[ServingService] = None
        self._stop = threading.Event()
        self._threads: List[threading.Thread] = []
        self._last: List[ServingService] = []
        os.makedirs(directory, exist_ok=True)

    def _my_path(self) -> str:
        assert self._service is not None
        return os.path.join(
            self.dir, self._service.serialize().replace(":", "_") + ".json")

    def register(self, service: ServingService) -> None:
        self._service = service
        self._write_heartbeat()
        t1 = threading.Thread(target=self._heartbeat_loop, daemon=True)
        t2 = threading.Thread(target=self._watch_loop, daemon=True)
        self._threads = [t1, t2]
        t1.start()
        t2.start()

    def unregister(self) -> None:
        self._stop.set()
        for t in self._threads:
            t.join(timeout=2)
        if self._service is not None:
            try:
                os.remove(self._my_path())
            except OSError:
                pass

    def _write_heartbeat(self) -> None:
        path = self._my_path()
        tmp = path + ".tmp"
        with open(tmp, "w") as f:
            json.dump({"member": self._service.serialize(),
                       "ts": time.time()}, f)
        os.replace(tmp, path)

    def _heartbeat_loop(self) -> None:
        while not self._stop.wait(self.ttl / 2):
            try:
                self._write_heartbeat()
            except OSError:
                log.exception("heartbeat write failed")

    def scan(self) -> List[ServingService]:
        now = time.time()
        members = []
        for fname in os.listdir(self.dir):
            if not fname.endswith(".json"):
                continue
            try:
                with open(os.path.join(self.dir, fname)) as f:
                    data = json.load(f)
                if now - data["ts"] <= 3 * self.ttl:
                    members.append(ServingService.parse(data["member"]))
            except (OSError, ValueError, KeyError):
                continue
        return sorted(members, key=lambda s_: s_.serialize())

    def _watch_loop(self) -> None:
        while not self._stop.wait(self.poll):
            members = self.scan()
            if members != self._last:
                self._last = members
                self._notify(members)
