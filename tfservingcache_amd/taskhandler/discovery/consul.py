"""Consul discovery backend (HTTP API).

Behavior mirrors the reference's consul integration
(pkg/taskhandler/discovery/consul/consul.go):
  * register an agent service with a TTL check and
    DeregisterCriticalServiceAfter = 100*TTL (consul.go:49-68);
  * heartbeat the TTL check every TTL/2, pass/fail from the health
    callback (consul.go:138-160);
  * watch /v1/health/service/<name>?passing with BLOCKING QUERIES
    (index + wait=55s long poll): membership changes propagate the
    moment consul sees them, instead of on the reference's 5s poll tick
    (consul.go:70-117). Falls back to plain polling on errors or
    non-indexed responses;
  * ports are encoded as tags "rest:<p>" / "grpc:<p>" (consul.go:54-57)
    plus "slot:<gpuN>" for per-GPU ring slots.
"""
from __future__ import annotations

import logging
import threading
import uuid
from typing import Callable, List, Optional

import requests

from .base import DiscoveryService, ServingService

log = logging.getLogger("tfsc.discovery.consul")

POLL_INTERVAL = 5.0


class ConsulDiscovery(DiscoveryService):
    def __init__(self, service_name: str, service_id: str = "",
                 address: str = "http://127.0.0.1:8500",
                 heartbeat_ttl: float = 5.0,
                 health_check: Optional[Callable[[], bool]] = None,
                 poll_interval: float = POLL_INTERVAL):
        super().__init__()
        self.name = service_name
        self.service_id = service_id or f"{service_name}-{uuid.uuid4().hex[:8]}"
        self.base = address.rstrip("/")
        self.ttl = heartbeat_ttl
        self.health_check = health_check or (lambda: True)
        self.poll_interval = poll_interval
        self._stop = threading.Event()
        self._threads: List[threading.Thread] = []
        self._session = requests.Session()

    def register(self, service: ServingService) -> None:
        tags = [f"rest:{service.rest_port}", f"grpc:{service.grpc_port}"]
        if service.slot:
            tags.append(f"slot:{service.slot}")
        payload = {
            "ID": self.service_id,
            "Name": self.name,
            "Address": service.host,
            "Port": service.grpc_port,
            "Tags": tags,
            "Check": {
                "CheckID": f"service:{self.service_id}",
                "TTL": f"{self.ttl}s",
                "DeregisterCriticalServiceAfter": f"{int(self.ttl * 100)}s",
            },
        }
        r = self._session.put(
            f"{self.base}/v1/agent/service/register", json=payload,
            timeout=10)
        r.raise_for_status()
        t1 = threading.Thread(target=self._ttl_loop, daemon=True)
        t2 = threading.Thread(target=self._poll_loop, daemon=True)
        self._threads = [t1, t2]
        t1.start()
        t2.start()

    def unregister(self) -> None:
        self._stop.set()
        for t in self._threads:
            t.join(timeout=2)
        try:
            self._session.put(
                f"{self.base}/v1/agent/service/deregister/{self.service_id}",
                timeout=10)
        except requests.RequestException:
            log.warning("consul deregister failed", exc_info=True)

    def _ttl_loop(self) -> None:
        while not self._stop.wait(self.ttl / 2):
            status = "pass" if self._safe_health() else "fail"
            try:
                self._session.put(
                    f"{self.base}/v1/agent/check/{status}/"
                    f"service:{self.service_id}", timeout=10)
            except requests.RequestException:
                log.warning("consul TTL update failed", exc_info=True)

    def _safe_health(self) -> bool:
        try:
            return bool(self.health_check())
        except Exception:       # noqa: BLE001
            return False

    def _poll_loop(self) -> None:
        last = None
        index = None
        while not self._stop.is_set():
            try:
                members, index = self.fetch_members(index)
                if members != last:
                    last = members
                    self._notify(members)
                if index is None:       # server doesn't support blocking
                    self._stop.wait(self.poll_interval)
            except requests.RequestException:
                log.warning("consul health watch failed", exc_info=True)
                index = None
                self._stop.wait(self.poll_interval)

    def fetch_members(self, index: Optional[int] = None
                      ) -> "tuple[List[ServingService], Optional[int]]":
        """One health query. With `index`, a blocking query: consul
        holds the request (wait=55s) until the service list changes past
        that index, so the loop is change-driven."""
        params = {"passing": "true"}
        timeout = 10.0
        if index is not None:
            params["index"] = str(index)
            params["wait"] = "55s"
            timeout = 70.0
        r = self._session.get(
            f"{self.base}/v1/health/service/{self.name}",
            params=params, timeout=timeout)
        r.raise_for_status()
        new_index = None
        try:
            new_index = int(r.headers.get("X-Consul-Index", ""))
            # consul docs: reset on non-monotonic or absurd indexes
            if new_index < 1 or (index is not None and new_index < index):
                new_index = None
        except ValueError:
            pass
        members = []
        for entry in r.json():
            svc = entry.get("Service", {})
            host = svc.get("Address") or entry.get("Node", {}).get("Address")
            rest, grpc_, slot = 0, svc.get("Port", 0), ""
            for tag in svc.get("Tags", []):
                if tag.startswith("rest:"):
                    rest = int(tag[5:])
                elif tag.startswith("grpc:"):
                    grpc_ = int(tag[5:])
                elif tag.startswith("slot:"):
                    slot = tag[5:]
            if host:
                members.append(ServingService(host, rest, grpc_, slot))
        return sorted(members, key=lambda s: s.serialize()), new_index
