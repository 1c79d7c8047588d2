"""Kubernetes discovery backend (Endpoints watch via the API server).

Behavior mirrors the reference's k8s integration
(pkg/taskhandler/discovery/kubernetes/kubernetes.go):
  * in-cluster config: token + CA + namespace from the serviceaccount
    files (kubernetes.go:163-180);
  * watch Endpoints with a fieldSelector (kubernetes.go:83-85), rebuild
    the full member list from subsets on every event, matching ports by
    NAME — grpccache/httpcache by default (kubernetes.go:102-124);
  * registration is implicit (pod membership in the Service) and
    unregister is a no-op (kubernetes.go:154).
"""
from __future__ import annotations

import json
import logging
import os
import threading
from typing import Dict, List, Optional

import requests

from .base import DiscoveryService, ServingService

log = logging.getLogger("tfsc.discovery.k8s")

SA_DIR = "/var/run/secrets/kubernetes.io/serviceaccount"


class KubernetesDiscovery(DiscoveryService):
    def __init__(self, field_selector: Optional[Dict[str, str]] = None,
                 port_names: Optional[Dict[str, str]] = None,
                 api_base: Optional[str] = None,
                 namespace: Optional[str] = None,
                 token: Optional[str] = None,
                 verify=None):
        super().__init__()
        self.field_selector = field_selector or {}
        port_names = port_names or {}
        self.grpc_port_name = port_names.get("grpcCache", "grpccache")
        self.http_port_name = port_names.get("httpCache", "httpcache")
        if api_base is None:
            host = os.environ.get("KUBERNETES_SERVICE_HOST", "kubernetes.default.svc")
            port = os.environ.get("KUBERNETES_SERVICE_PORT", "443")
            api_base = f"https://{host}:{port}"
        self.api_base = api_base.rstrip("/")
        self.namespace = namespace or self._read_sa("namespace") or "default"
        self.token = token if token is not None else self._read_sa("token")
        if verify is None:
            ca = os.path.join(SA_DIR, "ca.crt")
            verify = ca if os.path.exists(ca) else True
        self.verify = verify
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None
        self._session = requests.Session()
        if self.token:
            self._session.headers["Authorization"] = f"Bearer {self.token}"

    @staticmethod
    def _read_sa(name: str) -> Optional[str]:
        path = os.path.join(SA_DIR, name)
        try:
            with open(path) as f:
                return f.read().strip()
        except OSError:
            return None

    # -- DiscoveryService --------------------------------------------------
    def register(self, service: ServingService) -> None:
        # registration is implicit via pod Service membership
        self._thread = threading.Thread(target=self._watch_loop, daemon=True)
        self._thread.start()

    def unregister(self) -> None:
        self._stop.set()
        if self._thread:
            self._thread.join(timeout=2)

    # -- internals ---------------------------------------------------------
    def _selector(self) -> str:
        return ",".join(f"{k}={v}" for k, v in self.field_selector.items())

    def _endpoints_url(self) -> str:
        return (f"{self.api_base}/api/v1/namespaces/{self.namespace}"
                f"/endpoints")

    def parse_endpoints(self, obj: dict) -> List[ServingService]:
        members = []
        for subset in obj.get("subsets") or []:
            grpc_port = rest_port = 0
            for p in subset.get("ports") or []:
                if p.get("name") == self.grpc_port_name:
                    grpc_port = p.get("port", 0)
                elif p.get("name") == self.http_port_name:
                    rest_port = p.get("port", 0)
            # only READY addresses gate membership (readiness = liveness
            # in the mesh, kubernetes.go:102-115)
            for addr in subset.get("addresses") or []:
                ip = addr.get("ip")
                if ip and grpc_port:
                    members.append(ServingService(ip, rest_port, grpc_port))
        return sorted(members, key=lambda s: s.serialize())

    def _watch_loop(self) -> None:
        last = None
        while not self._stop.is_set():
            try:
                params = {"watch": "true"}
                if self._selector():
                    params["fieldSelector"] = self._selector()
                with self._session.get(self._endpoints_url(), params=params,
                                       stream=True, timeout=(10, 60),
                                       verify=self.verify) as r:
                    r.raise_for_status()
                    for line in r.iter_lines():
                        if self._stop.is_set():
                            return
                        if not line:
                            continue
                        try:
                            event = json.loads(line)
                        except ValueError:
                            continue
                        obj = event.get("object", {})
                        members = self.parse_endpoints(obj)
                        if members != last:
                            last = members
                            self._notify(members)
            except requests.RequestException:
                # re-watch on channel breakage (kubernetes.go:96-100)
                log.warning("k8s endpoints watch broke; retrying",
                            exc_info=True)
                self._stop.wait(1.0)
