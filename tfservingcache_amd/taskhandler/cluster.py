"""ClusterConnection: discovery -> consistent-hash ring.

Mirrors pkg/taskhandler/cluster.go: Connect registers this node with the
discovery service and re-seeds the ring on every full-membership update
(cluster.go:66-113); FindNodeForKey returns up to `replicas` distinct
owners via GetN (cluster.go:116-130). Keys are "model##version"
(taskhandler.go:84-92).

MI355X extension: when nodes register one member per GPU (slot tags),
keys land on (node, GPU) ring slots, so a model is owned by a specific
GPU's HBM pool and replicasPerModel replicas land on distinct GPUs.
"""
from __future__ import annotations

import logging
import random
import threading
from typing import List

from .discovery.base import DiscoveryService, ServingService
from .ring import ConsistentHashRing

log = logging.getLogger("tfsc.cluster")


def model_key(model_name: str, version) -> str:
    return f"{model_name}##{version}"


class ClusterConnection:
    def __init__(self, discovery: DiscoveryService,
                 replicas_per_model: int = 1):
        self.discovery = discovery
        self.replicas = max(1, replicas_per_model)
        self.ring = ConsistentHashRing()
        self._members_by_id = {}
        self._lock = threading.Lock()
        discovery.add_listener(self._on_members)

    def connect(self, service: ServingService) -> None:
        self.discovery.register(service)

    def disconnect(self) -> None:
        self.discovery.unregister()

    def _on_members(self, members: List[ServingService]) -> None:
        with self._lock:
            self._members_by_id = {m.serialize(): m for m in members}
            self.ring.set_members(list(self._members_by_id))
        log.info("cluster membership: %d members", len(members))

    def find_nodes_for_key(self, key: str) -> List[ServingService]:
        ids = self.ring.get_n(key, self.replicas)
        with self._lock:
            return [self._members_by_id[i] for i in ids
                    if i in self._members_by_id]

    def node_for_key(self, model_name: str, version) -> ServingService:
        """Random replica of the key's owner set (taskhandler.go:84-92)."""
        nodes = self.find_nodes_for_key(model_key(model_name, version))
        if not nodes:
            raise LookupError("no nodes in cluster")
        return random.choice(nodes)

    def n_members(self) -> int:
        return len(self.ring.members())
