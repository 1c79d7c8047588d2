"""Ring/cluster/discovery/proxy-tier tests.

Mirrors the reference's contract tests (pkg/taskhandler/cluster_test.go):
ring determinism, single-node case, minimal movement + exact reversion on
membership change — plus full proxy->cache forwarding over real sockets
(REST and gRPC) and the metrics merger.
"""
import threading
import time

import grpc
import numpy as np
import pytest
from aiohttp.test_utils import TestClient, TestServer

from tfservingcache_amd.cachemanager import (CacheManager, LRUCache,
                                             ModelPool, make_cpu_loader)
from tfservingcache_amd.cachemanager.providers import DiskModelProvider
from tfservingcache_amd.models import write_model_repo
from tfservingcache_amd.taskhandler import (ClusterConnection,
                                            ConsistentHashRing, model_key)
from tfservingcache_amd.taskhandler.discovery import (FileDiscovery,
                                                      MockDiscovery,
                                                      ServingService)
from tfservingcache_amd.tfservingproxy import (GrpcForwarder,
                                               LocalServingHandler,
                                               make_cache_grpc_server,
                                               make_cache_rest_app,
                                               make_proxy_grpc_server,
                                               make_proxy_rest_app)
from tfservingcache_amd.wire import messages as m
from tfservingcache_amd.wire.tensor import (numpy_to_tensorproto,
                                            tensorproto_to_numpy)


# ---------------------------------------------------------------------------
# ring contract (cluster_test.go:51-227)
# ---------------------------------------------------------------------------
def test_ring_deterministic():
    ring = ConsistentHashRing()
    members = [f"host_{i}:80:81" for i in range(8)]
    ring.set_members(members)
    keys = [model_key(f"model_{i}", 1) for i in range(6)]
    for key in keys:
        first = ring.get_n(key, 3)
        assert len(first) == 3
        assert len(set(first)) == 3
        for _ in range(1000):
            assert ring.get_n(key, 3) == first


def test_ring_single_node():
    ring = ConsistentHashRing()
    ring.set_members(["only:1:2"])
    for i in range(100):
        assert ring.get_n(f"key{i}", 3) == ["only:1:2"]


def test_ring_membership_change_reverts():
    ring = ConsistentHashRing()
    members = [f"host_{i}:80:81" for i in range(10)]
    ring.set_members(members)
    keys = [f"k{i}" for i in range(200)]
    before = {k: ring.get(k) for k in keys}

    ring.set_members(members[:-1])          # drop one node
    after = {k: ring.get(k) for k in keys}
    moved = [k for k in keys if before[k] != after[k]]
    # only keys owned by the dropped node may move
    for k in keys:
        if before[k] != members[-1]:
            assert after[k] == before[k], f"{k} moved needlessly"
    assert moved  # something moved

    ring.set_members(members)               # revert membership
    reverted = {k: ring.get(k) for k in keys}
    assert reverted == before               # exact reversion


def test_cluster_connection_updates():
    disc = MockDiscovery()
    cc = ClusterConnection(disc, replicas_per_model=2)
    cc.connect(ServingService("me", 1, 2))
    assert disc.registered is not None
    disc.generate_members(5)
    assert cc.n_members() == 5
    nodes = cc.find_nodes_for_key("m##1")
    assert len(nodes) == 2
    assert nodes[0].host.startswith("testhost_")
    # random replica pick stays within the owner set
    owners = {n.serialize() for n in nodes}
    for _ in range(50):
        assert cc.node_for_key("m", 1).serialize() in owners


def test_file_discovery(tmp_path):
    d1 = FileDiscovery(str(tmp_path / "cluster"), heartbeat_ttl=0.5,
                       poll_interval=0.1)
    d2 = FileDiscovery(str(tmp_path / "cluster"), heartbeat_ttl=0.5,
                       poll_interval=0.1)
    seen = []
    d1.add_listener(lambda ms: seen.append(list(ms)))
    try:
        d1.register(ServingService("hostA", 1, 2))
        d2.register(ServingService("hostB", 3, 4))
        deadline = time.time() + 5
        while time.time() < deadline:
            if seen and len(seen[-1]) == 2:
                break
            time.sleep(0.05)
        assert seen and len(seen[-1]) == 2
        hosts = {s.host for s in seen[-1]}
        assert hosts == {"hostA", "hostB"}
    finally:
        d1.unregister()
        d2.unregister()


# ---------------------------------------------------------------------------
# proxy -> cache forwarding over real sockets
# ---------------------------------------------------------------------------
@pytest.fixture()
def cache_node(tmp_path):
    write_model_repo(str(tmp_path / "repo"),
                     [("half_plus_two", 1, "half_plus_two")])
    provider = DiskModelProvider(str(tmp_path / "repo"))
    cache = LRUCache(str(tmp_path / "cache"), 10 ** 8)
    pool = ModelPool(make_cpu_loader(cache), 2)
    cm = CacheManager(provider, cache, pool)
    handler = LocalServingHandler(cm)
    server, _health = make_cache_grpc_server(handler)
    grpc_port = server.add_insecure_port("127.0.0.1:0")
    server.start()
    yield handler, grpc_port
    server.stop(None)


def test_proxy_grpc_forwarding(cache_node):
    handler, cache_port = cache_node
    picked = []

    def pick(model, version):
        picked.append((model, version))
        return f"127.0.0.1:{cache_port}"

    proxy, _h, fwd = make_proxy_grpc_server(pick)
    proxy_port = proxy.add_insecure_port("127.0.0.1:0")
    proxy.start()
    try:
        ch = grpc.insecure_channel(f"127.0.0.1:{proxy_port}")
        predict = ch.unary_unary(
            "/tensorflow.serving.PredictionService/Predict",
            request_serializer=lambda r: r.encode(),
            response_deserializer=m.PredictResponse.decode)
        resp = predict(m.PredictRequest(
            model_spec=m.ModelSpec(name="half_plus_two",
                                   version=m.Int64Value(value=1)),
            inputs={"x": numpy_to_tensorproto(
                np.array([2.0, 4.0], dtype=np.float32))}), timeout=10)
        np.testing.assert_allclose(
            tensorproto_to_numpy(resp.outputs["y"]), [3.0, 4.0])
        assert picked == [("half_plus_two", 1)]
        ch.close()
    finally:
        proxy.stop(None)
        fwd.close()


async def test_proxy_rest_forwarding(cache_node):
    handler, _ = cache_node
    # cache REST app on a real port
    cache_app = make_cache_rest_app(handler)
    cache_server = TestServer(cache_app)
    await cache_server.start_server()
    target = f"127.0.0.1:{cache_server.port}"

    proxy_app = make_proxy_rest_app(lambda mdl, v: target)
    client = TestClient(TestServer(proxy_app))
    await client.start_server()
    try:
        resp = await client.post(
            "/v1/models/half_plus_two/versions/1:predict",
            json={"instances": [1.0, 3.0]})
        assert resp.status == 200
        assert await resp.json() == {"predictions": [2.5, 3.5]}

        resp = await client.post("/v1/bogus", json={})
        assert resp.status == 404
    finally:
        await client.close()
        await cache_server.close()


def test_grpc_forwarder_channel_cache(cache_node):
    _handler, cache_port = cache_node
    fwd = GrpcForwarder()
    t = f"127.0.0.1:{cache_port}"
    ch1 = fwd.channel(t)
    ch2 = fwd.channel(t)
    assert ch1 is ch2           # dedup (reference raced here, §2.3)
    fwd.close()


def test_metrics_merger_merges_external():
    import http.server

    class H(http.server.BaseHTTPRequestHandler):
        def do_GET(self):
            body = (b"# TYPE tensorflow_core_counter counter\n"
                    b"tensorflow_core_counter 42\n")
            self.send_response(200)
            self.end_headers()
            self.wfile.write(body)

        def log_message(self, *a):
            pass

    srv = http.server.HTTPServer(("127.0.0.1", 0), H)
    thread = threading.Thread(target=srv.serve_forever, daemon=True)
    thread.start()
    try:
        from tfservingcache_amd.taskhandler import MetricsMerger
        merger = MetricsMerger(
            f"http://127.0.0.1:{srv.server_address[1]}/metrics")
        text = merger.render().decode()
        assert "tensorflow_core_counter 42" in text
        assert "tfservingcache_proxy_requests" in text
    finally:
        srv.shutdown()


def test_file_discovery_member_expiry(tmp_path):
    """A member that stops heartbeating drops out after 3*TTL (the
    lease-expiry liveness contract, etcd.go:139-146 analog)."""
    d1 = FileDiscovery(str(tmp_path / "c"), heartbeat_ttl=0.3,
                       poll_interval=0.1)
    d2 = FileDiscovery(str(tmp_path / "c"), heartbeat_ttl=0.3,
                       poll_interval=0.1)
    seen = []
    d1.add_listener(lambda ms: seen.append([s.host for s in ms]))
    try:
        d1.register(ServingService("alive", 1, 2))
        d2.register(ServingService("dying", 3, 4))
        deadline = time.time() + 5
        while time.time() < deadline and (
                not seen or sorted(seen[-1]) != ["alive", "dying"]):
            time.sleep(0.05)
        assert sorted(seen[-1]) == ["alive", "dying"]
        # kill d2's heartbeat without deregistering (simulated crash)
        d2._stop.set()
        deadline = time.time() + 6
        while time.time() < deadline and seen[-1] != ["alive"]:
            time.sleep(0.1)
        assert seen[-1] == ["alive"]
    finally:
        d1.unregister()
        d2.unregister()


def test_ring_load_balance():
    """20 vnodes/member should spread keys reasonably evenly: over 10k
    model##version keys on 8 members, no member owns more than 2.2x or
    less than 0.35x its fair share (the reference's ring lib uses the
    same CRC32 + 20-replica construction)."""
    from collections import Counter
    ring = ConsistentHashRing()
    members = [f"10.0.0.{i}:8094:8095" for i in range(8)]
    ring.set_members(members)
    counts = Counter(ring.get_n(f"model_{i:05d}##1", 1)[0]
                     for i in range(10000))
    fair = 10000 / 8
    assert set(counts) == set(members)           # everyone owns keys
    for m, c in counts.items():
        assert 0.35 * fair < c < 2.2 * fair, (m, c)


def test_ring_replica_distinctness_and_spread():
    """get_n returns N DISTINCT members, and replica PAIRS vary (a
    degenerate ring would co-locate every pair)."""
    ring = ConsistentHashRing()
    members = [f"n{i}" for i in range(6)]
    ring.set_members(members)
    pairs = set()
    for i in range(2000):
        owners = ring.get_n(f"m{i}##1", 2)
        assert len(owners) == 2 and len(set(owners)) == 2
        pairs.add(tuple(sorted(owners)))
    assert len(pairs) >= 10     # many distinct replica pairs in use
