"""Consul blocking-query and etcd watch discovery backends against
in-process mock servers: registration, membership, and CHANGE-DRIVEN
propagation (update must arrive much faster than the polling fallback
cadence would allow)."""
import json
import queue
import socket
import threading
import time
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer
from urllib.parse import parse_qs, urlparse

import grpc
import pytest

from tfservingcache_amd.taskhandler.discovery.base import ServingService
from tfservingcache_amd.taskhandler.discovery.consul import ConsulDiscovery
from tfservingcache_amd.taskhandler.discovery import etcd as e


def free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


# -- consul mock ------------------------------------------------------------
class ConsulState:
    def __init__(self):
        self.index = 1
        self.services = []          # consul health JSON entries
        self.cond = threading.Condition()
        self.registered = {}
        self.blocking_queries = 0

    def set_members(self, members):
        with self.cond:
            self.services = [
                {"Node": {"Address": m.host},
                 "Service": {"Address": m.host, "Port": m.grpc_port,
                             "Tags": [f"rest:{m.rest_port}",
                                      f"grpc:{m.grpc_port}"]}}
                for m in members]
            self.index += 1
            self.cond.notify_all()


def make_consul_handler(state: ConsulState):
    class Handler(BaseHTTPRequestHandler):
        def log_message(self, *a):      # noqa: D102
            pass

        def _json(self, obj, index=None):
            body = json.dumps(obj).encode()
            self.send_response(200)
            if index is not None:
                self.send_header("X-Consul-Index", str(index))
            self.send_header("Content-Length", str(len(body)))
            self.end_headers()
            self.wfile.write(body)

        def do_PUT(self):
            length = int(self.headers.get("Content-Length") or 0)
            body = self.rfile.read(length) if length else b""
            if self.path.startswith("/v1/agent/service/register"):
                payload = json.loads(body)
                state.registered[payload["ID"]] = payload
            self._json({})

        def do_GET(self):
            u = urlparse(self.path)
            if u.path.startswith("/v1/health/service/"):
                q = parse_qs(u.query)
                want_index = int(q.get("index", ["0"])[0])
                with state.cond:
                    if want_index:
                        state.blocking_queries += 1
                        deadline = time.time() + 5
                        while (state.index <= want_index and
                               time.time() < deadline):
                            state.cond.wait(deadline - time.time())
                    self._json(state.services, index=state.index)
            else:
                self._json({})
    return Handler


def test_consul_blocking_query_discovery():
    state = ConsulState()
    httpd = ThreadingHTTPServer(("127.0.0.1", 0),
                                make_consul_handler(state))
    port = httpd.server_address[1]
    threading.Thread(target=httpd.serve_forever, daemon=True).start()

    seen = queue.Queue()
    # poll_interval 30s: only the blocking query can deliver fast updates
    d = ConsulDiscovery("tfsc", address=f"http://127.0.0.1:{port}",
                        heartbeat_ttl=1.0, poll_interval=30.0)
    d.add_listener(seen.put)
    svc_a = ServingService("10.0.0.1", 8093, 8100)
    try:
        state.set_members([svc_a])
        d.register(ServingService("10.0.0.1", 8093, 8100))
        assert state.registered          # register call hit the mock
        members = seen.get(timeout=5)
        assert [m.serialize() for m in members] == [svc_a.serialize()]

        t0 = time.time()
        svc_b = ServingService("10.0.0.2", 8093, 8100)
        state.set_members([svc_a, svc_b])
        members = seen.get(timeout=5)
        assert len(members) == 2
        assert time.time() - t0 < 3.0    # change-driven, not 30s poll
        assert state.blocking_queries >= 1
    finally:
        d.unregister()
        httpd.shutdown()


# -- etcd mock --------------------------------------------------------------
class EtcdMock(grpc.GenericRpcHandler):
    def __init__(self):
        self.kv = {}
        self.lock = threading.Lock()
        self.watchers = []              # queues of WatchResponse
        self.watch_created = threading.Event()

    def put(self, key: bytes, value: bytes):
        with self.lock:
            self.kv[key] = value
            watchers = list(self.watchers)
        ev = e.Event(type=0, kv=e.KeyValue(key=key, value=value))
        for q in watchers:
            q.put(e.WatchResponse(watch_id=1, events=[ev]))

    # gRPC handlers
    def _range(self, req, ctx):
        with self.lock:
            kvs = [e.KeyValue(key=k, value=v)
                   for k, v in sorted(self.kv.items())
                   if req.key <= k < req.range_end]
        return e.RangeResponse(kvs=kvs, count=len(kvs))

    def _put(self, req, ctx):
        self.put(bytes(req.key), bytes(req.value))
        return e.PutResponse()

    def _lease(self, req, ctx):
        return e.LeaseGrantResponse(id=7, ttl=req.ttl)

    def _watch(self, req_iter, ctx):
        next(req_iter)                   # the create request
        q = queue.Queue()
        with self.lock:
            self.watchers.append(q)
        self.watch_created.set()
        yield e.WatchResponse(watch_id=1, created=True)
        while ctx.is_active():
            try:
                yield q.get(timeout=0.2)
            except queue.Empty:
                continue

    def service(self, call):
        enc = lambda r: r.encode()      # noqa: E731
        if call.method == "/etcdserverpb.KV/Range":
            return grpc.unary_unary_rpc_method_handler(
                self._range, request_deserializer=e.RangeRequest.decode,
                response_serializer=enc)
        if call.method == "/etcdserverpb.KV/Put":
            return grpc.unary_unary_rpc_method_handler(
                self._put, request_deserializer=e.PutRequest.decode,
                response_serializer=enc)
        if call.method == "/etcdserverpb.Lease/LeaseGrant":
            return grpc.unary_unary_rpc_method_handler(
                self._lease,
                request_deserializer=e.LeaseGrantRequest.decode,
                response_serializer=enc)
        if call.method == "/etcdserverpb.Watch/Watch":
            return grpc.stream_stream_rpc_method_handler(
                self._watch,
                request_deserializer=e.WatchRequest.decode,
                response_serializer=enc)
        return None


def test_etcd_watch_discovery():
    from concurrent import futures
    mock = EtcdMock()
    server = grpc.server(futures.ThreadPoolExecutor(max_workers=8))
    server.add_generic_rpc_handlers((mock,))
    port = server.add_insecure_port("127.0.0.1:0")
    server.start()

    seen = queue.Queue()
    # ttl 20 -> polling fallback runs every 10s; only the watch stream
    # can deliver updates fast
    d = e.EtcdDiscovery("tfsc", [f"127.0.0.1:{port}"], heartbeat_ttl=20)
    d.add_listener(seen.put)
    try:
        d.register(ServingService("10.0.0.1", 8093, 8100))
        members = seen.get(timeout=5)
        assert len(members) == 1
        assert members[0].host == "10.0.0.1"
        assert mock.watch_created.wait(timeout=5)

        t0 = time.time()
        other = ServingService("10.0.0.2", 8093, 8100)
        mock.put(b"/service/tfsc/otherid", other.serialize().encode())
        members = seen.get(timeout=5)
        assert len(members) == 2
        assert time.time() - t0 < 3.0    # watch-driven, not 10s poll
    finally:
        d.unregister()
        server.stop(grace=0.5)


# -- kubernetes mock --------------------------------------------------------
def test_k8s_endpoints_watch_discovery():
    """KubernetesDiscovery against a mock API server streaming watch
    events: readiness-gated members, port matching by NAME, live
    updates, re-watch after channel breakage."""
    from tfservingcache_amd.taskhandler.discovery.kubernetes import \
        KubernetesDiscovery

    events = queue.Queue()
    watch_count = [0]

    def endpoints_obj(ips):
        return {"type": "MODIFIED", "object": {"subsets": [{
            "addresses": [{"ip": ip} for ip in ips],
            "ports": [{"name": "grpccache", "port": 8100},
                      {"name": "httpcache", "port": 8093},
                      {"name": "other", "port": 9999}],
        }]}}

    class Handler(BaseHTTPRequestHandler):
        def log_message(self, *a):      # noqa: D102
            pass

        def do_GET(self):
            assert "watch=true" in self.path
            assert "fieldSelector=metadata.name%3Dtfsc" in self.path
            watch_count[0] += 1
            self.send_response(200)
            self.send_header("Transfer-Encoding", "chunked")
            self.end_headers()
            # stream events until the test closes the server
            sent = 0
            while sent < 10:
                try:
                    ev = events.get(timeout=3)
                except queue.Empty:
                    break
                body = json.dumps(ev).encode() + b"\n"
                self.wfile.write(f"{len(body):x}\r\n".encode() + body +
                                 b"\r\n")
                self.wfile.flush()
                sent += 1
            self.wfile.write(b"0\r\n\r\n")

    httpd = ThreadingHTTPServer(("127.0.0.1", 0), Handler)
    port = httpd.server_address[1]
    threading.Thread(target=httpd.serve_forever, daemon=True).start()

    seen = queue.Queue()
    d = KubernetesDiscovery(
        field_selector={"metadata.name": "tfsc"},
        api_base=f"http://127.0.0.1:{port}", namespace="ns",
        token="tok", verify=False)
    d.add_listener(seen.put)
    try:
        events.put(endpoints_obj(["10.1.0.1"]))
        d.register(ServingService("10.1.0.1", 8093, 8100))
        members = seen.get(timeout=5)
        assert [(m.host, m.rest_port, m.grpc_port) for m in members] == \
            [("10.1.0.1", 8093, 8100)]

        events.put(endpoints_obj(["10.1.0.1", "10.1.0.2"]))
        members = seen.get(timeout=5)
        assert len(members) == 2

        # channel breakage: the stream ends after the queued events;
        # the watcher must reconnect and pick up new state
        events.put(endpoints_obj(["10.1.0.3"]))
        members = seen.get(timeout=10)
        assert [m.host for m in members] == ["10.1.0.3"]
    finally:
        d.unregister()
        httpd.shutdown()
