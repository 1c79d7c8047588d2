"""Composition root — `python -m tfservingcache_amd.main`.

Mirrors the reference's cmd/taskhandler/main.go:
  * config.yaml from CWD + TFSC_ env overrides (cfg.go:10-26);
  * cache tier on cacheRestPort/cacheGrpcPort, proxy tier on
    proxyRestPort/proxyGrpcPort (config.yaml:1-4, main.go:45-113);
  * proxy tier disabled when serviceDiscovery.type is unset
    ("Proxy is disabled", main.go:103-105);
  * 30s health loop feeding the gRPC health servers from the probe-model
    check + provider check (main.go:35-42, cachemanager.go:76-89);
  * factories for model providers and discovery backends
    (main.go:115-192).

MI355X specifics: the serving backend is the in-process CDNA4 engine
(one model pool per visible GPU, models sharded over GPUs by the ring's
(node, gpu) slots; engine.gpus=-1 uses all).
"""
from __future__ import annotations

import asyncio
import logging
import os
import socket
import threading
import time
from typing import List, Optional

from aiohttp import web

from .cachemanager import (CacheManager, LRUCache, ModelPool,
                           make_cpu_loader, make_gpu_loader)
from .cachemanager.providers import DiskModelProvider
from .config import Config
from .taskhandler import ClusterConnection, MetricsMerger
from .taskhandler.discovery.base import (DiscoveryService, FileDiscovery,
                                         ServingService, StaticDiscovery)
from .tfservingproxy import (LocalServingHandler, make_cache_grpc_server,
                             make_cache_rest_app, make_proxy_grpc_server,
                             make_proxy_rest_app)
from .tfservingproxy.grpc_server import HealthState

log = logging.getLogger("tfsc.main")


def create_model_provider(cfg: Config):
    ptype = cfg.get_string("modelProvider.type")
    if ptype in ("diskProvider", "disk", ""):
        base = cfg.get_string("modelProvider.diskProvider.baseDir") or \
            cfg.get_string("modelProvider.diskProvider.basePath") or \
            "./model_repo"
        return DiskModelProvider(base)
    if ptype in ("s3Provider", "s3"):
        from .cachemanager.providers.s3 import S3ModelProvider
        return S3ModelProvider(
            bucket=cfg.get_string("modelProvider.s3.bucket"),
            base_path=cfg.get_string("modelProvider.s3.basePath"),
            endpoint_url=cfg.get_string("modelProvider.s3.endpoint") or None,
            region=cfg.get_string("modelProvider.s3.region") or "us-east-1")
    if ptype in ("azBlobProvider", "azblob"):
        from .cachemanager.providers.azblob import AZBlobModelProvider
        return AZBlobModelProvider(
            account=cfg.get_string("modelProvider.azBlob.account"),
            container=cfg.get_string("modelProvider.azBlob.container"),
            base_path=cfg.get_string("modelProvider.azBlob.basePath"),
            account_key=cfg.get_string("modelProvider.azBlob.accountKey"),
            endpoint=cfg.get_string("modelProvider.azBlob.endpoint") or None)
    raise ValueError(f"unknown modelProvider.type {ptype!r}")


def create_discovery_service(cfg: Config, health_check) -> Optional[DiscoveryService]:
    dtype = cfg.get_string("serviceDiscovery.type")
    ttl = cfg.get_float("serviceDiscovery.heartbeatTTL") or 5.0
    if not dtype:
        return None
    if dtype == "mock":
        from .taskhandler.discovery.base import MockDiscovery
        return MockDiscovery()
    if dtype == "static":
        return StaticDiscovery(
            [str(m) for m in cfg.get_list("serviceDiscovery.static.members")])
    if dtype == "file":
        return FileDiscovery(
            cfg.get_string("serviceDiscovery.file.directory") or
            "/tmp/tfsc-cluster", heartbeat_ttl=ttl)
    if dtype == "consul":
        from .taskhandler.discovery.consul import ConsulDiscovery
        return ConsulDiscovery(
            service_name=cfg.get_string(
                "serviceDiscovery.consul.serviceName") or "tfservingcache",
            service_id=cfg.get_string("serviceDiscovery.consul.serviceId"),
            address=cfg.get_string("serviceDiscovery.consul.address") or
            "http://127.0.0.1:8500",
            heartbeat_ttl=ttl, health_check=health_check)
    if dtype == "etcd":
        from .taskhandler.discovery.etcd import EtcdDiscovery
        auth = cfg.get_dict("serviceDiscovery.etcd.authorization")
        return EtcdDiscovery(
            service_name=cfg.get_string(
                "serviceDiscovery.etcd.serviceName") or "tfservingcache",
            endpoints=[str(e) for e in
                       cfg.get_list("serviceDiscovery.etcd.endpoints")],
            heartbeat_ttl=ttl,
            username=str(auth.get("username", "")),
            password=str(auth.get("password", "")))
    if dtype == "k8s":
        from .taskhandler.discovery.kubernetes import KubernetesDiscovery
        return KubernetesDiscovery(
            field_selector=cfg.get_dict(
                "serviceDiscovery.k8s.fieldSelector"),
            port_names=cfg.get_dict("serviceDiscovery.k8s.portNames"))
    raise ValueError(f"unknown serviceDiscovery.type {dtype!r}")


def _gpu_devices(cfg: Config) -> List[str]:
    n = cfg.get_int("engine.gpus")
    try:
        import torch
        if not torch.cuda.is_available():
            return []
        avail = torch.cuda.device_count()
    except Exception:       # noqa: BLE001
        return []
    if n < 0 or n > avail:
        n = avail
    return [f"cuda:{i}" for i in range(n)]


def create_cache_manager(cfg: Config,
                         devices: Optional[List[str]] = None
                         ) -> CacheManager:
    provider = create_model_provider(cfg)
    cache = LRUCache(
        cfg.get_string("modelCache.hostModelPath") or "./models",
        max_size_bytes=cfg.get_int("modelCache.size") or 10 ** 9,
        rebuild_from_disk=cfg.get_bool("modelCache.rebuildFromDisk"))
    max_models = cfg.get_int("serving.maxConcurrentModels") or 2
    if devices is None:
        devices = _gpu_devices(cfg)
    if devices:
        # models are hash-sharded over the node's visible GPUs (the
        # single-process analog of per-GPU ring slots; for one process
        # per GPU, set engine.gpus=1 and launch with CUDA_VISIBLE_DEVICES)
        loader = make_gpu_loader(
            cache, device=devices[0], devices=devices,
            max_batch=cfg.get_int("engine.maxbatch"),
            batching=cfg.get_bool("serving.batching.enabled"),
            batch_timeout_s=(cfg.get_float(
                "serving.batching.batchTimeoutMicros") or 2000.0) / 1e6,
            n_streams=cfg.get_int("engine.streamsPerGpu") or 6,
            dtype=cfg.get_string("engine.dtype") or "bf16")
        device = devices[0] if len(devices) == 1 else \
            f"cuda[0-{len(devices) - 1}]"
    else:
        loader = make_cpu_loader(cache)
        device = "cpu"
    hbm_budget = cfg.get_int("engine.hbmPoolBytes") or None

    def size_hint(name: str, version: int) -> int:
        # device footprint estimate: ~3x on-disk (fp32 master + bf16 +
        # transformed GEMM layouts)
        entry = cache.get(name, version)
        return 3 * entry.size_on_disk if entry else 0

    pool = ModelPool(loader, max_concurrent_models=max_models,
                     device=device, max_bytes=hbm_budget,
                     size_hint=size_hint if hbm_budget else None)
    fetch_timeout = cfg.get_float("serving.grpcConfigTimeout") or 10.0
    return CacheManager(provider, cache, pool,
                        model_fetch_timeout=fetch_timeout,
                        model_labels=cfg.get_bool("metrics.modelLabels"))


class Server:
    """One serving process. In multi-GPU production each GPU gets its
    own process (torchrun / torch.distributed.run, one rank per GPU):
    every rank registers a SLOT-TAGGED ring member (gpu<local_rank>)
    with its own 4 ports (base + local_rank * engine.portStride), so
    the ring's keys land on (node, GPU) slots — SURVEY §2.4. When
    proxy.replicasPerModel > 1 the cold-load path pushes model bytes
    to the other replica slots over the RCCL/xGMI plane instead of
    each replica re-fetching from the model store."""

    def __init__(self, cfg: Config):
        self.cfg = cfg
        self._plane = None
        self._rank = int(os.environ.get("RANK", "0"))
        self._world = int(os.environ.get("WORLD_SIZE", "1"))
        self._local_rank = int(os.environ.get("LOCAL_RANK",
                                              str(self._rank)))
        self._slot = ""
        devices = None
        plane_cfg = (cfg.get_string("engine.replicaPlane") or "").lower()
        self._plane_enabled = (self._world > 1 and
                               plane_cfg not in ("false", "0", "no"))
        if self._plane_enabled:
            self._slot = f"gpu{self._local_rank}"
            devices = self._init_distributed(cfg)
        self.cm = create_cache_manager(cfg, devices=devices)
        from .utils import metrics as mt
        mt.engine_stages.add_pool(self.cm.pool)
        self.handler = LocalServingHandler(self.cm)
        self.health = HealthState()
        self.cluster: Optional[ClusterConnection] = None
        self.discovery: Optional[DiscoveryService] = None
        self._stop = threading.Event()
        self._grpc_servers = []
        self._self_service = None
        self._last_member_ids = None
        self._handoff_fwd = None
        self._runners = []
        self._loop = None

        self.cache_rest_port = cfg.get_int("cacheRestPort") or 8094
        self.cache_grpc_port = cfg.get_int("cacheGrpcPort") or 8095
        self.proxy_rest_port = cfg.get_int("proxyRestPort") or 8093
        self.proxy_grpc_port = cfg.get_int("proxyGrpcPort") or 8100
        if self._plane_enabled:
            # per-GPU process: distinct ports per local rank
            stride = cfg.get_int("engine.portStride") or 10
            off = self._local_rank * stride
            self.cache_rest_port += off
            self.cache_grpc_port += off
            self.proxy_rest_port += off
            self.proxy_grpc_port += off

    def _init_distributed(self, cfg: Config) -> Optional[List[str]]:
        """One-process-per-GPU: init torch.distributed (nccl == RCCL on
        ROCm when GPUs are visible, gloo otherwise) and pin this process
        to its GPU. Returns the device list override for the loader."""
        import torch
        import torch.distributed as dist
        if not dist.is_initialized():
            backend = "nccl" if torch.cuda.is_available() else "gloo"
            dist.init_process_group(backend=backend)
        if torch.cuda.is_available():
            dev = self._local_rank % torch.cuda.device_count()
            torch.cuda.set_device(dev)
            return [f"cuda:{dev}"]
        return []

    def is_healthy(self) -> bool:
        probe = self.cfg.get_string("healthProbe.modelName")
        return self.cm.is_healthy(probe)

    # -- startup -----------------------------------------------------------
    def start(self) -> None:
        metrics_path = self.cfg.get_string("metrics.path") or \
            "/monitoring/prometheus/metrics"
        merger = MetricsMerger(
            self.cfg.get_string("serving.metricsScrapeUrl") or None,
            timeout=self.cfg.get_float("metrics.timeout") or 3.0)

        # cache tier gRPC: grpcio server, or the native nghttp2
        # front-end (serving.nativeFrontend) whose registered Predicts
        # never touch Python
        self._native_frontends = []
        if self.cfg.get_bool("serving.nativeFrontend"):
            from .tfservingproxy.native_frontend import NativeGrpcServer
            cache_grpc = NativeGrpcServer(
                self.handler, health=self.health,
                workers=self.cfg.get_int("serving.nativeWorkers") or 16)
            cache_grpc.add_insecure_port(f"[::]:{self.cache_grpc_port}")
            cache_grpc.start()
            self._grpc_servers.append(cache_grpc)
            self._native_frontends.append(cache_grpc)
        else:
            cache_grpc, _ = make_cache_grpc_server(
                self.handler, health=self.health,
                max_msg=self.cfg.get_int("serving.grpcMaxMsgSize") or
                16 * 1024 * 1024)
            cache_grpc.add_insecure_port(f"[::]:{self.cache_grpc_port}")
            cache_grpc.start()
            self._grpc_servers.append(cache_grpc)

        # proxy tier (only with service discovery — main.go:103-105)
        self.discovery = create_discovery_service(self.cfg, self.is_healthy)
        proxy_apps = []
        if self.discovery is not None:
            replicas = self.cfg.get_int("proxy.replicasPerModel") or 1
            self.cluster = ClusterConnection(self.discovery, replicas)
            host = self.cfg.get_string("proxy.advertiseHost") or \
                socket.gethostbyname(socket.gethostname())
            self._self_service = ServingService(
                host, self.cache_rest_port, self.cache_grpc_port,
                slot=self._slot)
            self.cluster.connect(self._self_service)
            if self._plane_enabled:
                self._wire_replica_plane(replicas)
            # warm handoff on ring changes: when membership shifts and a
            # locally-cached model's ownership moves away from this node,
            # nudge a new owner to load it NOW (a GetModelMetadata call —
            # existing wire surface, triggers its ensure_loaded) instead
            # of serving the first request cold. The reference relies on
            # the natural cache miss. Disable: proxy.warmHandoff: false.
            raw = (self.cfg.get_string("proxy.warmHandoff") or "").strip()
            if raw.lower() not in ("false", "0", "no"):
                self.discovery.add_listener(self._on_membership_change)

            def pick_rest(model, version):
                return self.cluster.node_for_key(model, version).rest_addr

            def pick_grpc(model, version):
                return self.cluster.node_for_key(model, version).grpc_addr

            from .tfservingproxy import GrpcForwarder
            fwd = GrpcForwarder(
                max_msg=self.cfg.get_int("serving.grpcMaxMsgSize") or
                16 * 1024 * 1024,
                timeout_s=self.cfg.get_float("proxy.grpcTimeout") or 10.0)
            proxy_grpc, _, self._fwd = make_proxy_grpc_server(
                pick_grpc, forwarder=fwd, health=self.health,
                max_msg=self.cfg.get_int("serving.grpcMaxMsgSize") or
                16 * 1024 * 1024)
            proxy_grpc.add_insecure_port(f"[::]:{self.proxy_grpc_port}")
            proxy_grpc.start()
            self._grpc_servers.append(proxy_grpc)
            proxy_apps.append((make_proxy_rest_app(
                pick_rest, metrics_path=metrics_path,
                metrics_render=merger.render,
                timeout_s=self.cfg.get_float("proxy.grpcTimeout") or 60.0),
                self.proxy_rest_port))
        else:
            log.info("Proxy is disabled (no serviceDiscovery.type)")

        # cache-tier REST: native (C++ HTTP/1.1 + JSON, registered
        # Predicts bypass Python) or the aiohttp app
        if self.cfg.get_bool("serving.nativeRestFrontend"):
            from .tfservingproxy.native_frontend import NativeRestServer
            cache_rest = NativeRestServer(self.handler,
                                          metrics_path=metrics_path,
                                          metrics_render=merger.render)
            cache_rest.add_insecure_port(f"[::]:{self.cache_rest_port}")
            cache_rest.start()
            self._grpc_servers.append(cache_rest)    # stop() plumbing
            self._native_frontends.append(cache_rest)
            apps = [] + proxy_apps
        else:
            cache_app = make_cache_rest_app(self.handler,
                                            metrics_path=metrics_path,
                                            metrics_render=merger.render)
            apps = [(cache_app, self.cache_rest_port)] + proxy_apps
        if self._native_frontends:
            self._wire_native_registry(self._native_frontends)

        # REST servers on a dedicated asyncio loop thread
        t = threading.Thread(target=self._run_rest, args=(apps,),
                             daemon=True)
        t.start()

        # health loop (30s — main.go:35-42)
        threading.Thread(target=self._health_loop, daemon=True).start()

    def _run_rest(self, apps) -> None:
        self._loop = asyncio.new_event_loop()
        asyncio.set_event_loop(self._loop)

        async def boot():
            for app, port in apps:
                runner = web.AppRunner(app)
                await runner.setup()
                site = web.TCPSite(runner, "0.0.0.0", port)
                await site.start()
                self._runners.append(runner)
        self._loop.run_until_complete(boot())
        self._loop.run_forever()
        # stop() requested: drain callbacks and release the loop's
        # resources (unclosed-loop ResourceWarning otherwise)
        try:
            self._loop.run_until_complete(asyncio.sleep(0))
            self._loop.close()
        except RuntimeError:
            pass

    def _wire_replica_plane(self, replicas: int) -> None:
        """Cold-load fan-out: after this rank fetches a model from the
        provider, push its bytes over the RCCL/xGMI plane to the other
        owner slots of the ring; receivers register the files in their
        LRU so their first request skips the provider."""
        from .cachemanager.lrucache import Model as CacheModel
        from .parallel.plane_service import PlaneService
        from .taskhandler.cluster import model_key

        cache = self.cm.cache

        def on_receive(name, version, vdir, total):
            if not cache.contains(name, version):
                cache.put(CacheModel(
                    name=name, version=version,
                    path=os.path.join(name, str(version)),
                    size_on_disk=total))

        self._plane = PlaneService(cache.base_dir,
                                   on_receive=on_receive)
        self._plane.announce_member(self._self_service.serialize())
        if replicas <= 1:
            return
        self_id = self._self_service.serialize()

        def on_cold_load(name, version):
            owners = self.cluster.find_nodes_for_key(
                model_key(name, version))
            ids = [o.serialize() for o in owners]
            if self_id not in ids:
                return                  # serving off-ring; no fan-out
            dsts = []
            for oid in ids:
                if oid == self_id:
                    continue
                r = self._plane.rank_of_member(oid)
                if r is not None:
                    dsts.append(r)
            if dsts:
                vdir = os.path.join(cache.base_dir, name, str(version))
                self._plane.push_files_async(name, version, vdir, dsts)

        self.cm.on_cold_load = on_cold_load

    def _wire_native_registry(self, natives) -> None:
        """Pool lifecycle -> C++ front-end FastModel registries (the
        native gRPC and native REST servers share one model pool)."""
        pool = self.cm.pool

        def on_available(name, version, model):
            fast = getattr(getattr(model, "_gpu", None), "_fast", None)
            if fast is not None:
                for fe in natives:
                    fe.register_model(name, version, fast)

        def on_unload(name, version, model):
            for fe in natives:
                fe.unregister_model(name, version)

        pool.on_available = on_available
        pool.on_unload = on_unload

    # -- warm handoff ------------------------------------------------------
    def _on_membership_change(self, members) -> None:
        ids = sorted(mm.serialize() for mm in members)
        prev, self._last_member_ids = self._last_member_ids, ids
        if prev is None or prev == ids or self._stop.is_set():
            return                         # initial seed / no-op update
        threading.Thread(target=self._warm_handoff, daemon=True).start()

    def _warm_handoff(self) -> None:
        from .taskhandler.cluster import model_key
        from .wire import messages as wm
        if self._handoff_fwd is None:
            from .tfservingproxy import GrpcForwarder
            self._handoff_fwd = GrpcForwarder(
                timeout_s=self.cfg.get_float("proxy.grpcTimeout") or 10.0)
        self_id = self._self_service.serialize()
        for entry in self.cm.list_cached():
            if self._stop.is_set():
                return
            try:
                owners = self.cluster.find_nodes_for_key(
                    model_key(entry.name, entry.version))
                if not owners or self_id in [o.serialize()
                                             for o in owners]:
                    continue
                # byte push first: if the new owner is a replica-plane
                # member (one-process-per-GPU deployment), send the
                # model FILES over RCCL/xGMI — it then loads from its
                # own disk cache without touching the model store
                # (ROADMAP round-1 item 7's byte-push candidate)
                pushed = False
                if self._plane is not None:
                    dsts = []
                    for o in owners:
                        r = self._plane.rank_of_member(o.serialize(),
                                                       timeout_s=0.5)
                        if r is not None and r != self._plane.rank:
                            dsts.append(r)
                    if dsts:
                        vdir = os.path.join(
                            self.cm.cache.base_dir, entry.name,
                            str(entry.version))
                        self._plane.push_files_async(
                            entry.name, entry.version, vdir, dsts)
                        pushed = True
                        log.info("warm handoff (plane): %s:%s -> ranks "
                                 "%s", entry.name, entry.version, dsts)
                # nudge the owner to LOAD now (from the pushed bytes if
                # the plane delivered, from the store otherwise)
                req = wm.GetModelMetadataRequest(
                    model_spec=wm.ModelSpec(
                        name=entry.name,
                        version=wm.Int64Value(value=entry.version)),
                    metadata_field=["signature_def"])
                self._handoff_fwd.call(
                    owners[0].grpc_addr,
                    f"/{wm.PREDICTION_SERVICE}/GetModelMetadata",
                    req.encode())
                log.info("warm handoff: %s:%s -> %s%s", entry.name,
                         entry.version, owners[0].grpc_addr,
                         " (bytes via plane)" if pushed else "")
            except Exception:       # noqa: BLE001
                log.warning("warm handoff failed for %s:%s",
                            entry.name, entry.version, exc_info=True)

    def _health_loop(self) -> None:
        while not self._stop.wait(30.0):
            try:
                self.health.set_serving(self.is_healthy())
            except Exception:       # noqa: BLE001
                self.health.set_serving(False)

    def stop(self) -> None:
        self._stop.set()
        if self._plane is not None:
            self._plane.stop()
        if self.cluster is not None:
            self.cluster.disconnect()
        for s in self._grpc_servers:
            s.stop(grace=1.0)
        if self._loop is not None:
            async def shutdown():
                for r in self._runners:
                    await r.cleanup()
            coro = shutdown()
            try:
                if self._loop.is_running() and not self._loop.is_closed():
                    fut = asyncio.run_coroutine_threadsafe(coro, self._loop)
                    coro = None
                    try:
                        fut.result(timeout=5)
                    except Exception:       # noqa: BLE001
                        pass
            except RuntimeError:
                pass                # loop died between check and schedule
            finally:
                if coro is not None:
                    coro.close()    # never scheduled: silence the
                                    # "never awaited" RuntimeWarning
            try:
                self._loop.call_soon_threadsafe(self._loop.stop)
            except RuntimeError:
                pass
        self.cm.close()


def main() -> int:
    cfg = Config.load()
    from .utils.logsetup import setup_logging
    setup_logging(cfg)
    server = Server(cfg)
    server.start()
    log.info("tfservingcache-amd serving: cacheRest=%d cacheGrpc=%d",
             server.cache_rest_port, server.cache_grpc_port)
    try:
        while True:
            time.sleep(3600)
    except KeyboardInterrupt:
        server.stop()
    return 0


if __name__ == "__main__":
    raise SystemExit(main())
