"""Production multi-GPU wiring (CPU/gloo stand-in, world_size=2).

VERDICT round-1 item 2: N server PROCESSES (one per GPU in production)
must form an N-slot ring with slot-tagged members, and when
proxy.replicasPerModel > 1 a cold load on one slot must push the model
bytes to the other replica slot over the RCCL plane — the replica's
first request then never touches the model provider
(reference contrast: every replica re-downloads from the store,
pkg/cachemanager/cachemanager.go:122).
"""
import json
import os
import socket
import subprocess
import sys
import tempfile
import time
from pathlib import Path

import pytest

REPO = Path(__file__).resolve().parent.parent

WORKER = r"""
import json, os, sys, time
sys.path.insert(0, os.environ["TFSC_REPO"])

import numpy as np
import torch.distributed as dist

from tfservingcache_amd.config import Config
from tfservingcache_amd.main import Server
from tfservingcache_amd.models import write_model_repo
from tfservingcache_amd.cachemanager.providers import DiskModelProvider
from tfservingcache_amd.taskhandler.cluster import model_key
from tfservingcache_amd.wire import messages as m
from tfservingcache_amd.wire.tensor import numpy_to_tensorproto

import tfservingcache_amd.main as main_mod


class CountingDisk(DiskModelProvider):
    calls = 0
    def load_model(self, name, version, dest):
        type(self).calls += 1
        return super().load_model(name, version, dest)


def main():
    rank = int(os.environ["RANK"])
    shared = os.environ["TFSC_SHARED"]
    base = int(os.environ["TFSC_BASE_PORT"])

    repo = os.path.join(shared, "repo")
    if rank == 0:
        write_model_repo(repo, [("mlp_a", 1, "mlp"), ("mlp_b", 1, "mlp"),
                                ("mlp_c", 1, "mlp")])

    # patch the provider factory so we can count provider fetches
    main_mod.create_model_provider = lambda cfg: CountingDisk(repo)

    cfg = Config({
        "cacheRestPort": base, "cacheGrpcPort": base + 1,
        "proxyRestPort": base + 2, "proxyGrpcPort": base + 3,
        "modelCache": {"hostModelPath":
                       os.path.join(shared, f"cache_r{rank}"),
                       "size": 10 ** 9},
        "serving": {"maxConcurrentModels": 4},
        "proxy": {"replicasPerModel": 2, "advertiseHost": "127.0.0.1",
                  "warmHandoff": False},
        "engine": {"portStride": 10},
        "serviceDiscovery": {
            "type": "file",
            "heartbeatTTL": 1.0,
            "file": {"directory": os.path.join(shared, "disc")}},
    })
    srv = Server(cfg)
    srv.start()
    out = {"rank": rank}
    try:
        # both slot-tagged members must join the ring
        deadline = time.time() + 20
        while srv.cluster.n_members() < 2 and time.time() < deadline:
            time.sleep(0.1)
        out["n_members"] = srv.cluster.n_members()
        members = srv.cluster.ring.members()
        out["slots"] = sorted(mm.split(":")[3] for mm in members)

        dist.barrier()

        # pick a model and let its FIRST owner slot cold-load it
        owners = srv.cluster.find_nodes_for_key(model_key("mlp_a", 1))
        out["n_owners"] = len(owners)
        owner_ids = [o.serialize() for o in owners]
        self_id = srv._self_service.serialize()
        i_am_first = owner_ids and owner_ids[0] == self_id
        if i_am_first:
            img = np.ones((1, 16), dtype=np.float32)
            req = m.PredictRequest(
                model_spec=m.ModelSpec(
                    name="mlp_a", version=m.Int64Value(value=1)),
                inputs={"x": numpy_to_tensorproto(img)}).encode()
            resp = srv.handler.predict_bytes(req)
            out["first_predict_ok"] = len(resp) > 0
            out["loads_after_first"] = CountingDisk.calls
        dist.barrier()

        if not i_am_first:
            # replica slot: the bytes must ARRIVE over the plane
            deadline = time.time() + 30
            while not srv.cm.cache.contains("mlp_a", 1) and \
                    time.time() < deadline:
                time.sleep(0.1)
            out["replica_has_files"] = srv.cm.cache.contains("mlp_a", 1)
            calls_before = CountingDisk.calls
            img = np.ones((1, 16), dtype=np.float32)
            req = m.PredictRequest(
                model_spec=m.ModelSpec(
                    name="mlp_a", version=m.Int64Value(value=1)),
                inputs={"x": numpy_to_tensorproto(img)}).encode()
            resp = srv.handler.predict_bytes(req)
            out["replica_predict_ok"] = len(resp) > 0
            out["replica_provider_calls"] = \
                CountingDisk.calls - calls_before
        dist.barrier()
    finally:
        with open(os.path.join(shared, f"out_r{rank}.json"), "w") as f:
            json.dump(out, f)
        srv.stop()
    dist.destroy_process_group()


main()
"""


@pytest.mark.gpu
def test_plane_nccl_initializes_single_gpu(tmp_path, monkeypatch):
    """The RCCL path of the production plane must initialize on a real
    GPU (world_size=1: store plumbing + plane thread + member mapping;
    the 2-rank transfer itself is covered by the gloo test above and
    the driver's 8-GPU run)."""
    import torch
    import torch.distributed as dist
    if not torch.cuda.is_available():
        pytest.skip("needs GPU")
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    monkeypatch.setenv("MASTER_ADDR", "127.0.0.1")
    monkeypatch.setenv("MASTER_PORT", str(port))
    monkeypatch.setenv("RANK", "0")
    monkeypatch.setenv("WORLD_SIZE", "1")
    dist.init_process_group("nccl", rank=0, world_size=1)
    try:
        # a real RCCL collective must run
        t = torch.ones(4, device="cuda:0")
        dist.all_reduce(t)
        assert t.cpu().tolist() == [1.0, 1.0, 1.0, 1.0]

        from tfservingcache_amd.parallel.plane_service import PlaneService
        got = []
        plane = PlaneService(str(tmp_path),
                             on_receive=lambda *a: got.append(a),
                             device="cuda:0")
        plane.announce_member("127.0.0.1:8094:8095:gpu0")
        assert plane.rank_of_member("127.0.0.1:8094:8095:gpu0") == 0
        assert plane.backend == "nccl"
        # push with no remote dsts is a clean no-op
        plane.push_files("m", 1, str(tmp_path), [0])
        plane.stop()
    finally:
        dist.destroy_process_group()


def _free_port_block(n=30):
    """Find a base port where the WHOLE block [base, base+n) binds —
    rank 1's ports are base+portStride.., so checking only the base
    left a bind-race window (one flaky run in ~30)."""
    import random
    for _ in range(50):
        base = random.randint(20000, 60000 - n)
        socks = []
        try:
            for off in range(n):
                s = socket.socket()
                s.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
                s.bind(("127.0.0.1", base + off))
                socks.append(s)
            return base
        except OSError:
            continue
        finally:
            for s in socks:
                s.close()
    return 29500  # last resort


def _spawn_two_slot_ring(tmp_path):
    os.makedirs(tmp_path, exist_ok=True)
    master_port = _free_port_block()
    base_port = _free_port_block()
    if abs(base_port - master_port) < 40:
        base_port = master_port + 200
    procs = []
    for rank in range(2):
        env = dict(os.environ,
                   RANK=str(rank), WORLD_SIZE="2",
                   LOCAL_RANK=str(rank),
                   MASTER_ADDR="127.0.0.1",
                   MASTER_PORT=str(master_port),
                   TFSC_REPO=str(REPO),
                   TFSC_SHARED=str(tmp_path),
                   TFSC_BASE_PORT=str(base_port))
        procs.append(subprocess.Popen(
            [sys.executable, "-c", WORKER], env=env,
            stdout=subprocess.PIPE, stderr=subprocess.STDOUT))
    outs = []
    for p in procs:
        stdout, _ = p.communicate(timeout=150)
        outs.append(stdout.decode(errors="replace"))
    return procs, outs


@pytest.mark.timeout(400)
def test_two_slot_ring_with_plane_fanout(tmp_path):
    # one retry: the free-port probe can't fully exclude a bind race
    # with concurrently-allocated ephemeral ports
    for attempt in range(2):
        procs, outs = _spawn_two_slot_ring(tmp_path / str(attempt))
        if all(p.returncode == 0 for p in procs):
            break
    for rank, (p, o) in enumerate(zip(procs, outs)):
        assert p.returncode == 0, \
            f"rank0 out:\n{outs[0][-2500:]}\nrank1 out:\n{outs[1][-2500:]}"
    tmp_path = tmp_path / str(attempt)

    results = {}
    for rank in range(2):
        with open(tmp_path / f"out_r{rank}.json") as f:
            results[rank] = json.load(f)

    for rank in range(2):
        r = results[rank]
        assert r["n_members"] == 2, r
        assert r["slots"] == ["gpu0", "gpu1"], r
        assert r["n_owners"] == 2, r

    first = [r for r in results.values() if "first_predict_ok" in r]
    replica = [r for r in results.values() if "replica_predict_ok" in r]
    assert len(first) == 1 and len(replica) == 1
    assert first[0]["first_predict_ok"]
    assert first[0]["loads_after_first"] == 1
    # the replica got the files via the plane and NEVER hit the provider
    assert replica[0]["replica_has_files"], replica[0]
    assert replica[0]["replica_predict_ok"]
    assert replica[0]["replica_provider_calls"] == 0
