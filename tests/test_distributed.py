"""Multi-process distributed tests (gloo backend, CPU, world_size=2).

Covers the distributed paths the driver exercises at round end on 8
GPUs: ring-consistent model ownership across ranks and the replica
plane's stage-once/fan-out (RCCL broadcast on GPU; gloo here).
"""
import json
import os
import subprocess
import sys
import tempfile
from pathlib import Path

import pytest

REPO = Path(__file__).resolve().parent.parent

WORKER = r"""
import json, os, sys
sys.path.insert(0, os.environ["TFSC_REPO"])
import torch.distributed as dist

from tfservingcache_amd.models import write_model_repo
from tfservingcache_amd.cachemanager import LRUCache
from tfservingcache_amd.cachemanager.providers import DiskModelProvider
from tfservingcache_amd.parallel import ReplicaPlane
from tfservingcache_amd.taskhandler import ConsistentHashRing, model_key

def main():
    rank = int(os.environ["RANK"])
    world = int(os.environ["WORLD_SIZE"])
    dist.init_process_group("gloo")
    shared = os.environ["TFSC_SHARED"]

    # rank 0 has the provider repo; replicas receive over the plane
    repo = os.path.join(shared, "repo")
    if rank == 0:
        write_model_repo(repo, [("mlp_a", 1, "mlp"), ("mlp_b", 1, "mlp"),
                                ("mlp_c", 2, "mlp")])
    dist.barrier()

    cache_dir = os.path.join(shared, f"cache_r{rank}")
    cache = LRUCache(cache_dir, 10**8)
    provider = DiskModelProvider(repo)

    # ring over ranks: every rank must agree on ownership
    ring = ConsistentHashRing()
    ring.set_members([f"rank{r}" for r in range(world)])
    models = [("mlp_a", 1), ("mlp_b", 1), ("mlp_c", 2)]

    def owners_of(name, version):
        mems = ring.get_n(model_key(name, version), 2)
        return [int(m[4:]) for m in mems]

    plane = ReplicaPlane()

    def fetch_local(name, version):
        entry = provider.load_model(name, version, cache.base_dir)
        cache.put(entry)

    mine = plane.preload_replicated(models, owners_of, cache.base_dir,
                                    fetch_local)

    # verify: every owned model dir exists and loads
    from tfservingcache_amd.engine.model import load_model_from_dir
    import numpy as np
    ok = []
    for name, version in mine:
        vdir = os.path.join(cache.base_dir, name, str(version))
        assert os.path.exists(os.path.join(vdir, "saved_model.pb")), vdir
        lm = load_model_from_dir(vdir, name, version)
        out = lm.predict({"x": np.zeros((1, 16), dtype=np.float32)})
        assert out["probs"].shape == (1, 8)
        ok.append(f"{name}:{version}")

    all_owned = [None] * world
    dist.all_gather_object(all_owned, sorted(ok))
    if rank == 0:
        print(json.dumps({"owned": all_owned}))
    dist.barrier()
    dist.destroy_process_group()

main()
"""


def test_replica_plane_two_ranks(tmp_path):
    script = tmp_path / "worker.py"
    script.write_text(WORKER)
    shared = tmp_path / "shared"
    shared.mkdir()
    env = dict(os.environ)
    env.update({
        "TFSC_REPO": str(REPO),
        "TFSC_SHARED": str(shared),
        "MASTER_ADDR": "127.0.0.1",
        "MASTER_PORT": "29611",
        "WORLD_SIZE": "2",
    })
    procs = []
    for rank in range(2):
        e = dict(env)
        e["RANK"] = str(rank)
        e["LOCAL_RANK"] = str(rank)
        procs.append(subprocess.Popen(
            [sys.executable, str(script)], env=e,
            stdout=subprocess.PIPE, stderr=subprocess.PIPE))
    outs = []
    for p in procs:
        out, err = p.communicate(timeout=180)
        assert p.returncode == 0, err.decode()[-3000:]
        outs.append(out.decode())
    payload = json.loads(outs[0].strip().splitlines()[-1])
    owned = payload["owned"]
    # with 3 models x 2 replicas over 2 ranks, both ranks own all 3
    assert sorted(owned[0]) == sorted(owned[1])
    assert len(owned[0]) == 3
