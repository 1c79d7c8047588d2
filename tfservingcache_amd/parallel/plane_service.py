"""Replica-plane service: on-demand model byte push between serving
processes over RCCL/xGMI.

The library-level `ReplicaPlane` (replica.py) does COLLECTIVE preloads —
every rank walks the model list together. A production server can't do
that: cold loads happen on whatever rank a request lands on, whenever it
lands. This service makes the fan-out demand-driven:

  * every serving process (one per GPU, torch.distributed initialized,
    backend "nccl" == RCCL on ROCm / "gloo" on CPU) runs ONE plane
    thread;
  * transfers are published to a global command log in the c10d store
    (seq = store.add) and executed by every involved rank in global
    sequence order — a total order, so concurrent pushes between
    overlapping rank sets cannot deadlock;
  * the bytes move as one flat uint8 tensor via point-to-point
    dist.send/recv — on the nccl backend that is an RCCL p2p transfer
    over the direct xGMI link between the two GPUs, never through the
    model store;
  * the receiver writes the SavedModel files into its own disk cache and
    fires `on_receive` so the CacheManager registers them (the next
    request for the model on that rank skips the provider entirely).

Used by main.py when `proxy.replicasPerModel > 1` and the server runs
one-process-per-GPU (WORLD_SIZE > 1): the rank that cold-loads a model
pushes its files to the other owner slots of the ring
(reference behavior being replaced: every replica independently
downloads from the store, pkg/cachemanager/cachemanager.go:122).
"""
from __future__ import annotations

import json
import logging
import os
import threading
from datetime import timedelta
from typing import Callable, Dict, List, Optional, Sequence

log = logging.getLogger("tfsc.plane")


class PlaneService:
    POLL_S = 0.25

    def __init__(self, cache_base_dir: str,
                 on_receive: Optional[Callable[[str, int, str, int],
                                               None]] = None,
                 device: Optional[str] = None):
        import torch
        import torch.distributed as dist
        if not dist.is_initialized():
            raise RuntimeError("torch.distributed is not initialized")
        self.torch = torch
        self.dist = dist
        self.rank = dist.get_rank()
        self.world = dist.get_world_size()
        self.backend = dist.get_backend()
        self.device = device if self.backend == "nccl" else "cpu"
        self.cache_base_dir = cache_base_dir
        self.on_receive = on_receive
        from torch.distributed.distributed_c10d import _get_default_store
        from torch.distributed import PrefixStore
        self._store = PrefixStore("tfsc_plane/", _get_default_store())
        self._stop = threading.Event()
        self._waiters: Dict[int, threading.Event] = {}
        self._errors: Dict[int, str] = {}
        self._wlock = threading.Lock()
        self._next = 1
        self._thread = threading.Thread(target=self._run, daemon=True,
                                        name="tfsc-plane")
        self._thread.start()

    # -- member <-> rank mapping ------------------------------------------
    def announce_member(self, member_id: str) -> None:
        """Publish this process's ring-member identity so other ranks can
        address it (member serialize() -> rank)."""
        self._store.set(f"member/{member_id}", str(self.rank))

    def rank_of_member(self, member_id: str,
                       timeout_s: float = 5.0) -> Optional[int]:
        try:
            self._store.wait([f"member/{member_id}"],
                             timedelta(seconds=timeout_s))
            return int(self._store.get(f"member/{member_id}"))
        except Exception:       # noqa: BLE001
            return None

    # -- publishing --------------------------------------------------------
    def push_files(self, name: str, version: int, version_dir: str,
                   dst_ranks: Sequence[int],
                   timeout_s: float = 120.0) -> None:
        """Push the SavedModel files under version_dir to dst_ranks'
        disk caches. Blocks until this rank's send completes."""
        dsts = sorted(set(int(d) for d in dst_ranks) - {self.rank})
        if not dsts:
            return
        entries: List = []
        for root, _dirs, files in os.walk(version_dir):
            for f in sorted(files):
                full = os.path.join(root, f)
                rel = os.path.relpath(full, version_dir)
                entries.append([rel, os.path.getsize(full)])
        cmd = {"name": name, "version": int(version),
               "src": self.rank, "dsts": dsts, "entries": entries,
               "_local_dir": version_dir}   # src-local source path
        seq = self._store.add("seq", 1)
        ev = threading.Event()
        with self._wlock:
            self._waiters[seq] = ev
        self._store.set(f"cmd/{seq}", json.dumps(cmd))
        if not ev.wait(timeout_s):
            raise TimeoutError(
                f"plane push of {name}:{version} to {dsts} timed out")
        err = self._errors.pop(seq, None)
        if err:
            raise RuntimeError(f"plane push failed: {err}")

    def push_files_async(self, name: str, version: int, version_dir: str,
                         dst_ranks: Sequence[int]) -> None:
        t = threading.Thread(
            target=self._push_logged, daemon=True,
            args=(name, version, version_dir, dst_ranks))
        t.start()

    def _push_logged(self, name, version, version_dir, dst_ranks):
        try:
            self.push_files(name, version, version_dir, dst_ranks)
        except Exception:       # noqa: BLE001
            log.exception("replica push failed for %s:%s", name, version)

    # -- the plane thread --------------------------------------------------
    def _run(self) -> None:
        while not self._stop.is_set():
            key = f"cmd/{self._next}"
            try:
                self._store.wait([key], timedelta(seconds=self.POLL_S))
            except Exception:   # timeout — re-check stop flag
                continue
            try:
                cmd = json.loads(self._store.get(key))
            except Exception:   # noqa: BLE001
                log.exception("bad plane command at seq %d", self._next)
                self._next += 1
                continue
            seq = self._next
            self._next += 1
            try:
                self._execute(cmd)
            except Exception as e:      # noqa: BLE001
                log.exception("plane transfer failed (seq %d)", seq)
                with self._wlock:
                    if seq in self._waiters:
                        self._errors[seq] = str(e)
            finally:
                with self._wlock:
                    ev = self._waiters.pop(seq, None)
                if ev is not None:
                    ev.set()

    def _execute(self, cmd: dict) -> None:
        torch, dist = self.torch, self.dist
        src = int(cmd["src"])
        dsts = [int(d) for d in cmd["dsts"]]
        if self.rank != src and self.rank not in dsts:
            return
        entries = cmd["entries"]
        total = sum(int(size) for _rel, size in entries)
        name, version = cmd["name"], int(cmd["version"])
        vdir = os.path.join(self.cache_base_dir, name, str(version))
        buf = torch.empty(max(total, 1), dtype=torch.uint8,
                          device=self.device)
        if self.rank == src:
            off = 0
            src_dir = cmd.get("_local_dir") or vdir
            for rel, size in entries:
                with open(os.path.join(src_dir, rel), "rb") as f:
                    data = f.read()
                buf[off:off + int(size)] = torch.frombuffer(
                    bytearray(data), dtype=torch.uint8)
                off += int(size)
            for d in dsts:
                dist.send(buf, dst=d)
            from ..utils import metrics as mt
            mt.plane_transfers.labels("send").inc(len(dsts))
            mt.plane_bytes.labels("send").inc(total * len(dsts))
        else:
            dist.recv(buf, src=src)
            host = buf.cpu().numpy().tobytes()
            off = 0
            for rel, size in entries:
                path = os.path.join(vdir, rel)
                os.makedirs(os.path.dirname(path), exist_ok=True)
                tmp = path + ".planetmp"
                with open(tmp, "wb") as f:
                    f.write(host[off:off + int(size)])
                os.replace(tmp, path)
                off += int(size)
            log.info("plane: received %s:%s (%d files, %.1f MB) from "
                     "rank %d", name, version, len(entries),
                     total / 1e6, src)
            from ..utils import metrics as mt
            mt.plane_transfers.labels("recv").inc()
            mt.plane_bytes.labels("recv").inc(total)
            if self.on_receive is not None:
                try:
                    self.on_receive(name, version, vdir, total)
                except Exception:       # noqa: BLE001
                    log.exception("plane on_receive failed")

    def stop(self) -> None:
        self._stop.set()
        self._thread.join(timeout=2 * self.POLL_S + 1)
