"""CPU-testable concurrency pieces of the GPU cold-load path: the
two-mode capture/upload guard (pure threading) and the plane service's
store plumbing (world_size=1, gloo)."""
import os
import socket
import threading
import time

import pytest

pytest.importorskip("torch")


def _mk_guard():
    from tfservingcache_amd.engine.gpu import _CaptureGuard
    g = _CaptureGuard()
    g.enabled = True
    return g


def test_guard_concurrent_same_mode():
    g = _mk_guard()
    inside = []
    barrier = threading.Barrier(4, timeout=10)

    def cap():
        with g.capture():
            inside.append(1)
            barrier.wait()          # all 4 captures inside AT ONCE

    threads = [threading.Thread(target=cap) for _ in range(4)]
    for t in threads:
        t.start()
    for t in threads:
        t.join(timeout=10)
    assert len(inside) == 4


def test_guard_mutual_exclusion_and_writer_preference():
    g = _mk_guard()
    log = []
    cap_started = threading.Event()
    release_cap = threading.Event()

    def long_capture():
        with g.capture():
            cap_started.set()
            release_cap.wait(10)
            log.append("cap_end")

    def upload():
        with g.unsafe_host_op():
            log.append("upload")

    late_cap_done = threading.Event()

    def late_capture():
        with g.capture():
            log.append("late_cap")
            late_cap_done.set()

    t1 = threading.Thread(target=long_capture)
    t1.start()
    cap_started.wait(10)
    t2 = threading.Thread(target=upload)
    t2.start()
    time.sleep(0.1)                 # upload now QUEUED behind t1
    t3 = threading.Thread(target=late_capture)
    t3.start()
    time.sleep(0.1)
    # writer preference: the late capture must NOT start while an
    # upload is queued
    assert "late_cap" not in log
    release_cap.set()
    for t in (t1, t2, t3):
        t.join(timeout=10)
    # upload ran before the late capture
    assert log.index("upload") < log.index("late_cap")


def test_guard_disabled_is_noop():
    from tfservingcache_amd.engine.gpu import _CaptureGuard
    g = _CaptureGuard()
    g.enabled = False
    with g.capture():
        with g.unsafe_host_op():    # would deadlock if enforced
            pass


def test_plane_service_world1(tmp_path):
    """Store plumbing + plane thread on a single gloo rank: member
    mapping round-trips, a push with no remote destinations is a clean
    no-op, and stop() terminates the thread."""
    import torch.distributed as dist
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=0, world_size=1)
    try:
        from tfservingcache_amd.parallel.plane_service import PlaneService
        got = []
        plane = PlaneService(str(tmp_path),
                             on_receive=lambda *a: got.append(a))
        plane.announce_member("host:1:2:gpu0")
        assert plane.rank_of_member("host:1:2:gpu0") == 0
        assert plane.rank_of_member("nope", timeout_s=0.2) is None
        # no remote dsts -> no-op, returns immediately
        (tmp_path / "m" / "1").mkdir(parents=True)
        (tmp_path / "m" / "1" / "f").write_bytes(b"x" * 128)
        plane.push_files("m", 1, str(tmp_path / "m" / "1"), [0])
        plane.stop()
        assert not plane._thread.is_alive()
        assert got == []
    finally:
        dist.destroy_process_group()


def test_plane_metrics_on_scrape_endpoint():
    """The replica-plane counters must be in the SERVED registry (a
    counter on prometheus_client's default registry would silently
    vanish from /monitoring)."""
    from tfservingcache_amd.utils import metrics as mt
    mt.plane_transfers.labels("send").inc(0)
    mt.plane_bytes.labels("send").inc(0)
    text = mt.render().decode()
    assert "tfservingcache_replica_plane_transfers_total" in text
    assert "tfservingcache_replica_plane_bytes_total" in text
