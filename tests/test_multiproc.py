"""Multi-process scaling correctness: N registrar processes, shared-nothing
except the ZK ensemble (SURVEY §2.3 'multi-process axis'), plus a
torch.distributed (gloo) world_size=2 test of the bench coordination path."""
import json
import multiprocessing as mp
import os
import sys

import pytest

import registrar_amd as ra
from conftest import REPO_ROOT, wait_for


def _worker(connect, idx, q):
    sys.path.insert(0, REPO_ROOT)
    import registrar_amd as ra  # noqa: F811 (fresh import in child)

    servers = []
    for hp in connect.split(","):
        host, port = hp.rsplit(":", 1)
        servers.append((host, int(port)))
    c = ra.ZkClient(servers=servers, session_timeout_ms=10000)
    c.start()
    if not c.wait_connected(15000):
        q.put((idx, "connect-failed", []))
        return
    registration = {
        "domain": "proc%d.mp.test" % idx,
        "type": "host",
        "adminIp": "127.0.0.1",
        "hostname": "host%d" % idx,
        "settleMs": 0,
        "aliases": ["a%02d.proc%d.mp.test" % (i, idx) for i in range(9)],
    }
    rc, err, znodes = ra.register_node(c, json.dumps(registration))
    if rc != 0:
        q.put((idx, "register-failed: %s" % err, []))
        return
    rc, rtt = c.heartbeat(znodes)
    q.put((idx, "ok" if rc == 0 else "heartbeat-failed", znodes))
    c.close()


@pytest.mark.parametrize("nprocs", [4, 8])
def test_processes_share_ensemble(ensemble3, nprocs):
    # the 8-process axis of the north star, hermetically on CPU (the GPU
    # bench covers the same shape with timing)
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    connect = ensemble3.connect_string()
    procs = [ctx.Process(target=_worker, args=(connect, i, q)) for i in range(nprocs)]
    for p in procs:
        p.start()
    results = [q.get(timeout=120) for _ in procs]
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0
    assert all(status == "ok" for _, status, _ in results), results
    all_nodes = [n for _, _, zn in results for n in zn]
    assert len(all_nodes) == nprocs * 10
    assert len(set(all_nodes)) == nprocs * 10  # no collisions across processes
    # processes have exited ⇒ sessions close ⇒ ephemerals vanish
    assert wait_for(lambda: ensemble3.ephemeral_count() == 0, timeout=20)


def _dist_worker(rank, world, port, connect, q):
    sys.path.insert(0, REPO_ROOT)
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    import torch.distributed as dist

    import registrar_amd as ra  # noqa: F811

    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        # rank 0 broadcasts the ensemble connect string (the bench.py pattern)
        obj = [connect if rank == 0 else None]
        dist.broadcast_object_list(obj, src=0)
        servers = []
        for hp in obj[0].split(","):
            host, p = hp.rsplit(":", 1)
            servers.append((host, int(p)))
        c = ra.ZkClient(servers=servers, session_timeout_ms=10000)
        c.start()
        assert c.wait_connected(15000)
        registration = {
            "domain": "rank%d.dist.test" % rank,
            "type": "host",
            "adminIp": "127.0.0.1",
            "hostname": "r%d" % rank,
            "settleMs": 0,
        }
        rc, err, znodes = ra.register_node(c, json.dumps(registration))
        assert rc == 0, err
        dist.barrier()
        rc, rtt = c.heartbeat(znodes)
        assert rc == 0
        dist.barrier()
        c.close()
        q.put((rank, "ok"))
    finally:
        dist.destroy_process_group()


def test_torch_distributed_gloo_two_ranks(ensemble):
    torch = pytest.importorskip("torch")
    del torch
    from conftest import free_port

    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = free_port()
    connect = ensemble.connect_string()
    procs = [ctx.Process(target=_dist_worker, args=(r, 2, port, connect, q)) for r in range(2)]
    for p in procs:
        p.start()
    results = [q.get(timeout=120) for _ in procs]
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0
    assert sorted(r for r, _ in results) == [0, 1]
    assert all(s == "ok" for _, s in results)


def test_bench_partition_cpus(monkeypatch):
    """bench.partition_cpus: rank slices are disjoint, within the quota, and
    the ensemble slice is the remainder — for every rank of several worlds
    (this partition is what makes the driver-measured scaling curve monotone
    under a CFS quota)."""
    import importlib.util

    spec = importlib.util.spec_from_file_location("bench", os.path.join(REPO_ROOT, "bench.py"))
    bench = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(bench)

    cpus = set(range(16))
    monkeypatch.setattr(os, "sched_getaffinity", lambda pid: cpus)
    monkeypatch.setattr(bench, "read_cpu_quota", lambda: 16.0)
    monkeypatch.delenv("BENCH_AFFINITY", raising=False)
    monkeypatch.delenv("BENCH_RANK_CPUS", raising=False)

    for world in (2, 4, 8):
        slices = []
        ens_ref = None
        for rank in range(world):
            my, ens, eff = bench.partition_cpus(world, rank)
            assert eff == 16
            assert my and ens, "partition expected for world=%d" % world
            assert my.isdisjoint(ens)
            slices.append(my)
            ens_ref = ens
        # all rank slices pairwise disjoint and inside the quota
        seen = set()
        for s in slices:
            assert not (s & seen)
            seen |= s
        assert seen | ens_ref <= cpus
        assert len(ens_ref) >= 2

    # world=1: no pinning by default
    assert bench.partition_cpus(1, 0)[:2] == (None, None)
    # BENCH_AFFINITY=0 disables
    monkeypatch.setenv("BENCH_AFFINITY", "0")
    assert bench.partition_cpus(8, 0)[:2] == (None, None)
    monkeypatch.delenv("BENCH_AFFINITY")
    # BENCH_RANK_CPUS override honored
    monkeypatch.setenv("BENCH_RANK_CPUS", "1")
    my, ens, _ = bench.partition_cpus(8, 3)
    assert my == {3} and ens == set(range(8, 16))
    # too few CPUs for the requested split: graceful no-pinning
    monkeypatch.setenv("BENCH_RANK_CPUS", "4")
    assert bench.partition_cpus(8, 0)[:2] == (None, None)
    monkeypatch.delenv("BENCH_RANK_CPUS")
    # tight quota (world+2 > cpus): no pinning rather than starving ranks
    monkeypatch.setattr(os, "sched_getaffinity", lambda pid: set(range(4)))
    monkeypatch.setattr(bench, "read_cpu_quota", lambda: 4.0)
    assert bench.partition_cpus(4, 0)[:2] == (None, None)
