"""ZooKeeper multi (transactions) and the atomic-swap registration mode.

The atomic swap is a beyond-the-reference feature: re-registration replaces
stale znodes and creates the new set in one transaction, so consumers never
observe a partially-registered domain (the gap the reference's 1 s settle
delay papered over)."""
import json
import threading
import time

import pytest

import registrar_amd as ra
from conftest import make_client, orch_config, wait_for


def test_multi_applies_atomically(ensemble, client):
    client.mkdirp("/m")
    rc, per_op = client.multi([
        ("create", "/m/a", b"1", False),
        ("create", "/m/b", b"2", True),
    ])
    assert rc == ra.ZOK and per_op == [ra.ZOK, ra.ZOK]
    assert ensemble.get("/m/a")["exists"] and ensemble.get("/m/b")["exists"]
    assert ensemble.get("/m/b")["stat"]["ephemeralOwner"] == client.session_id()


def test_multi_delete_then_recreate_same_path(ensemble, client):
    client.mkdirp("/m2")
    client.create("/m2/x", b"old", True)
    rc, per_op = client.multi([
        ("delete", "/m2/x"),
        ("create", "/m2/x", b"new", True),
    ])
    assert rc == ra.ZOK
    assert ensemble.get("/m2/x")["data"] == b"new"
    assert ensemble.ephemeral_count() == 1  # no leak from the replaced node


def test_multi_aborts_all_on_failure(ensemble, client):
    client.mkdirp("/m3")
    client.create("/m3/exists", b"")
    rc, per_op = client.multi([
        ("create", "/m3/new", b"x", False),
        ("create", "/m3/exists", b"y", False),  # fails: NODE_EXISTS
        ("delete", "/m3/new"),
    ])
    assert rc == ra.ZNODEEXISTS
    assert per_op[1] == ra.ZNODEEXISTS
    assert not ensemble.get("/m3/new")["exists"]  # first op rolled back
    assert ensemble.get("/m3/exists")["data"] == b""  # untouched


def test_multi_parent_rules(ensemble, client):
    rc, per_op = client.multi([("create", "/noparent/child", b"", False)])
    assert rc == ra.ZNONODE
    # ephemeral parent created IN the txn: children must be rejected
    client.mkdirp("/m4")
    rc, per_op = client.multi([
        ("create", "/m4/eph", b"", True),
        ("create", "/m4/eph/kid", b"", False),
    ])
    assert rc == ra.ZNOCHILDRENFOREPHEMERALS
    assert not ensemble.get("/m4/eph")["exists"]


def test_multi_fires_watches_only_on_success(ensemble, client):
    other = make_client(ensemble)
    client.mkdirp("/m5")
    other.get_children("/m5", watch=True)
    rc, _ = client.multi([
        ("create", "/m5/a", b"", False),
        ("create", "/m5/a", b"", False),  # NODE_EXISTS within the txn
    ])
    assert rc == ra.ZNODEEXISTS
    time.sleep(0.2)
    assert not any(w["path"] == "/m5" for w in other.poll_watches())
    rc, _ = client.multi([("create", "/m5/b", b"", False)])
    assert rc == ra.ZOK
    assert wait_for(lambda: any(w["path"] == "/m5" for w in other.poll_watches()), 5)
    other.close()


REG = {
    "domain": "swap.test",
    "type": "host",
    "adminIp": "127.0.0.1",
    "hostname": "s0",
    "settleMs": 0,
    "atomicSwap": True,
    "aliases": ["a%02d.swap.test" % i for i in range(19)],
}


def test_atomic_swap_registers(ensemble, client):
    prep = ra.PreparedRegistration(json.dumps(REG))
    rc, err, znodes = prep.register_(client)
    assert rc == ra.ZOK, err
    assert len(znodes) == 20
    assert ensemble.ephemeral_count() == 20
    # re-register through the same swap path
    rc, err, znodes2 = prep.register_(client)
    assert rc == ra.ZOK, err
    assert znodes2 == znodes
    assert ensemble.ephemeral_count() == 20


def test_atomic_swap_no_observable_gap(ensemble):
    """The headline property: during continuous re-registration, a reader
    NEVER observes the host znode missing (the reference's non-atomic
    cleanup→create always has a window where it is gone)."""
    writer = make_client(ensemble)
    reader = make_client(ensemble)
    prep = ra.PreparedRegistration(json.dumps(REG))
    rc, err, znodes = prep.register_(writer)
    assert rc == ra.ZOK, err
    host_node = [n for n in znodes if n.endswith("/s0")][0]

    stop = threading.Event()
    gaps = []

    def poll():
        while not stop.is_set():
            rc, _ = reader.exists(host_node)
            if rc != ra.ZOK:
                gaps.append(rc)

    t = threading.Thread(target=poll)
    t.start()
    for _ in range(30):  # 30 full re-registrations under continuous polling
        rc, err, _ = prep.register_(writer)
        assert rc == ra.ZOK, err
    stop.set()
    t.join()
    assert gaps == [], "reader observed %d missing-node windows" % len(gaps)
    writer.close()
    reader.close()


def test_nonatomic_register_has_gap(ensemble):
    """Control experiment: the reference-shaped pipeline (cleanup, settle
    delay, create — the settle is fixed at 1000 ms in the reference,
    lib/register.js:232-235; shortened here) DOES expose a missing-node
    window under the same polling — the property the atomic swap removes.
    (With settleMs=0 this build pipelines the stages into one round trip,
    shrinking the gap to microseconds; atomicSwap removes it entirely.)"""
    cfg = dict(REG)
    cfg["atomicSwap"] = False
    cfg["settleMs"] = 30
    writer = make_client(ensemble)
    reader = make_client(ensemble)
    prep = ra.PreparedRegistration(json.dumps(cfg))
    rc, err, znodes = prep.register_(writer)
    assert rc == ra.ZOK, err
    host_node = [n for n in znodes if n.endswith("/s0")][0]
    stop = threading.Event()
    gaps = []

    def poll():
        while not stop.is_set():
            rc, _ = reader.exists(host_node)
            if rc != ra.ZOK:
                gaps.append(rc)

    t = threading.Thread(target=poll)
    t.start()
    for _ in range(30):
        rc, err, _ = prep.register_(writer)
        assert rc == ra.ZOK, err
    stop.set()
    t.join()
    assert gaps, "expected the non-atomic pipeline to expose a gap"
    writer.close()
    reader.close()


def test_atomic_swap_via_orchestrator(ensemble):
    cfg = orch_config(ensemble, REG, heartbeatInterval=100)
    o = ra.Orchestrator(json.dumps(cfg))
    o.start()
    assert o.wait_registered(15000)
    assert ensemble.ephemeral_count() == 20
    # expiry storm through the atomic path
    sid1 = o.session_id()
    ensemble.expire_session(sid1)
    assert wait_for(lambda: ensemble.ephemeral_count() == 20
                    and o.session_id() not in (0, sid1), timeout=20)
    o.stop()


def test_multi_delete_of_other_sessions_ephemeral(ensemble):
    """Deleting another session's ephemeral in a txn must update that owner's
    bookkeeping: when the owner later dies, counts stay exact."""
    owner = make_client(ensemble)
    actor = make_client(ensemble)
    owner.mkdirp("/x")
    owner.create("/x/theirs", b"", True)
    assert ensemble.ephemeral_count() == 1
    rc, _ = actor.multi([("delete", "/x/theirs"), ("create", "/x/mine", b"", True)])
    assert rc == ra.ZOK
    assert ensemble.ephemeral_count() == 1  # theirs removed from the owner's books
    ensemble.expire_session(owner.session_id())
    assert wait_for(lambda: owner.state() == "expired", timeout=10)
    assert ensemble.get("/x/mine")["exists"]  # actor's node untouched by owner's death
    assert ensemble.ephemeral_count() == 1
    actor.close()
    owner.close()
