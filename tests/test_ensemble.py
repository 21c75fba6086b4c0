"""Synthetic ensemble behaviors beyond what the client tests cover:
multi-server mode, leader kill/election, latency injection, introspection.

These capabilities exist so the 8-process and expiry-storm benchmarks mean
something (SURVEY §4 'Implication for the new build', §7.3)."""
import time

import pytest

import registrar_amd as ra
from conftest import make_client, wait_for


def test_three_server_ports_distinct(ensemble3):
    ports = ensemble3.ports()
    assert len(ports) == 3 and len(set(ports)) == 3
    assert ensemble3.connect_string().count(",") == 2


def test_kill_restart_server(ensemble3):
    assert ensemble3.server_up(1)
    ensemble3.kill_server(1)
    assert not ensemble3.server_up(1)
    ensemble3.restart_server(1)
    assert ensemble3.server_up(1)
    # port is stable across restart
    p_before = ensemble3.ports()
    ensemble3.kill_server(1)
    ensemble3.restart_server(1)
    assert ensemble3.ports() == p_before


def test_leader_kill_moves_leader(ensemble3):
    first = ensemble3.leader()
    killed = ensemble3.kill_leader()
    assert killed == first
    assert not ensemble3.server_up(killed)
    # data survives: shared tree
    c = make_client(ensemble3)
    c.create("/after-kill", b"x")
    assert ensemble3.get("/after-kill")["exists"]
    c.close()
    assert ensemble3.leader() != killed or ensemble3.server_up(ensemble3.leader())


def test_latency_injection(ensemble):
    c = make_client(ensemble)
    t0 = time.monotonic()
    c.create("/fast", b"")
    fast = time.monotonic() - t0
    ensemble.set_latency_ms(100)
    t0 = time.monotonic()
    c.create("/slow", b"")
    slow = time.monotonic() - t0
    assert slow >= 0.09 and slow > fast
    ensemble.set_latency_ms(0)
    c.close()


def test_counters(ensemble):
    c = make_client(ensemble)
    c.create("/cnt", b"")
    c.exists("/cnt")
    c.get("/cnt")
    c.delete_("/cnt")
    cnt = ensemble.counters()
    assert cnt["connect"] >= 1
    assert cnt["create"] >= 1
    assert cnt["exists"] >= 1
    assert cnt["getData"] >= 1
    assert cnt["delete"] >= 1
    c.close()


def test_introspection(ensemble):
    c = make_client(ensemble)
    c.mkdirp("/intro/a")
    c.create("/intro/a/e", b"payload", True)
    assert ensemble.node_count() >= 3
    assert ensemble.children("/intro") == ["a"]
    info = ensemble.get("/intro/a/e")
    assert info["exists"] and info["data"] == b"payload"
    assert info["stat"]["ephemeralOwner"] == c.session_id()
    assert c.session_id() in ensemble.session_ids()
    assert ensemble.zxid() > 0
    c.close()


def test_two_clients_share_tree(ensemble):
    c1 = make_client(ensemble)
    c2 = make_client(ensemble)
    assert c1.session_id() != c2.session_id()
    c1.create("/shared", b"from-c1")
    rc, data, _ = c2.get("/shared")
    assert rc == ra.ZOK and data == b"from-c1"
    # c1's ephemerals are not c2's
    c1.create("/shared/e1", b"", True)
    ensemble.expire_session(c2.session_id())
    assert wait_for(lambda: c2.state() == "expired", timeout=5)
    assert ensemble.get("/shared/e1")["exists"]  # c1 unaffected
    c1.close()
    c2.close()


def test_election_pause_blocks_connects():
    ens = ra.Ensemble(servers=2, tick_ms=50, election_ms=500, min_session_timeout_ms=200)
    ens.start()
    try:
        ens.kill_leader()
        t0 = time.monotonic()
        c = make_client(ens)  # must wait out the election window
        dt = time.monotonic() - t0
        assert dt >= 0.3, "connect should have been refused during election, took %.3fs" % dt
        c.close()
    finally:
        ens.stop()
