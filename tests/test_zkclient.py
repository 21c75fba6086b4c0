"""Native ZK client: connect/retry, ops, pipelining, session semantics.

Mirrors the reference's test/zk.test.js (connect-failure retry with abort,
successful connect exposes heartbeat) and extends it with the session tests
the reference could not run hermetically (expiry, same-session reconnect)."""
import time

import pytest

import registrar_amd as ra
from conftest import free_port, make_client, wait_for


def test_connect_and_heartbeat_surface(ensemble, client):
    # reference test/zk.test.js:54-71: connected client has a heartbeat verb
    assert client.state() == "connected"
    assert client.session_id() != 0
    rc, rtt = client.heartbeat([])
    assert rc == ra.ZOK
    assert "session" in client.to_string()


def test_connect_failure_retries_then_abort():
    # reference test/zk.test.js:30-51: dead port, watch attempts, abort stops
    port = free_port()  # nothing listens here
    c = ra.ZkClient(
        servers=[("127.0.0.1", port)],
        connect_timeout_ms=200,
        connect_initial_delay_ms=50,
        connect_max_delay_ms=100,
    )
    c.start()
    assert wait_for(lambda: sum(1 for e in _drain(c) if e["type"] == "attempt") >= 2, timeout=10)
    c.abort_connect()
    assert not c.wait_connected(2000)
    assert c.state() == "closed"
    c.close()


_seen = {}


def _drain(c):
    evs = _seen.setdefault(id(c), [])
    evs.extend(c.poll_events())
    return evs


def test_connect_failure_exhaustion():
    port = free_port()
    c = ra.ZkClient(
        servers=[("127.0.0.1", port)],
        connect_timeout_ms=200,
        connect_initial_delay_ms=10,
        connect_max_delay_ms=20,
        connect_max_attempts=3,
    )
    c.start()
    assert not c.wait_connected(15000)
    assert c.state() == "closed"
    c.close()


def test_basic_ops(client, ensemble):
    rc, _ = client.create("/t", b"v0")
    assert rc == ra.ZOK
    rc, data, st = client.get("/t")
    assert rc == ra.ZOK and data == b"v0" and st["version"] == 0
    assert client.set("/t", b"v1") == ra.ZOK
    rc, data, st = client.get("/t")
    assert data == b"v1" and st["version"] == 1
    # version conflict
    assert client.set("/t", b"v2", version=0) == ra.ZBADVERSION
    # children
    client.create("/t/a", b"")
    client.create("/t/b", b"")
    rc, ch = client.get_children("/t")
    assert rc == ra.ZOK and sorted(ch) == ["a", "b"]
    # not empty
    assert client.delete_("/t") == ra.ZNOTEMPTY
    assert client.delete_("/t/a") == ra.ZOK
    assert client.delete_("/t/b") == ra.ZOK
    assert client.delete_("/t") == ra.ZOK
    assert client.exists("/t")[0] == ra.ZNONODE
    assert client.delete_("/t") == ra.ZNONODE


def test_ephemeral_owner_stat(client):
    # the ephemeralOwner assertion from reference test/register.test.js:41-42
    client.create("/e", b"x", True)
    rc, st = client.exists("/e")
    assert rc == ra.ZOK
    assert st["ephemeralOwner"] == client.session_id()


def test_no_children_for_ephemerals(client):
    client.create("/eph", b"", True)
    rc, _ = client.create("/eph/child", b"")
    assert rc == ra.ZNOCHILDRENFOREPHEMERALS


def test_create_no_parent(client):
    rc, _ = client.create("/missing/child", b"")
    assert rc == ra.ZNONODE


def test_node_exists(client):
    client.create("/dup", b"")
    rc, _ = client.create("/dup", b"")
    assert rc == ra.ZNODEEXISTS


def test_mkdirp_and_put(client, ensemble):
    assert client.mkdirp("/x/y/z") == ra.ZOK
    assert client.mkdirp("/x/y/z") == ra.ZOK  # idempotent
    assert client.exists("/x/y/z")[0] == ra.ZOK
    assert client.put("/x/y/z", b"payload") == ra.ZOK  # overwrite existing
    assert ensemble.get("/x/y/z")["data"] == b"payload"
    assert client.put("/x/y/new", b"n") == ra.ZOK  # create path
    assert ensemble.get("/x/y/new")["data"] == b"n"


def test_pipelined_batches(client, ensemble):
    paths = ["/batch/n%04d" % i for i in range(200)]
    assert client.mkdirp("/batch") == ra.ZOK
    rcs = client.create_many(paths, b"d", True)
    assert all(rc == ra.ZOK for rc in rcs)
    assert ensemble.ephemeral_count() == 200
    rcs = client.exists_many(paths)
    assert all(rc == ra.ZOK for rc in rcs)
    rcs = client.delete_many(paths)
    assert all(rc == ra.ZOK for rc in rcs)
    assert ensemble.ephemeral_count() == 0


def test_session_expiry_event(ensemble, client):
    sid = client.session_id()
    ensemble.expire_session(sid)
    assert wait_for(lambda: client.state() == "expired", timeout=10)
    evs = client.poll_events()
    assert any(e["type"] == "session_expired" for e in evs)
    # ephemerals die with the session
    assert ensemble.ephemeral_count() == 0


def test_ephemerals_vanish_on_close(ensemble):
    c = make_client(ensemble)
    c.create("/gone", b"", True)
    assert ensemble.ephemeral_count() == 1
    c.close()
    assert wait_for(lambda: ensemble.ephemeral_count() == 0, timeout=5)


def test_session_survives_server_kill(ensemble3):
    c = make_client(ensemble3)
    sid = c.session_id()
    c.create("/sticky", b"", True)
    # kill whichever server the client landed on — kill all but one, one at a
    # time, and the session must survive via same-session reconnect
    ports = ensemble3.ports()
    assert len(ports) == 3
    ensemble3.kill_server(0)
    ensemble3.kill_server(1)
    assert wait_for(lambda: c.exists("/sticky")[0] == ra.ZOK, timeout=10)
    assert c.session_id() == sid
    assert ensemble3.get("/sticky")["exists"]
    c.close()


def test_session_timeout_expires_ephemerals(ensemble):
    # client with a short session dies (simulated by killing the server so no
    # pings flow), ephemerals vanish after the timeout
    c = make_client(ensemble, session_timeout_ms=400)
    c.create("/shortlived", b"", True)
    ensemble.kill_server(0)
    assert wait_for(lambda: ensemble.ephemeral_count() == 0, timeout=5)
    # the client is now reconnecting; bring the server back and the session is
    # gone ⇒ expired handshake
    ensemble.restart_server(0)
    assert wait_for(lambda: c.state() == "expired", timeout=10)
    c.close()


def test_initial_connect_skips_dead_first_server(ensemble):
    """Connect-string failover at START: the first server in the list is
    unreachable; the client must rotate to the live one without burning a
    full backoff cycle per attempt forever (lib/zk.js failover semantics)."""
    from conftest import free_port

    dead = ("127.0.0.1", free_port())
    live_host, live_port = ensemble.connect_string().rsplit(":", 1)
    c = ra.ZkClient(servers=[dead, (live_host, int(live_port))],
                    session_timeout_ms=10000, randomize_start=False,
                    connect_initial_delay_ms=50, connect_max_delay_ms=100)
    c.start()
    try:
        assert c.wait_connected(15000), "failover to the live server failed"
        rc, _ = c.create("/fo", b"", True)
        assert rc == ra.ZOK
    finally:
        c.close()
