"""Storm and scale scenarios (BASELINE.json configs 4 and 5 analogs, CPU-sized):
1k-znode registration, session-expiry storm with verified re-registration,
leader-kill with session survival."""
import json
import time

import pytest

import registrar_amd as ra
from conftest import make_client, orch_config, wait_for

N = 1000


def registration_1k(idx=0):
    # hostname node + 999 aliases ⇒ 1000 ephemeral znodes, the BASELINE shape
    return {
        "domain": "p%d.storm.test" % idx,
        "type": "host",
        "adminIp": "127.0.0.1",
        "hostname": "h%d" % idx,
        "settleMs": 0,
        "aliases": ["a%04d.p%d.storm.test" % (i, idx) for i in range(N - 1)],
    }


def test_register_1k_znodes(ensemble, client):
    t0 = time.monotonic()
    rc, err, znodes = ra.register_node(client, json.dumps(registration_1k()))
    dt = time.monotonic() - t0
    assert rc == ra.ZOK, err
    assert len(znodes) == N
    assert ensemble.ephemeral_count() == N
    # sanity floor, not a benchmark: 1k pipelined creates (plus cleanup +
    # mkdirp) should take well under 5 s even on a loaded CI box
    assert dt < 5.0, "1k-node register took %.2fs" % dt
    # heartbeat over all 1k inside the reference's 3000ms cadence envelope
    rc, rtt_us = client.heartbeat(znodes)
    assert rc == ra.ZOK
    assert rtt_us < 3_000_000, "1k heartbeat took %dus" % rtt_us
    rcs = client.delete_many(znodes)
    assert all(r == ra.ZOK for r in rcs)


def test_expiry_storm_reregisters_1k(ensemble):
    cfg = orch_config(ensemble, registration_1k(), heartbeatInterval=200)
    o = ra.Orchestrator(json.dumps(cfg))
    o.start()
    assert o.wait_registered(30000)
    assert ensemble.ephemeral_count() == N
    sid1 = o.session_id()
    ensemble.expire_session(sid1)
    # all 1k ephemerals vanish server-side, then the orchestrator re-registers
    # every one of them under a fresh session (BASELINE config 4)
    assert wait_for(lambda: o.metrics()["registers"] >= 2, timeout=30)
    assert wait_for(lambda: ensemble.ephemeral_count() == N, timeout=30)
    sid2 = o.session_id()
    assert sid2 not in (0, sid1)
    for n in o.znodes()[:10]:
        assert ensemble.get(n)["stat"]["ephemeralOwner"] == sid2
    o.stop()


def test_leader_kill_storm_session_survives(ensemble3):
    cfg = orch_config(ensemble3, registration_1k(), heartbeatInterval=200)
    o = ra.Orchestrator(json.dumps(cfg))
    o.start()
    assert o.wait_registered(30000)
    sid = o.session_id()
    hb0 = o.metrics()["heartbeats"]
    ensemble3.kill_leader()
    # session survives via same-session reconnect to a surviving server:
    # no re-register, no ephemeral loss, heartbeats resume
    assert wait_for(lambda: o.metrics()["heartbeats"] > hb0 + 2, timeout=30)
    assert o.session_id() == sid
    assert ensemble3.ephemeral_count() == N
    assert o.metrics()["registers"] == 1
    o.stop()


def test_aliases_multipath_flap(ensemble):
    """BASELINE config 5: aliases + flap damping under intermittent failures."""
    import os
    import tempfile

    tmpdir = tempfile.mkdtemp()
    once = os.path.join(tmpdir, "fail-once")
    always = os.path.join(tmpdir, "fail-always")
    # fails exactly once per `once` flag (self-consuming), or continuously
    # while `always` exists
    command = ("if [ -f {o} ]; then rm -f {o}; exit 1; fi; " "test ! -f {a}").format(o=once, a=always)
    registration = {
        "domain": "flap5.test",
        "type": "host",
        "adminIp": "127.0.0.1",
        "hostname": "fh",
        "settleMs": 0,
        "aliases": ["a.flap5.test", "b.flap5.test"],
    }
    cfg = orch_config(
        ensemble,
        registration,
        heartbeatInterval=100,
        healthCheck={"command": command, "interval": 30, "timeout": 500,
                     "threshold": 3, "period": 60000},
    )
    o = ra.Orchestrator(json.dumps(cfg))
    o.start()
    assert o.wait_registered(10000)
    znodes = o.znodes()
    assert len(znodes) == 3

    # one isolated failure (below threshold) must NOT unregister
    open(once, "w").close()
    assert wait_for(lambda: not os.path.exists(once), timeout=5)  # consumed
    time.sleep(0.3)
    assert all(ensemble.get(n)["exists"] for n in znodes)
    assert o.metrics()["unregisters"] == 0

    # sustained failure crosses the threshold: all three paths unregistered
    open(always, "w").close()
    assert wait_for(lambda: all(not ensemble.get(n)["exists"] for n in znodes), timeout=10)

    # recovery: all three come back
    os.unlink(always)
    assert wait_for(lambda: all(ensemble.get(n)["exists"] for n in znodes), timeout=10)
    o.stop()
    os.rmdir(tmpdir)


def test_register_10k_znodes_scale(ensemble, client):
    """10x the BASELINE scale in one registration: exercises buffer growth,
    shard distribution, and large pipelined batches."""
    registration = {
        "domain": "big.scale.test",
        "type": "host",
        "adminIp": "127.0.0.1",
        "hostname": "big0",
        "settleMs": 0,
        "aliases": ["a%05d.big.scale.test" % i for i in range(9999)],
    }
    import registrar_amd as ra

    prep = ra.PreparedRegistration(json.dumps(registration))
    t0 = time.monotonic()
    rc, err, znodes = prep.register_(client)
    dt = time.monotonic() - t0
    assert rc == ra.ZOK, err
    assert len(znodes) == 10000
    assert ensemble.ephemeral_count() == 10000
    assert dt < 30.0, "10k-node register took %.2fs" % dt
    rc, rtt_us = client.heartbeat(znodes)
    assert rc == ra.ZOK
    assert rtt_us < 3_000_000  # still inside the heartbeat cadence envelope
    rcs = client.delete_many(znodes)
    assert all(r == ra.ZOK for r in rcs)


def test_expiry_during_election_recovers(ensemble3):
    """Worst-case storm: session expired AND leader killed at once; the
    orchestrator's bounded re-register retries must ride out the chaos."""
    cfg = orch_config(ensemble3, registration_1k(), heartbeatInterval=200)
    o = ra.Orchestrator(json.dumps(cfg))
    o.start()
    assert o.wait_registered(30000)
    sid1 = o.session_id()
    ensemble3.kill_leader()
    ensemble3.expire_session(sid1)
    assert wait_for(lambda: ensemble3.ephemeral_count() == N
                    and o.session_id() not in (0, sid1), timeout=40)
    assert not o.expired()
    o.stop()
