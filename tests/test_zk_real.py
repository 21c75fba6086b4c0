"""Interop tests against a REAL Apache ZooKeeper (`-m zk_real`).

The reference ran its whole ZK suite against a real server on localhost
(/root/reference/test/helper.js:57-62). This container has no JVM, so no real
ZooKeeper can run here — the wire protocol is instead pinned by the
independent byte-level vectors in test_wire_golden.py. Wherever a real
ensemble IS reachable, point ZK_HOST/ZK_PORT at it and run

    ZK_HOST=10.0.0.5 ZK_PORT=2181 python -m pytest tests -m zk_real

to exercise the native client's session handshake, register pipeline,
heartbeat, watches and unregister against Apache ZooKeeper itself. Tests use
a unique chroot-style path prefix and clean up after themselves.
"""
import json
import os
import time
import uuid

import pytest

import registrar_amd as ra
from conftest import wait_for

pytestmark = [
    pytest.mark.zk_real,
    pytest.mark.skipif("ZK_HOST" not in os.environ,
                       reason="set ZK_HOST[:ZK_PORT] to a real ZooKeeper to run interop tests"),
]

RUN_ID = "zkreal%s" % uuid.uuid4().hex[:8]


def real_servers():
    host = os.environ["ZK_HOST"]
    port = int(os.environ.get("ZK_PORT", "2181"))
    return [(host, port)]


@pytest.fixture
def client():
    c = ra.ZkClient(servers=real_servers(), session_timeout_ms=15000)
    c.start()
    assert c.wait_connected(15000), "cannot connect to real ZooKeeper at $ZK_HOST"
    yield c
    c.close()


def cleanup(c, path):
    rc, children = c.get_children(path)
    if rc == ra.ZOK:
        for ch in children:
            cleanup(c, "%s/%s" % (path, ch))
    c.delete_(path)


def test_real_session_and_crud(client):
    base = "/%s-crud" % RUN_ID
    try:
        rc, created = client.create(base, b'{"k":1}')
        assert rc == ra.ZOK and created == base
        rc, data, stat = client.get(base)
        assert rc == ra.ZOK and data == b'{"k":1}' and stat["version"] == 0
        assert client.set(base, b'{"k":2}', 0) == ra.ZOK
        rc, data, stat = client.get(base)
        assert data == b'{"k":2}' and stat["version"] == 1
        assert client.set(base, b"x", 0) == ra.ZBADVERSION
        assert client.delete_(base, 1) == ra.ZOK
        assert client.exists(base)[0] == ra.ZNONODE
    finally:
        cleanup(client, base)


def test_real_ephemeral_dies_with_session(client):
    base = "/%s-eph" % RUN_ID
    try:
        c2 = ra.ZkClient(servers=real_servers(), session_timeout_ms=15000)
        c2.start()
        assert c2.wait_connected(15000)
        rc, _ = c2.create(base, b"", True)
        assert rc == ra.ZOK
        rc, stat = client.exists(base)
        assert rc == ra.ZOK and stat["ephemeralOwner"] == c2.session_id()
        c2.close()
        assert wait_for(lambda: client.exists(base)[0] == ra.ZNONODE, timeout=20)
    finally:
        cleanup(client, base)


def test_real_register_heartbeat_unregister(client):
    domain = "h1.%s.interop.test" % RUN_ID
    registration = {"domain": domain, "type": "host", "adminIp": "127.0.0.1",
                    "hostname": "real-zk-host", "settleMs": 0,
                    "aliases": ["a%d.%s.interop.test" % (i, RUN_ID) for i in range(10)]}
    prep = ra.PreparedRegistration(json.dumps(registration))
    try:
        rc, err, znodes = prep.register_(client)
        assert rc == ra.ZOK, err
        assert len(znodes) == 11  # hostname node + 10 aliases
        for z in znodes:
            rc, data, stat = client.get(z)
            assert rc == ra.ZOK
            assert stat["ephemeralOwner"] == client.session_id()
            rec = json.loads(data)
            assert rec["type"] == "host" and rec["address"] == "127.0.0.1"
        rc, rtt_us = prep.heartbeat(client)
        assert rc == ra.ZOK and rtt_us > 0
        # re-register is idempotent (cleanup of previous entries first)
        rc, err, znodes2 = prep.register_(client)
        assert rc == ra.ZOK and sorted(znodes2) == sorted(znodes)
        assert prep.unregister(client) == ra.ZOK
        for z in znodes:
            assert client.exists(z)[0] == ra.ZNONODE
    finally:
        cleanup(client, ra.domain_to_path(domain))
        cleanup(client, "/test")


def test_real_watches(client):
    base = "/%s-watch" % RUN_ID
    try:
        c2 = ra.ZkClient(servers=real_servers(), session_timeout_ms=15000)
        c2.start()
        assert c2.wait_connected(15000)
        rc, _ = client.exists(base, watch=True)
        assert rc == ra.ZNONODE
        c2.create(base, b"v0")
        assert wait_for(lambda: any(w["type"] == "created" and w["path"] == base
                                    for w in client.poll_watches()), timeout=10)
        client.get(base, watch=True)
        c2.set(base, b"v1", -1)
        assert wait_for(lambda: any(w["type"] == "changed" and w["path"] == base
                                    for w in client.poll_watches()), timeout=10)
        c2.close()
    finally:
        cleanup(client, base)


def test_real_multi(client):
    base = "/%s-multi" % RUN_ID
    try:
        rc, per_op = client.multi([("create", base, b"", False),
                                   ("create", base + "/a", b"x", True),
                                   ("delete", base + "/a", b"", False)])
        assert rc == ra.ZOK
        # rollback on failure
        rc, per_op = client.multi([("create", base + "/b", b"", False),
                                   ("delete", base + "/missing", b"", False)])
        assert rc == ra.ZNONODE
        assert client.exists(base + "/b")[0] == ra.ZNONODE
    finally:
        cleanup(client, base)
