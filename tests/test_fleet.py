"""Fleet integration: BASELINE config 3's shape end-to-end — 8 registrard
daemons (one per GPU slot) publishing host records under ONE domain, a
binder-lite resolver answering A/SRV from them, and the failure semantics a
consumer actually sees: a killed daemon's record leaves DNS when its session
expires, and a health-gated daemon unregisters (and recovers) without
touching its 7 siblings.

The reference never had a multi-instance test at all; its SMF manifest is
single_instance (smf/manifests/registrar.xml.in:13). The per-GPU fleet is
this build's north-star deployment (one process per MI355X GPU).
"""
import json
import os
import signal
import socket
import struct
import subprocess
import tempfile
import time

import pytest

import registrar_amd as ra
from registrar_amd.binder_lite import BinderLite, _encode_name
from conftest import REPO_ROOT, wait_for

FLEET = 8
DOMAIN = "workers.fleet.test"


def servers_of(ens):
    out = []
    for hp in ens.connect_string().split(","):
        host, port = hp.rsplit(":", 1)
        out.append((host, int(port)))
    return out


def dns_a(addr, name):
    q = struct.pack(">HHHHHH", 0x7a7a, 0x0100, 1, 0, 0, 0)
    q += _encode_name(name) + struct.pack(">HH", 1, 1)
    s = socket.socket(socket.AF_INET, socket.SOCK_DGRAM)
    s.settimeout(5)
    s.sendto(q, addr)
    buf, _ = s.recvfrom(4096)
    s.close()
    an = struct.unpack(">H", buf[6:8])[0]
    # walk answers for A rdata
    off = 12
    while buf[off] != 0:
        off += 1 + buf[off]
    off += 1 + 4
    addrs = set()
    for _ in range(an):
        off += 2
        rtype, _, _, rdlen = struct.unpack(">HHIH", buf[off:off + 10])
        off += 10
        if rtype == 1:
            addrs.add(socket.inet_ntoa(buf[off:off + 4]))
        off += rdlen
    return addrs


@pytest.fixture
def fleet(ensemble3, daemon_bin, tmp_path):
    gates = []
    procs = []
    cfgs = []
    servers = [{"host": h, "port": p} for h, p in servers_of(ensemble3)]
    for i in range(FLEET):
        gate = tmp_path / ("gate%d" % i)
        gate.write_text("")
        gates.append(str(gate))
        cfg = {
            "zookeeper": {"servers": servers, "timeout": 3000, "connectTimeout": 4000},
            "registration": {
                "domain": DOMAIN,
                "type": "host",
                "adminIp": "10.77.0.%d" % (i + 1),
                "hostname": "gpu%d" % i,
                "settleMs": 0,
                "ports": [9000 + i],
            },
            "heartbeatInterval": 150,
            "healthCheck": {"command": "test -e %s" % gate, "interval": 100,
                            "timeout": 1000, "threshold": 1, "period": 60000},
        }
        p = tmp_path / ("cfg%d.json" % i)
        p.write_text(json.dumps(cfg))
        cfgs.append(str(p))
        procs.append(subprocess.Popen([daemon_bin, "-f", str(p)],
                                      stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL))
    yield procs, gates
    for p in procs:
        if p.poll() is None:
            p.send_signal(signal.SIGTERM)
    for p in procs:
        try:
            p.wait(timeout=10)
        except subprocess.TimeoutExpired:
            p.kill()


def test_fleet_of_eight_under_one_domain(ensemble3, fleet):
    procs, gates = fleet
    path = ra.domain_to_path(DOMAIN)
    # all 8 register under the single domain path
    assert wait_for(lambda: len(ensemble3.children(path)) == FLEET, timeout=30), \
        "fleet did not fully register: %r" % ensemble3.children(path)
    assert sorted(ensemble3.children(path)) == ["gpu%d" % i for i in range(FLEET)]

    # a binder answers with all 8 addresses
    binder = BinderLite(servers_of(ensemble3))
    binder.start()
    try:
        addrs = dns_a(binder.address, DOMAIN)
        assert addrs == {"10.77.0.%d" % (i + 1) for i in range(FLEET)}

        # hard-kill daemon 3: its EPHEMERAL record dies with the session
        # (session timeout 3 s), siblings unaffected
        procs[3].kill()
        assert wait_for(lambda: "gpu3" not in ensemble3.children(path), timeout=20)
        addrs = dns_a(binder.address, DOMAIN)
        assert "10.77.0.4" not in addrs and len(addrs) == FLEET - 1

        # health-gate daemon 5: it unregisters itself (no session death)
        os.unlink(gates[5])
        assert wait_for(lambda: "gpu5" not in ensemble3.children(path), timeout=20)
        assert len(dns_a(binder.address, DOMAIN)) == FLEET - 2
        # and recovers
        open(gates[5], "w").close()
        assert wait_for(lambda: "gpu5" in ensemble3.children(path), timeout=20)
        assert "10.77.0.6" in dns_a(binder.address, DOMAIN)

        # the survivors never flapped: their records are continuously owned
        # by live sessions
        for name in sorted(ensemble3.children(path)):
            st = ensemble3.get("%s/%s" % (path, name))["stat"]
            assert st["ephemeralOwner"] != 0
    finally:
        binder.stop()
