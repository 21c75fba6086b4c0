"""binder_lite: DNS answers from registrar records — the discovery triangle
end-to-end (registrar writes → binder-lite reads → DNS A/SRV answers)."""
import json
import socket
import struct
import time

import pytest

import registrar_amd as ra
from registrar_amd.binder_lite import BinderLite, _encode_name
from conftest import make_client, wait_for


def dns_query(addr, name, qtype):
    q = struct.pack(">HHHHHH", 0x1234, 0x0100, 1, 0, 0, 0)
    q += _encode_name(name) + struct.pack(">HH", qtype, 1)
    s = socket.socket(socket.AF_INET, socket.SOCK_DGRAM)
    s.settimeout(5)
    s.sendto(q, addr)
    buf, _ = s.recvfrom(4096)
    s.close()
    txid, flags, qd, an, ns, ar = struct.unpack(">HHHHHH", buf[:12])
    assert txid == 0x1234
    return buf, flags & 0xF, an


def parse_answers(buf, an):
    # skip header + question
    off = 12
    while buf[off] != 0:
        off += 1 + buf[off]
    off += 1 + 4
    answers = []
    for _ in range(an):
        off += 2  # name pointer
        rtype, rclass, ttl, rdlen = struct.unpack(">HHIH", buf[off:off + 10])
        off += 10
        rdata = buf[off:off + rdlen]
        off += rdlen
        answers.append((rtype, ttl, rdata))
    return answers


@pytest.fixture
def binder(ensemble):
    servers = []
    for hp in ensemble.connect_string().split(","):
        host, port = hp.rsplit(":", 1)
        servers.append((host, int(port)))
    b = BinderLite(servers)
    b.start()
    yield b
    b.stop()


def register(client, domain, hosts, ttl=None, service=None):
    for i, addr in enumerate(hosts):
        reg = {"domain": domain, "type": "host", "adminIp": addr,
               "hostname": "n%d" % i, "settleMs": 0}
        if ttl is not None:
            reg["ttl"] = ttl
        if service is not None:
            reg["service"] = service
        rc, err, _ = ra.register_node(client, json.dumps(reg))
        assert rc == ra.ZOK, err


def test_a_records(ensemble, client, binder):
    register(client, "web.dns.test", ["10.0.0.1", "10.0.0.2"], ttl=120)
    buf, rcode, an = dns_query(binder.address, "web.dns.test", 1)
    assert rcode == 0 and an == 2
    answers = parse_answers(buf, an)
    addrs = sorted(socket.inet_ntoa(rd) for _, _, rd in answers)
    assert addrs == ["10.0.0.1", "10.0.0.2"]
    assert all(ttl == 120 for _, ttl, _ in answers)


def test_srv_records_and_ttl_precedence(ensemble, client, binder):
    service = {"type": "service", "service": {"srvce": "_http", "proto": "_tcp", "port": 8080, "ttl": 45}}
    register(client, "api.dns.test", ["10.1.0.1"], service=service)  # no host ttl
    buf, rcode, an = dns_query(binder.address, "api.dns.test", 33)
    assert rcode == 0 and an == 1
    rtype, ttl, rdata = parse_answers(buf, an)[0]
    assert rtype == 33
    assert ttl == 45  # service-record ttl (host ttl absent)
    prio, weight, port = struct.unpack(">HHH", rdata[:6])
    assert port == 8080


def test_host_ttl_wins_over_service(ensemble, client, binder):
    service = {"type": "service", "service": {"srvce": "_http", "proto": "_tcp", "port": 80, "ttl": 45}}
    register(client, "ttl.dns.test", ["10.2.0.1"], ttl=7, service=service)
    buf, rcode, an = dns_query(binder.address, "ttl.dns.test", 1)
    assert an == 1
    assert parse_answers(buf, an)[0][1] == 7  # record-level ttl wins


def test_nxdomain(binder):
    _, rcode, an = dns_query(binder.address, "nosuch.dns.test", 1)
    assert rcode == 3 and an == 0


def test_liveness_via_ephemerals(ensemble, binder):
    c2 = make_client(ensemble)
    register(c2, "gone.dns.test", ["10.3.0.1"])
    _, rcode, an = dns_query(binder.address, "gone.dns.test", 1)
    assert an == 1
    # instance dies ⇒ session closes ⇒ ephemeral vanishes ⇒ out of DNS
    c2.close()
    assert wait_for(lambda: ensemble.ephemeral_count() == 0, timeout=10)
    _, rcode, an = dns_query(binder.address, "gone.dns.test", 1)
    assert an == 0  # path (parents) persists: NOERROR, no answers


def test_gpu_payload_does_not_break_answers(ensemble, client, binder):
    reg = {"domain": "gpu.dns.test", "type": "host", "adminIp": "10.4.0.1", "hostname": "g0",
           "settleMs": 0, "gpu": {"index": 0, "xgmiRank": 0, "uuid": "GPU-x"},
           "service": {"type": "service", "service": {"srvce": "_infer", "proto": "_tcp", "port": 8000}}}
    rc, err, _ = ra.register_node(client, json.dumps(reg))
    assert rc == ra.ZOK
    buf, rcode, an = dns_query(binder.address, "gpu.dns.test", 33)
    assert an == 1
    _, _, rdata = parse_answers(buf, an)[0]
    assert struct.unpack(">HHH", rdata[:6])[2] == 8000


def test_full_triangle_health_drives_dns(ensemble, binder, tmp_path):
    """The complete discovery loop: an orchestrated registrar serves DNS via
    binder-lite; a health failure takes it out of DNS; recovery brings it
    back — the end-to-end behavior the reference system delivers with three
    separate components (registrar + ZooKeeper + Binder)."""
    import registrar_amd as ra
    from conftest import orch_config, wait_for

    gate = tmp_path / "gate"
    gate.write_text("")
    cfg = orch_config(
        ensemble,
        {"domain": "tri.dns.test", "type": "host", "adminIp": "10.9.0.1", "hostname": "t0",
         "settleMs": 0,
         "service": {"type": "service", "service": {"srvce": "_svc", "proto": "_tcp", "port": 9000}}},
        heartbeatInterval=100,
        healthCheck={"command": "test -e %s" % gate, "interval": 40, "timeout": 500,
                     "threshold": 2, "period": 60000},
    )
    o = ra.Orchestrator(json.dumps(cfg))
    o.start()
    assert o.wait_registered(15000)

    def answers(qtype=1):
        buf, rcode, an = dns_query(binder.address, "tri.dns.test", qtype)
        return an

    assert wait_for(lambda: answers() == 1, timeout=5)
    # SRV answer carries the service port
    buf, rcode, an = dns_query(binder.address, "tri.dns.test", 33)
    assert an == 1
    assert struct.unpack(">HHH", parse_answers(buf, an)[0][2][:6])[2] == 9000

    gate.unlink()  # GPU/instance goes sick → flap damping → unregister
    assert wait_for(lambda: answers() == 0, timeout=10)

    gate.write_text("")  # recovery → re-register → back in DNS
    assert wait_for(lambda: answers() == 1, timeout=10)
    o.stop()


def dns_query_tcp(addr, name, qtype):
    q = struct.pack(">HHHHHH", 0x4321, 0x0100, 1, 0, 0, 0)
    q += _encode_name(name) + struct.pack(">HH", qtype, 1)
    s = socket.create_connection(addr, timeout=10)
    s.sendall(struct.pack(">H", len(q)) + q)

    def recv_exact(n):
        out = b""
        while len(out) < n:
            chunk = s.recv(n - len(out))
            assert chunk, "server closed TCP connection mid-response"
            out += chunk
        return out

    hdr = recv_exact(2)
    buf = recv_exact(struct.unpack(">H", hdr)[0])
    s.close()
    txid, flags, qd, an, ns, ar = struct.unpack(">HHHHHH", buf[:12])
    assert txid == 0x4321
    return buf, flags, an


def test_truncation_and_tcp_fallback(ensemble, client, binder):
    """>512 B answer sets (easy at 1k records per domain) must come back
    TC-truncated to whole RRs over UDP and complete over TCP on the same
    port."""
    domain = "big.dns.test"
    # 1000 host records behind one domain (VERDICT r1 #7's scale: each A
    # answer is 16 bytes, so 1000 × 16 ≫ 512 for UDP and ~16 KB over TCP)
    NREC = 1000
    path = ra.domain_to_path(domain)
    client.mkdirp(path)
    for i in range(NREC):
        rec = {"type": "host", "address": "10.1.%d.%d" % (i // 250, i % 250)}
        client.create("%s/h%04d" % (path, i), json.dumps(rec).encode(), True)

    buf, flags_rcode, an = dns_query(binder.address, domain, 1)
    flags = struct.unpack(">H", buf[2:4])[0]
    assert flags & 0x0200, "TC bit not set on oversized UDP answer"
    assert len(buf) <= 512
    # the partial UDP payload still parses: only whole RRs included
    answers = parse_answers(buf, an)
    assert 0 < len(answers) < NREC
    assert all(rtype == 1 and len(rdata) == 4 for rtype, _, rdata in answers)

    # TCP retry: the complete set
    buf, flags, an = dns_query_tcp(binder.address, domain, 1)
    assert not (flags & 0x0200)
    assert an == NREC
    answers = parse_answers(buf, an)
    assert len(answers) == NREC
    addrs = {socket.inet_ntoa(rdata) for _, _, rdata in answers}
    assert "10.1.0.0" in addrs and len(addrs) == NREC

    # small answers remain untruncated over UDP
    small = "small.dns.test"
    spath = ra.domain_to_path(small)
    client.mkdirp(spath)
    client.create(spath + "/only", json.dumps({"type": "host", "address": "10.2.0.1"}).encode(), True)
    buf, _, an = dns_query(binder.address, small, 1)
    assert an == 1 and not (struct.unpack(">H", buf[2:4])[0] & 0x0200)


def test_tcp_srv_large(ensemble, client, binder):
    domain = "bigsrv.dns.test"
    path = ra.domain_to_path(domain)
    client.mkdirp(path)
    # service record as registrar writes it: {"type":"service","service":
    # <registration.service verbatim>} (docs/data-format.md)
    svc = {"type": "service",
           "service": {"type": "service",
                       "service": {"srvce": "_x", "proto": "_tcp", "port": 8080, "ttl": 60}}}
    client.set(path, json.dumps(svc).encode(), -1)
    for i in range(80):
        # host-record shape: ports live inside the typed sub-object
        addr = "10.3.0.%d" % (i + 1)
        rec = {"type": "host", "address": addr, "host": {"address": addr, "ports": [9000 + i]}}
        client.create("%s/s%03d" % (path, i), json.dumps(rec).encode(), True)
    buf, flags, an = dns_query_tcp(binder.address, domain, 33)
    assert an == 80
    answers = parse_answers(buf, an)
    ports = set()
    for rtype, _, rdata in answers:
        assert rtype == 33
        _, _, port = struct.unpack(">HHH", rdata[:6])
        ports.add(port)
    assert ports == set(range(9000, 9080))
