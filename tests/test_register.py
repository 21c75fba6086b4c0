"""Registration engine: domain mapping, payload parity, pipeline, unregister.

Mirrors the reference's test/register.test.js cases (host-only, unregister,
exact payload equality with adminIp and adminIp+ttl, service record) with the
payload expectations copied from those assertions
(test/register.test.js:122-130, 145-153, 176-182)."""
import json

import pytest

import registrar_amd as ra


def reg(client, registration):
    registration = dict(registration)
    registration.setdefault("settleMs", 0)
    rc, err, znodes = ra.register_node(client, json.dumps(registration))
    assert rc == ra.ZOK, err
    return znodes


def test_domain_to_path():
    # reference lib/register.js:34-39
    assert ra.domain_to_path("1.moray.us-east.joyent.com") == "/com/joyent/us-east/moray/1"
    assert ra.domain_to_path("FOO.Example.COM") == "/com/example/foo"
    assert ra.domain_to_path("single") == "/single"


def test_register_host_only(ensemble, client):
    znodes = reg(client, {"domain": "test.laptop.joyent.us", "type": "host", "hostname": "h1"})
    assert znodes == ["/us/joyent/laptop/test/h1"]
    info = ensemble.get(znodes[0])
    assert info["exists"]
    assert info["stat"]["ephemeralOwner"] == client.session_id()


def test_register_unregister(ensemble, client):
    znodes = reg(client, {"domain": "test.laptop.joyent.us", "type": "host", "hostname": "h1"})
    assert ra.unregister_node(client, znodes) == ra.ZOK
    for n in znodes:
        assert not ensemble.get(n)["exists"]


def test_unregister_deletes_all_nodes(ensemble, client):
    # the reference bug deleted only the first node; all must go
    znodes = reg(
        client,
        {
            "domain": "test.laptop.joyent.us",
            "type": "host",
            "hostname": "h1",
            "aliases": ["a1.test.laptop.joyent.us", "a2.test.laptop.joyent.us"],
        },
    )
    assert len(znodes) == 3
    assert ra.unregister_node(client, znodes) == ra.ZOK
    for n in znodes:
        assert not ensemble.get(n)["exists"]
    # idempotent
    assert ra.unregister_node(client, znodes) == ra.ZOK


def test_payload_adminip(ensemble, client):
    # expected shape from reference test/register.test.js:122-130
    znodes = reg(client, {"domain": "test.laptop.joyent.us", "type": "host", "adminIp": "127.0.0.1",
                          "hostname": "h1"})
    obj = json.loads(ensemble.get(znodes[0])["data"])
    assert obj == {"type": "host", "address": "127.0.0.1", "host": {"address": "127.0.0.1"}}


def test_payload_adminip_ttl(ensemble, client):
    # expected shape from reference test/register.test.js:145-153
    znodes = reg(client, {"domain": "test.laptop.joyent.us", "type": "host", "adminIp": "127.0.0.1",
                          "ttl": 120, "hostname": "h1"})
    obj = json.loads(ensemble.get(znodes[0])["data"])
    assert obj == {
        "type": "host",
        "address": "127.0.0.1",
        "host": {"address": "127.0.0.1"},
        "ttl": 120,
    }


SERVICE = {
    "type": "service",
    "service": {"srvce": "_http", "proto": "_tcp", "ttl": 60, "port": 80},
}


def test_payload_service_record(ensemble, client):
    # expected shape from reference test/register.test.js:158-186: the
    # persistent node at the service path carries the registration.service
    # block verbatim, and the service path joins the heartbeat list
    registration = {
        "domain": "test.laptop.joyent.us",
        "type": "host",
        "ttl": 120,
        "adminIp": "127.0.0.1",
        "hostname": "h1",
        "service": SERVICE,
    }
    znodes = reg(client, registration)
    svc_path = ra.domain_to_path("test.laptop.joyent.us")
    assert svc_path in znodes
    obj = json.loads(ensemble.get(svc_path)["data"])
    assert obj == {"type": "service", "service": SERVICE}
    # the service node is persistent, host node ephemeral
    assert ensemble.get(svc_path)["stat"]["ephemeralOwner"] == 0
    host_node = [n for n in znodes if n.endswith("/h1")][0]
    assert ensemble.get(host_node)["stat"]["ephemeralOwner"] != 0


def test_ports_from_service(client, ensemble):
    # ports defaulted from service.service.port (lib/register.js:146-151)
    registration = {"domain": "p.x", "type": "load_balancer", "adminIp": "10.0.0.1", "hostname": "h1",
                    "service": SERVICE}
    znodes = reg(client, registration)
    host_node = [n for n in znodes if n.endswith("/h1")][0]
    obj = json.loads(ensemble.get(host_node)["data"])
    assert obj["load_balancer"]["ports"] == [80]


def test_explicit_ports_win(client, ensemble):
    registration = {"domain": "p.x", "type": "load_balancer", "adminIp": "10.0.0.1", "hostname": "h1",
                    "ports": [80, 443], "service": SERVICE}
    znodes = reg(client, registration)
    host_node = [n for n in znodes if n.endswith("/h1")][0]
    obj = json.loads(ensemble.get(host_node)["data"])
    assert obj["load_balancer"]["ports"] == [80, 443]


def test_service_ttl_defaulted_to_60():
    # lib/register.js:197
    registration = {
        "domain": "a.b",
        "type": "host",
        "service": {"type": "service", "service": {"srvce": "_http", "proto": "_tcp", "port": 80}},
    }
    rec = json.loads(ra.build_service_record(json.dumps(registration)))
    assert rec["service"]["service"]["ttl"] == 60


def test_aliases_reversed_independently(client, ensemble):
    # lib/register.js:217-227: aliases are full domains, each reversed
    registration = {
        "domain": "test.laptop.joyent.us",
        "type": "host",
        "hostname": "h1",
        "aliases": ["alias-1.other.example.com"],
    }
    znodes = reg(client, registration)
    assert "/com/example/other/alias-1" in znodes
    assert ensemble.get("/com/example/other/alias-1")["exists"]


def test_reregister_cleans_previous(client, ensemble):
    registration = {"domain": "re.reg", "type": "host", "hostname": "h1"}
    z1 = reg(client, registration)
    zxid1 = ensemble.get(z1[0])["stat"]["czxid"]
    z2 = reg(client, registration)  # cleanup deletes, then re-creates
    assert z1 == z2
    zxid2 = ensemble.get(z2[0])["stat"]["czxid"]
    assert zxid2 > zxid1  # really re-created, not left over


def test_register_validation_errors():
    with pytest.raises(RuntimeError, match="domain"):
        ra.build_node_list(json.dumps({"type": "host"}))
    with pytest.raises(RuntimeError, match="type"):
        ra.build_node_list(json.dumps({"domain": "a.b"}))
    with pytest.raises(RuntimeError, match="service.type"):
        ra.build_service_record(json.dumps({"domain": "a.b", "type": "host",
                                            "service": {"type": "nope", "service": {}}}))


def test_gpu_extension_payload():
    registration = {"domain": "a.b", "type": "host", "adminIp": "1.2.3.4",
                    "gpu": {"index": 3, "xgmiRank": 5, "uuid": "GPU-abc"}}
    rec = json.loads(ra.build_host_record(json.dumps(registration)))
    assert rec["host"]["gpu"] == {"index": 3, "xgmiRank": 5, "uuid": "GPU-abc"}
    # and without gpu config the key is absent
    rec2 = json.loads(ra.build_host_record(json.dumps({"domain": "a.b", "type": "host", "adminIp": "1.2.3.4"})))
    assert "gpu" not in rec2["host"]


def test_prepared_registration_matches_register_node(ensemble, client):
    import registrar_amd as ra

    registration = {
        "domain": "prep.test",
        "type": "host",
        "adminIp": "127.0.0.1",
        "hostname": "ph",
        "settleMs": 0,
        "aliases": ["a1.prep.test", "a2.prep.test"],
        "service": SERVICE,
    }
    prep = ra.PreparedRegistration(json.dumps(registration))
    assert prep.path == "/test/prep"
    assert len(prep.nodes) == 3
    rc, err, znodes = prep.register_(client)
    assert rc == ra.ZOK, err
    assert "/test/prep" in znodes  # service path joins the heartbeat list
    # same observable result as the one-shot engine call
    obj = json.loads(ensemble.get("/test/prep/ph")["data"])
    assert obj == json.loads(prep.host_payload)
    # idempotent re-register through the same prepared handle
    rc, err, znodes2 = prep.register_(client)
    assert rc == ra.ZOK and znodes2 == znodes


def test_payload_address_fallback_without_adminip(ensemble, client):
    """Without adminIp the payload advertises the first non-loopback IPv4
    interface address (reference lib/register.js:22-31's address()); with no
    such interface it falls back to 127.0.0.1 rather than crashing."""
    import ipaddress

    znodes = reg(client, {"domain": "noadmin.test", "type": "host", "hostname": "nf"})
    obj = json.loads(ensemble.get(znodes[0])["data"])
    addr = obj["address"]
    ipaddress.IPv4Address(addr)  # must be a syntactically valid IPv4 literal
    assert obj["host"]["address"] == addr
