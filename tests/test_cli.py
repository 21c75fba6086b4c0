"""python -m registrar_amd CLI subcommands."""
import json
import os
import signal
import subprocess
import sys
import time

import pytest

from conftest import REPO_ROOT, orch_config, wait_for

ENV = dict(os.environ, PYTHONPATH=REPO_ROOT)


def run_cli(*args, **kw):
    return subprocess.run([sys.executable, "-m", "registrar_amd", *args],
                          capture_output=True, text=True, env=ENV, timeout=60, **kw)


def test_check_valid(tmp_path, ensemble):
    cfg = orch_config(ensemble, {"domain": "c.test", "type": "host"})
    p = tmp_path / "cfg.json"
    p.write_text(json.dumps(cfg))
    r = run_cli("check", "-f", str(p))
    assert r.returncode == 0 and "ok" in r.stdout


def test_check_invalid(tmp_path):
    p = tmp_path / "bad.json"
    p.write_text(json.dumps({"registration": {"domain": "x", "type": "y"}}))  # no zookeeper
    r = run_cli("check", "-f", str(p))
    assert r.returncode == 1 and "zookeeper" in r.stderr


def test_gpus_runs_anywhere():
    r = run_cli("gpus")
    assert r.returncode == 0  # "no GPUs" on CPU boxes is fine


def test_tree_dump(ensemble, client):
    client.mkdirp("/com/example/svc")
    client.create("/com/example/svc/h1", json.dumps({"type": "host", "address": "1.2.3.4"}).encode(), True)
    r = run_cli("tree", "--servers", ensemble.connect_string())
    assert r.returncode == 0
    assert "h1" in r.stdout and "ephemeral" in r.stdout and "1.2.3.4" in r.stdout


def test_daemon_subcommand(ensemble, tmp_path):
    cfg = orch_config(
        ensemble,
        {"domain": "pycli.test", "type": "host", "hostname": "pyh", "settleMs": 0},
        heartbeatInterval=200,
    )
    p = tmp_path / "cfg.json"
    p.write_text(json.dumps(cfg))
    proc = subprocess.Popen([sys.executable, "-m", "registrar_amd", "daemon", "-f", str(p)],
                            stdout=subprocess.PIPE, text=True, env=ENV)
    try:
        assert wait_for(lambda: ensemble.get("/test/pycli/pyh")["exists"], timeout=15)
    finally:
        proc.send_signal(signal.SIGTERM)
        out, _ = proc.communicate(timeout=15)
    assert proc.returncode == 0
    assert any('"event": "register"' in line for line in out.splitlines())


def test_ensemble_subcommand():
    proc = subprocess.Popen([sys.executable, "-m", "registrar_amd", "ensemble", "-n", "2"],
                            stdout=subprocess.PIPE, text=True, env=ENV)
    try:
        line = proc.stdout.readline()
        info = json.loads(line)
        assert len(info["ports"]) == 2
        import registrar_amd as ra

        c = ra.ZkClient(servers=[("127.0.0.1", info["ports"][0])], connect_max_attempts=3)
        c.start()
        assert c.wait_connected(10000)
        rc, _ = c.create("/via-cli", b"x")
        assert rc == ra.ZOK
        c.close()
    finally:
        proc.send_signal(signal.SIGTERM)
        proc.wait(timeout=15)
    assert proc.returncode == 0


def test_verify_subcommand(ensemble, tmp_path):
    cfg = orch_config(ensemble, {"domain": "v.test", "type": "host", "hostname": "vh",
                                 "adminIp": "127.0.0.1", "settleMs": 0})
    p = tmp_path / "cfg.json"
    p.write_text(json.dumps(cfg))
    r = run_cli("verify", "-f", str(p))
    assert r.returncode == 0, r.stdout + r.stderr
    out = json.loads(r.stdout)
    assert out["verify"] == "ok"
    assert out["znodes"] == ["/test/v/vh"]
    assert out["heartbeat_rtt_ms"] > 0
    # nothing left behind
    assert not ensemble.get("/test/v/vh")["exists"]


def test_verify_unreachable(tmp_path):
    from conftest import free_port

    cfg = {"registration": {"domain": "x.y", "type": "host"},
           "zookeeper": {"servers": [{"host": "127.0.0.1", "port": free_port()}],
                         "timeout": 2000, "connectTimeout": 200}}
    p = tmp_path / "cfg.json"
    p.write_text(json.dumps(cfg))
    r = run_cli("verify", "-f", str(p), "--timeout", "10", "--attempts", "2")
    assert r.returncode == 1
    assert "connect-failed" in r.stderr


def test_check_rejects_bad_stdout_match(tmp_path):
    """Fail-fast regex validation at config time."""
    base = {"zookeeper": {"servers": [{"host": "127.0.0.1", "port": 2181}]},
            "registration": {"domain": "a.b", "type": "host"}}
    bad_pat = dict(base, healthCheck={"command": "true", "stdoutMatch": {"pattern": "(unclosed"}})
    p = tmp_path / "badpat.json"
    p.write_text(json.dumps(bad_pat))
    r = run_cli("check", "-f", str(p))
    assert r.returncode == 1 and "invalid regex" in r.stderr

    bad_flag = dict(base, healthCheck={"command": "true",
                                       "stdoutMatch": {"pattern": "x", "flags": "su"}})
    p2 = tmp_path / "badflag.json"
    p2.write_text(json.dumps(bad_flag))
    r = run_cli("check", "-f", str(p2))
    assert r.returncode == 1 and "unsupported flag" in r.stderr


def test_binder_subcommand(ensemble, client):
    """README quickstart: `python -m registrar_amd binder` answers DNS from
    the registration tree; SIGTERM exits 0."""
    import signal as _signal
    import socket
    import struct
    import subprocess
    import sys

    import registrar_amd as ra

    reg = {"domain": "cli.binder.test", "type": "host", "adminIp": "10.5.0.9",
           "hostname": "cb0", "settleMs": 0}
    rc, err, _ = ra.register_node(client, json.dumps(reg))
    assert rc == ra.ZOK

    proc = subprocess.Popen(
        [sys.executable, "-m", "registrar_amd", "binder",
         "--servers", ensemble.connect_string(), "--port", "0"],
        stdout=subprocess.PIPE, stderr=subprocess.DEVNULL, text=True, cwd=REPO_ROOT)
    try:
        line = proc.stdout.readline()
        host, port = json.loads(line)["dns"].rsplit(":", 1)
        q = struct.pack(">HHHHHH", 0x31ca, 0x0100, 1, 0, 0, 0)
        for label in "cli.binder.test".split("."):
            q += bytes([len(label)]) + label.encode()
        q += b"\x00" + struct.pack(">HH", 1, 1)
        s = socket.socket(socket.AF_INET, socket.SOCK_DGRAM)
        s.settimeout(5)
        s.sendto(q, (host, int(port)))
        buf, _ = s.recvfrom(512)
        s.close()
        assert struct.unpack(">H", buf[6:8])[0] == 1  # one A answer
        assert socket.inet_ntoa(buf[-4:]) == "10.5.0.9"
    finally:
        proc.send_signal(_signal.SIGTERM)
        assert proc.wait(timeout=10) == 0


def test_consumer_example(ensemble, client):
    """examples/consumer.py lists live instances with their xGMI rank — the
    documented Binder-style consumer pattern."""
    import registrar_amd as ra

    reg = {"domain": "svc.consumer.test", "type": "host", "adminIp": "10.6.0.2",
           "hostname": "cx0", "settleMs": 0,
           "gpu": {"index": 3, "xgmiRank": 5, "uuid": "GPU-test"},
           "service": {"type": "service",
                       "service": {"srvce": "_infer", "proto": "_tcp", "port": 8000, "ttl": 30}}}
    rc, err, _ = ra.register_node(client, json.dumps(reg))
    assert rc == ra.ZOK
    r = subprocess.run(
        [sys.executable, os.path.join(REPO_ROOT, "examples", "consumer.py"),
         "--servers", ensemble.connect_string(), "svc.consumer.test"],
        capture_output=True, text=True, timeout=60, cwd=REPO_ROOT)
    assert r.returncode == 0, r.stderr
    assert "1 live instance(s)" in r.stdout
    assert "10.6.0.2" in r.stdout and "xgmiRank=5" in r.stdout
    assert "SRV _infer._tcp.svc.consumer.test port=8000" in r.stdout


def test_shipped_sample_configs_are_valid():
    """Every etc/*.json sample must pass `registrard check`-level validation
    (schema drift in shipped samples is a doc bug users hit first)."""
    import glob

    samples = sorted(glob.glob(os.path.join(REPO_ROOT, "etc", "*.json")))
    assert samples, "no sample configs found"
    for path in samples:
        r = run_cli("check", "-f", path)
        assert r.returncode == 0, "%s failed check: %s" % (path, r.stderr)
