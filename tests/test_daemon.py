"""registrard daemon-level integration: spawn the real binary against an
in-process ensemble and observe its bunyan log stream and ZK effects.

The reference has no daemon-level tests at all; its SMF manifest contract
(exit on expiry → restarter relaunches) is verified here
with --exit-on-expiry."""
import json
import os
import signal
import subprocess
import time

import pytest

import registrar_amd as ra
from conftest import REPO_ROOT, orch_config, wait_for


def spawn_daemon(daemon_bin, cfg, *args):
    cfg_path = os.path.join("/tmp", "registrard-test-%d.json" % os.getpid())
    with open(cfg_path, "w") as f:
        json.dump(cfg, f)
    proc = subprocess.Popen(
        [daemon_bin, "-f", cfg_path, *args],
        stdout=subprocess.PIPE,
        stderr=subprocess.STDOUT,
        text=True,
    )
    return proc, cfg_path


def read_logs(proc):
    out, _ = proc.communicate(timeout=10)
    recs = []
    for line in out.splitlines():
        try:
            recs.append(json.loads(line))
        except ValueError:
            pass
    return recs


def test_daemon_registers_and_heartbeats(ensemble, daemon_bin):
    cfg = orch_config(
        ensemble,
        {"domain": "d.test", "type": "host", "hostname": "dh", "adminIp": "127.0.0.1", "settleMs": 0},
        heartbeatInterval=100,
    )
    proc, cfg_path = spawn_daemon(daemon_bin, cfg)
    try:
        assert wait_for(lambda: ensemble.get("/test/d/dh")["exists"], timeout=10)
        payload = json.loads(ensemble.get("/test/d/dh")["data"])
        assert payload == {"type": "host", "address": "127.0.0.1", "host": {"address": "127.0.0.1"}}
        time.sleep(0.5)
    finally:
        proc.send_signal(signal.SIGTERM)
        recs = read_logs(proc)
        os.unlink(cfg_path)
    assert proc.returncode == 0
    msgs = [r["msg"] for r in recs]
    assert "registrar: registered" in msgs
    assert any(r.get("name") == "registrar" and "hostname" in r and "pid" in r and "v" in r for r in recs), \
        "bunyan record shape missing"


def test_daemon_exit_on_expiry(ensemble, daemon_bin):
    cfg = orch_config(
        ensemble,
        {"domain": "e.test", "type": "host", "hostname": "eh", "settleMs": 0},
        heartbeatInterval=100,
    )
    proc, cfg_path = spawn_daemon(daemon_bin, cfg, "--exit-on-expiry")
    try:
        assert wait_for(lambda: ensemble.get("/test/e/eh")["exists"], timeout=10)
        sid = ensemble.get("/test/e/eh")["stat"]["ephemeralOwner"]
        ensemble.expire_session(sid)
        assert wait_for(lambda: proc.poll() is not None, timeout=15)
    finally:
        if proc.poll() is None:
            proc.kill()
        recs = read_logs(proc)
        os.unlink(cfg_path)
    # exit(1) on expiry, with the reference's fatal message contract
    # (main.js:141-144)
    assert proc.returncode == 1
    assert any("session_expired" in r["msg"] for r in recs)


def test_daemon_reregisters_in_process_on_expiry(ensemble, daemon_bin):
    cfg = orch_config(
        ensemble,
        {"domain": "r.test", "type": "host", "hostname": "rh", "settleMs": 0},
        heartbeatInterval=100,
    )
    proc, cfg_path = spawn_daemon(daemon_bin, cfg)
    try:
        assert wait_for(lambda: ensemble.get("/test/r/rh")["exists"], timeout=10)
        sid1 = ensemble.get("/test/r/rh")["stat"]["ephemeralOwner"]
        ensemble.expire_session(sid1)
        # default policy: daemon stays up, re-registers under a new session
        assert wait_for(
            lambda: ensemble.get("/test/r/rh")["exists"]
            and ensemble.get("/test/r/rh")["stat"]["ephemeralOwner"] not in (0, sid1),
            timeout=15,
        )
        assert proc.poll() is None
    finally:
        proc.send_signal(signal.SIGTERM)
        read_logs(proc)
        os.unlink(cfg_path)
    assert proc.returncode == 0


def test_daemon_bad_config(daemon_bin):
    proc = subprocess.Popen([daemon_bin, "-f", "/nonexistent.json"], stdout=subprocess.PIPE, text=True)
    out, _ = proc.communicate(timeout=10)
    assert proc.returncode == 1
    assert "unable to read configuration" in out


def test_daemon_usage():
    daemon = os.path.join(REPO_ROOT, "bin", "registrard")
    if not os.path.exists(daemon):
        pytest.skip("daemon not built")
    proc = subprocess.run([daemon, "-h"], capture_output=True, text=True, timeout=10)
    assert proc.returncode == 0
    assert "usage:" in proc.stdout


def test_daemon_sigusr1_metrics(ensemble, daemon_bin):
    cfg = orch_config(
        ensemble,
        {"domain": "m.test", "type": "host", "hostname": "mh", "settleMs": 0},
        heartbeatInterval=100,
    )
    proc, cfg_path = spawn_daemon(daemon_bin, cfg)
    try:
        assert wait_for(lambda: ensemble.get("/test/m/mh")["exists"], timeout=10)
        time.sleep(0.5)
        proc.send_signal(signal.SIGUSR1)
        time.sleep(0.5)
    finally:
        proc.send_signal(signal.SIGTERM)
        recs = read_logs(proc)
        os.unlink(cfg_path)
    metrics = [r for r in recs if r["msg"] == "registrar: metrics"]
    assert metrics, "no metrics record after SIGUSR1"
    assert metrics[0]["registers"] >= 1
    assert metrics[0]["heartbeats"] >= 1
    assert metrics[0]["p50HeartbeatRttUs"] > 0


def test_debug_logs_carry_src(ensemble, daemon_bin):
    """bunyan src parity (reference main.js:75-76): at debug verbosity every
    record carries the emitting file:line."""
    cfg = orch_config(
        ensemble,
        {"domain": "src.test", "type": "host", "hostname": "sh", "settleMs": 0},
        heartbeatInterval=100,
        logLevel="debug",
    )
    proc, cfg_path = spawn_daemon(daemon_bin, cfg)
    try:
        time.sleep(1.0)
    finally:
        proc.send_signal(signal.SIGTERM)
    recs = read_logs(proc)
    os.unlink(cfg_path)
    assert recs, "no log records"
    with_src = [r for r in recs if "src" in r]
    assert with_src, "no record carries src at debug level"
    s = with_src[0]["src"]
    assert s["file"].endswith(".cpp") or s["file"].endswith(".hpp")
    assert isinstance(s["line"], int) and s["line"] > 0
    # at default (info) verbosity src is absent
    cfg.pop("logLevel")
    proc, cfg_path = spawn_daemon(daemon_bin, cfg)
    try:
        time.sleep(0.8)
    finally:
        proc.send_signal(signal.SIGTERM)
    recs = read_logs(proc)
    os.unlink(cfg_path)
    assert recs and all("src" not in r for r in recs)


def test_verbose_flag_steps_log_level(ensemble, daemon_bin):
    """-v lowers the level one step per repeat (reference main.js:66-76:
    bunyan level minus 10 each): info → debug with one -v, visible as debug
    records (which also carry src at that point)."""
    cfg = orch_config(
        ensemble,
        {"domain": "vflag.test", "type": "host", "hostname": "vh", "settleMs": 0},
        heartbeatInterval=100,
    )
    proc, cfg_path = spawn_daemon(daemon_bin, cfg, "-v")
    try:
        time.sleep(0.9)
    finally:
        proc.send_signal(signal.SIGTERM)
    recs = read_logs(proc)
    os.unlink(cfg_path)
    assert any(r["level"] == 20 for r in recs), "one -v must enable debug (level 20) records"
    assert not any(r["level"] == 10 for r in recs), "one -v must not reach trace"


def test_zkensembled_binary_quickstart():
    """The README quickstart path: spawn the standalone synthetic ensemble
    binary, parse its JSON ports line, speak ZK to it, shut it down clean."""
    bin_path = os.path.join(REPO_ROOT, "bin", "zkensembled")
    assert os.path.exists(bin_path), "run `make daemon` first"
    proc = subprocess.Popen([bin_path, "-n", "3", "--tick-ms", "100"],
                            stdout=subprocess.PIPE, stderr=subprocess.DEVNULL, text=True)
    try:
        line = proc.stdout.readline()
        info = json.loads(line)
        assert len(info["ports"]) == 3
        servers = [("127.0.0.1", p) for p in info["ports"]]
        import registrar_amd as ra

        c = ra.ZkClient(servers=servers, session_timeout_ms=10000)
        c.start()
        assert c.wait_connected(15000)
        rc, _ = c.create("/standalone", b"x", True)
        assert rc == ra.ZOK
        rc, data, _ = c.get("/standalone")
        assert rc == ra.ZOK and data == b"x"
        c.close()
    finally:
        proc.send_signal(signal.SIGTERM)
        assert proc.wait(timeout=10) == 0
