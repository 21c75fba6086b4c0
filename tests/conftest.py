import json
import os
import socket
import subprocess
import sys

import pytest

REPO_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO_ROOT)


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: test requires an AMD GPU (run on an MI355X box)")
    config.addinivalue_line(
        "markers",
        "zk_real: interop test against a REAL Apache ZooKeeper at $ZK_HOST:$ZK_PORT "
        "(skipped when unset; no JVM exists in the build container)")


@pytest.fixture
def ensemble():
    """A started 1-server synthetic ensemble with fast ticks."""
    import registrar_amd as ra

    ens = ra.Ensemble(servers=1, tick_ms=50, min_session_timeout_ms=200)
    ens.start()
    yield ens
    ens.stop()


@pytest.fixture
def ensemble3():
    """A started 3-server synthetic ensemble."""
    import registrar_amd as ra

    ens = ra.Ensemble(servers=3, tick_ms=50, min_session_timeout_ms=200)
    ens.start()
    yield ens
    ens.stop()


def make_client(ens, **kw):
    import registrar_amd as ra

    servers = []
    for hp in ens.connect_string().split(","):
        host, port = hp.rsplit(":", 1)
        servers.append((host, int(port)))
    kw.setdefault("session_timeout_ms", 5000)
    c = ra.ZkClient(servers=servers, **kw)
    c.start()
    assert c.wait_connected(10000), "client failed to connect to the ensemble"
    return c


@pytest.fixture
def client(ensemble):
    c = make_client(ensemble)
    yield c
    c.close()


def free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


@pytest.fixture(scope="session")
def daemon_bin():
    """Path to the registrard binary; make keeps it current vs csrc/."""
    subprocess.run(["make", "daemon"], cwd=REPO_ROOT, check=True, capture_output=True)
    return os.path.join(REPO_ROOT, "bin", "registrard")


def orch_config(ens, registration, **extra):
    """Build a daemon/orchestrator config JSON dict for a live ensemble."""
    servers = []
    for hp in ens.connect_string().split(","):
        host, port = hp.rsplit(":", 1)
        servers.append({"host": host, "port": int(port)})
    cfg = {
        "registration": registration,
        "zookeeper": {"servers": servers, "timeout": 5000, "connectTimeout": 2000},
    }
    cfg.update(extra)
    return cfg


def wait_for(predicate, timeout=10.0, interval=0.02):
    import time

    deadline = time.monotonic() + timeout
    while time.monotonic() < deadline:
        if predicate():
            return True
        time.sleep(interval)
    return False
