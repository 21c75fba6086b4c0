"""GPU integration: KFD topology discovery, xGMI rank, GPU-gated health check.

The fake-sysfs tests run anywhere; the @pytest.mark.gpu tests need a real
AMD GPU box (MI355X) and verify the real discovery + rocm-smi health gate +
the per-GPU registration payload (BASELINE config 3)."""
import json
import os

import pytest

import registrar_amd as ra
from conftest import orch_config, wait_for


# ---------- fake-sysfs tests (CPU) ----------

def make_fake_kfd(tmp_path, nodes):
    """nodes: list of dicts with properties key/values."""
    root = tmp_path / "nodes"
    root.mkdir()
    for i, props in enumerate(nodes):
        d = root / str(i)
        d.mkdir()
        with open(d / "properties", "w") as f:
            for k, v in props.items():
                if k == "name":
                    continue
                f.write("%s %d\n" % (k, v))
        with open(d / "name", "w") as f:
            f.write(props.get("name", ""))
    return str(root)


def test_discover_fake_8gpu_hive(tmp_path):
    # 1 CPU node + 8 GPUs in one xGMI hive (the MI355X 8-GPU node shape)
    nodes = [{"simd_count": 0, "cpu_cores_count": 96}]
    for i in range(8):
        nodes.append({"simd_count": 1024, "hive_id": 0xABCD, "location_id": 0x1000 + i,
                      "unique_id": 0x1111 + i, "name": "gfx950"})
    root = make_fake_kfd(tmp_path, nodes)
    gpus = ra.discover_gpus(root)
    assert len(gpus) == 8
    for i, g in enumerate(gpus):
        assert g["device_index"] == i
        assert g["xgmi_rank"] == i
        assert g["hive_id"] == 0xABCD
        assert g["name"] == "gfx950"
        assert g["uuid"].startswith("GPU-")


def test_discover_fake_two_hives(tmp_path):
    nodes = [{"simd_count": 0}]
    for i in range(4):
        nodes.append({"simd_count": 512, "hive_id": 1})
    for i in range(4):
        nodes.append({"simd_count": 512, "hive_id": 2})
    root = make_fake_kfd(tmp_path, nodes)
    gpus = ra.discover_gpus(root)
    assert [g["xgmi_rank"] for g in gpus] == [0, 1, 2, 3, 0, 1, 2, 3]


def test_discover_fake_no_hive(tmp_path):
    root = make_fake_kfd(tmp_path, [{"simd_count": 256}])
    gpus = ra.discover_gpus(root)
    assert len(gpus) == 1 and gpus[0]["xgmi_rank"] == 0 and gpus[0]["hive_id"] == 0


def test_discover_empty(tmp_path):
    assert ra.discover_gpus(str(tmp_path / "missing")) == []


def test_gpu_health_command_shape():
    cmd = ra.gpu_health_command(0)
    assert isinstance(cmd, str) and cmd


# ---------- real-GPU tests (MI355X box) ----------

@pytest.mark.gpu
def test_real_gpu_discovery():
    gpus = ra.discover_gpus("")
    assert len(gpus) >= 1, "no GPUs visible through KFD on a GPU box"
    g = gpus[0]
    assert g["device_index"] == 0
    assert g["xgmi_rank"] >= 0
    assert ra.gpu_count() == len(gpus)
    assert ra.xgmi_local_rank(0) == g["xgmi_rank"]
    assert ra.gpu_alive(0)
    assert not ra.gpu_alive(len(gpus))


@pytest.mark.gpu
def test_real_gpu_health_command_passes():
    cmd = ra.gpu_health_command(0)
    # cold boxes can take tens of seconds on the FIRST rocm-smi invocation
    # (driver/library paging); warm up once, then assert with headroom
    ra.exec_with_timeout(cmd, 120000)
    res = ra.exec_with_timeout(cmd, 60000)
    assert res["exit_status"] == 0 and not res["timed_out"], res


@pytest.mark.gpu
def test_real_gpu_torch_sees_device():
    torch = pytest.importorskip("torch")
    assert torch.cuda.is_available()
    assert torch.cuda.device_count() == ra.gpu_count()


@pytest.mark.gpu
def test_gpu_gated_registration(ensemble):
    """BASELINE config 3: per-GPU registrar with rocm-smi health gate and
    xGMI rank advertised in the payload."""
    registration = {
        "domain": "gpu0.mi355x.test",
        "type": "host",
        "adminIp": "127.0.0.1",
        "hostname": "g0",
        "settleMs": 0,
        "service": {"type": "service", "service": {"srvce": "_infer", "proto": "_tcp", "port": 8000}},
    }
    cfg = orch_config(
        ensemble,
        registration,
        gpuIndex=0,
        heartbeatInterval=200,
        healthCheck={"command": "gpu-liveness", "interval": 1000, "timeout": 30000, "threshold": 3},
    )
    o = ra.Orchestrator(json.dumps(cfg))
    o.start()
    assert o.wait_registered(30000)
    znodes = o.znodes()
    host_node = [n for n in znodes if n.endswith("/g0")][0]
    payload = json.loads(ensemble.get(host_node)["data"])
    gpu = payload["host"]["gpu"]
    assert gpu["index"] == 0
    assert gpu["xgmiRank"] == ra.xgmi_local_rank(0)
    # the health gate actually ran rocm-smi and stayed up
    import time

    time.sleep(1.5)
    assert all(ensemble.get(n)["exists"] for n in znodes)
    assert o.metrics()["unregisters"] == 0
    o.stop()
