#!/usr/bin/env python3
"""BASELINE config 3 on real hardware: 8 registrard daemons (one per GPU
slot; on a 1-GPU box they all gate on GPU 0) publishing per-GPU host records
with the real rocm-smi/amd-smi liveness gate and xGMI rank in the payload,
a binder-lite resolver answering A/SRV over them, and a kill/recover cycle.
Prints a JSON summary artifact.

Usage: python tools/fleet_gpu_demo.py [--fleet 8] [--seconds 20]
"""
import argparse
import json
import os
import signal
import socket
import struct
import subprocess
import sys
import tempfile
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import registrar_amd as ra  # noqa: E402
from registrar_amd.binder_lite import BinderLite, _encode_name  # noqa: E402

DOMAIN = "workers.mi355x"


def dns_count(addr, name, qtype):
    q = struct.pack(">HHHHHH", 0x55aa, 0x0100, 1, 0, 0, 0)
    q += _encode_name(name) + struct.pack(">HH", qtype, 1)
    s = socket.socket(socket.AF_INET, socket.SOCK_DGRAM)
    s.settimeout(5)
    s.sendto(q, addr)
    buf, _ = s.recvfrom(4096)
    s.close()
    return struct.unpack(">H", buf[6:8])[0]


def wait_for(pred, timeout):
    deadline = time.monotonic() + timeout
    while time.monotonic() < deadline:
        if pred():
            return True
        time.sleep(0.05)
    return False


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--fleet", type=int, default=8)
    ap.add_argument("--seconds", type=int, default=20)
    args = ap.parse_args()

    gpus = ra.discover_gpus("")
    ens = ra.Ensemble(servers=3, tick_ms=100, min_session_timeout_ms=1000)
    ens.start()
    servers = []
    for hp in ens.connect_string().split(","):
        host, port = hp.rsplit(":", 1)
        servers.append({"host": host, "port": int(port)})

    daemon = os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
                          "bin", "registrard")
    tmp = tempfile.mkdtemp(prefix="fleet-gpu-")
    procs = []
    for i in range(args.fleet):
        gi = i % max(1, len(gpus)) if gpus else 0
        cfg = {
            "zookeeper": {"servers": servers, "timeout": 3000, "connectTimeout": 4000},
            "gpuIndex": gi,  # top-level, like etc/config.gpu0.json
            "registration": {
                "domain": DOMAIN, "type": "host", "adminIp": "10.88.0.%d" % (i + 1),
                "hostname": "gpu%d" % i, "settleMs": 0, "ports": [9000 + i],
            },
            "heartbeatInterval": 200,
            "healthCheck": {"command": "gpu-liveness", "interval": 500,
                            "timeout": 5000, "threshold": 2, "period": 60000},
        }
        path = os.path.join(tmp, "cfg%d.json" % i)
        with open(path, "w") as f:
            json.dump(cfg, f)
        procs.append(subprocess.Popen([daemon, "-f", path],
                                      stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL))

    summary = {"fleet": args.fleet, "gpus_discovered": len(gpus), "failures": []}
    zkpath = ra.domain_to_path(DOMAIN)
    try:
        if not wait_for(lambda: len(ens.children(zkpath)) == args.fleet, 60):
            summary["failures"].append("registration incomplete: %r" % ens.children(zkpath))
        else:
            summary["registered"] = sorted(ens.children(zkpath))
            rec = json.loads(ens.get(zkpath + "/gpu0")["data"])
            summary["sample_payload"] = rec

        bl = BinderLite([(s["host"], s["port"]) for s in servers])
        bl.start()
        summary["dns_a_answers"] = dns_count(bl.address, DOMAIN, 1)
        summary["dns_srv_answers"] = dns_count(bl.address, DOMAIN, 33)

        # hard-kill one daemon: record leaves ZK (and so DNS) on session expiry
        procs[2].kill()
        gone = wait_for(lambda: "gpu2" not in ens.children(zkpath), 30)
        if not gone:
            summary["failures"].append("killed daemon's record did not expire")
        summary["dns_a_after_kill"] = dns_count(bl.address, DOMAIN, 1)

        # let the survivors heartbeat under the real GPU gate for a while
        time.sleep(args.seconds)
        alive = ens.children(zkpath)
        summary["survivors_after_%ds" % args.seconds] = sorted(alive)
        if len(alive) != args.fleet - 1:
            summary["failures"].append("survivor set wrong: %r" % alive)
        bl.stop()
    finally:
        for p in procs:
            if p.poll() is None:
                p.send_signal(signal.SIGTERM)
        for p in procs:
            try:
                p.wait(timeout=10)
            except subprocess.TimeoutExpired:
                p.kill()
        ens.stop()

    summary["ok"] = not summary["failures"]
    print(json.dumps(summary, indent=2))
    return 0 if summary["ok"] else 1


if __name__ == "__main__":
    sys.exit(main())
