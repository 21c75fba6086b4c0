#!/usr/bin/env python3
"""Fleet endurance: 8 registrard daemons (one per GPU slot, BASELINE
config 3) under sustained ensemble chaos — server kills, leader kills with
election pauses, and random session expiries — with a convergence verifier
that requires the FULL fleet (all 8 host records, each owned by a live
session) back within a bounded window after every chaos action.

This is the multi-instance counterpart of tools/endurance.py: an expiry
kills one daemon's ephemerals and that daemon alone must re-register while
its 7 siblings stay registered throughout.

Usage: python tools/fleet_endurance.py [--seconds 120] [--fleet 8] [--gpu]
Exit 0 = converged after every action; prints a JSON summary.
"""
import argparse
import json
import os
import random
import signal
import subprocess
import sys
import tempfile
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import registrar_amd as ra  # noqa: E402

DOMAIN = "fleet.endure.mi355x"


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--seconds", type=int, default=120)
    ap.add_argument("--fleet", type=int, default=8)
    ap.add_argument("--gpu", action="store_true", help="gpu-liveness health gate")
    args = ap.parse_args()

    ens = ra.Ensemble(servers=3, tick_ms=100, election_ms=200, min_session_timeout_ms=1000)
    ens.start()
    servers = []
    for hp in ens.connect_string().split(","):
        host, port = hp.rsplit(":", 1)
        servers.append({"host": host, "port": int(port)})

    daemon = os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
                          "bin", "registrard")
    tmp = tempfile.mkdtemp(prefix="fleet-endure-")
    procs = []
    for i in range(args.fleet):
        cfg = {
            "zookeeper": {"servers": servers, "timeout": 4000, "connectTimeout": 1000},
            "registration": {
                "domain": DOMAIN, "type": "host", "adminIp": "10.99.0.%d" % (i + 1),
                "hostname": "gpu%d" % i, "settleMs": 0, "ports": [9000 + i],
            },
            "heartbeatInterval": 500,
            "heartbeat": {"retry": {"maxAttempts": 3, "initialDelay": 100, "maxDelay": 500}},
        }
        if args.gpu:
            cfg["gpuIndex"] = 0
            cfg["healthCheck"] = {"command": "gpu-liveness", "interval": 2000,
                                  "timeout": 8000, "threshold": 3}
        path = os.path.join(tmp, "cfg%d.json" % i)
        with open(path, "w") as f:
            json.dump(cfg, f)
        procs.append(subprocess.Popen([daemon, "-f", path],
                                      stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL))

    zkpath = ra.domain_to_path(DOMAIN)
    want = {"gpu%d" % i for i in range(args.fleet)}

    def converged():
        names = set(ens.children(zkpath))
        if names != want:
            return False
        for n in names:
            info = ens.get("%s/%s" % (zkpath, n))
            if not info["exists"] or info["stat"]["ephemeralOwner"] == 0:
                return False
        return True

    def wait_converged(timeout):
        deadline = time.monotonic() + timeout
        while time.monotonic() < deadline:
            if converged():
                return True
            time.sleep(0.05)
        return False

    summary = {"seconds": args.seconds, "fleet": args.fleet, "gpu_gate": args.gpu,
               "actions": {"kill": 0, "restart": 0, "expire": 0, "leader": 0},
               "failures": []}
    actions = summary["actions"]
    failures = summary["failures"]
    rng = random.Random(77)
    try:
        if not wait_converged(60):
            failures.append("initial fleet registration failed: %r" % sorted(ens.children(zkpath)))
        t_end = time.monotonic() + args.seconds
        while not failures and time.monotonic() < t_end:
            choice = rng.random()
            if choice < 0.25:
                idx = rng.randrange(3)
                ups = [i for i in range(3) if ens.server_up(i)]
                if ens.server_up(idx) and len(ups) > 1:
                    ens.kill_server(idx)
                    actions["kill"] += 1
            elif choice < 0.5:
                for i in range(3):
                    if not ens.server_up(i):
                        ens.restart_server(i)
                        actions["restart"] += 1
            elif choice < 0.7:
                ups = [i for i in range(3) if ens.server_up(i)]
                if len(ups) > 1:
                    ens.kill_leader()
                    actions["leader"] += 1
            else:
                sids = ens.session_ids()
                if sids:
                    # expire a RANDOM fleet member's session: exactly that
                    # daemon must re-register
                    ens.expire_session(rng.choice(sids))
                    actions["expire"] += 1
            if not wait_converged(30):
                failures.append("fleet not converged (have %r)" % sorted(ens.children(zkpath)))
                break
            for i, p in enumerate(procs):
                if p.poll() is not None:
                    failures.append("daemon %d exited %d" % (i, p.returncode))
            time.sleep(rng.uniform(0.1, 0.5))
        for i in range(3):
            if not ens.server_up(i):
                ens.restart_server(i)
        if not failures and not wait_converged(30):
            failures.append("final convergence failed")
    finally:
        for p in procs:
            if p.poll() is None:
                p.send_signal(signal.SIGTERM)
        for p in procs:
            try:
                p.wait(timeout=15)
            except subprocess.TimeoutExpired:
                p.kill()
        ens.stop()

    summary["ok"] = not failures
    print(json.dumps(summary, indent=2))
    return 0 if summary["ok"] else 1


if __name__ == "__main__":
    sys.exit(main())
