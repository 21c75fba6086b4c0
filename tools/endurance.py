#!/usr/bin/env python3
"""Endurance run: the real registrard binary under sustained ensemble chaos.

A 3-server synthetic ensemble (with election pauses) hosts a registrard
subprocess (100 znodes, fast heartbeat, optional GPU health gate). A chaos
loop kills/restarts servers and expires the daemon's session while a
verifier continuously checks convergence: within a bounded window after
each chaos action, all znodes are registered under a live session.

Usage: python tools/endurance.py [--seconds 60] [--gpu] [--znodes 100]
Exit 0 = converged after every chaos action; prints a JSON summary.
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


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--seconds", type=int, default=60)
    ap.add_argument("--znodes", type=int, default=100)
    ap.add_argument("--gpu", action="store_true", help="use the gpu-liveness health gate")
    ap.add_argument("--atomic", action="store_true", help="atomicSwap registration mode")
    args = ap.parse_args()

    ens = ra.Ensemble(servers=3, tick_ms=100, election_ms=200, min_session_timeout_ms=1000)
    ens.start()
    servers = []
    for hp in ens.connect_string().split(","):
        host, port = hp.rsplit(":", 1)
        servers.append({"host": host, "port": int(port)})

    domain = "endure.mi355x"
    cfg = {
        "registration": {
            "domain": domain,
            "type": "host",
            "adminIp": "127.0.0.1",
            "hostname": "e0",
            "settleMs": 0,
            "aliases": ["a%03d.%s" % (i, domain) for i in range(args.znodes - 1)],
            "atomicSwap": bool(args.atomic),
        },
        "zookeeper": {"servers": servers, "timeout": 4000, "connectTimeout": 1000},
        "heartbeatInterval": 500,
        "heartbeat": {"retry": {"maxAttempts": 3, "initialDelay": 100, "maxDelay": 500}},
    }
    if args.gpu:
        cfg["gpuIndex"] = 0
        cfg["healthCheck"] = {"command": "gpu-liveness", "interval": 2000, "timeout": 8000, "threshold": 3}

    cfg_path = tempfile.NamedTemporaryFile(suffix=".json", delete=False, mode="w")
    json.dump(cfg, cfg_path)
    cfg_path.close()
    daemon = os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))), "bin", "registrard")
    proc = subprocess.Popen([daemon, "-f", cfg_path.name], stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL)

    host_node = ra.domain_to_path(domain) + "/e0"

    def converged():
        if ens.ephemeral_count() != args.znodes:
            return False
        info = ens.get(host_node)
        return info["exists"] and info["stat"]["ephemeralOwner"] != 0

    def wait_converged(timeout):
        deadline = time.monotonic() + timeout
        while time.monotonic() < deadline:
            if converged():
                return True
            time.sleep(0.05)
        return False

    assert wait_converged(30), "initial registration failed"

    rng = random.Random(42)
    actions = {"kill": 0, "restart": 0, "expire": 0, "leader": 0}
    failures = []
    t_end = time.monotonic() + args.seconds
    try:
        while time.monotonic() < t_end:
            choice = rng.random()
            if choice < 0.3:
                idx = rng.randrange(3)
                ups = [i for i in range(3) if ens.server_up(i)]
                if ens.server_up(idx) and len(ups) > 1:
                    ens.kill_server(idx)
                    actions["kill"] += 1
            elif choice < 0.55:
                for i in range(3):
                    if not ens.server_up(i):
                        ens.restart_server(i)
                        actions["restart"] += 1
            elif choice < 0.75:
                ups = [i for i in range(3) if ens.server_up(i)]
                if len(ups) > 1:
                    ens.kill_leader()
                    actions["leader"] += 1
            else:
                sids = ens.session_ids()
                if sids:
                    ens.expire_session(rng.choice(sids))
                    actions["expire"] += 1
            # bounded convergence window after every action
            if not wait_converged(30):
                failures.append("not converged after %s at t=%.1f" % (max(actions, key=actions.get),
                                                                      t_end - time.monotonic()))
                break
            time.sleep(rng.uniform(0.2, 0.8))
            if proc.poll() is not None:
                failures.append("daemon exited %d" % proc.returncode)
                break
        # final: restore all servers and verify steady state
        for i in range(3):
            if not ens.server_up(i):
                ens.restart_server(i)
        if not wait_converged(30):
            failures.append("final convergence failed")
    finally:
        proc.send_signal(signal.SIGTERM)
        try:
            proc.wait(timeout=15)
        except subprocess.TimeoutExpired:
            proc.kill()
        os.unlink(cfg_path.name)
        summary = {
            "seconds": args.seconds,
            "znodes": args.znodes,
            "gpu_gate": args.gpu,
            "atomic_swap": bool(args.atomic),
            "actions": actions,
            "failures": failures,
            "daemon_exit": proc.returncode,
        }
        print(json.dumps(summary, indent=2))
        ens.stop()
    return 1 if failures else 0


if __name__ == "__main__":
    sys.exit(main())
