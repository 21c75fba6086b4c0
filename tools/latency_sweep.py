#!/usr/bin/env python3
"""Latency/throughput sweep across znode counts: evidence artifact for
profiles/ (p50/p99 heartbeat RTT and register cycle time at 100 / 1k / 10k
znodes, single process, in-process ensemble)."""
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import registrar_amd as ra  # noqa: E402


def percentile(values, p):
    vs = sorted(values)
    return vs[min(len(vs) - 1, int(round(p / 100.0 * (len(vs) - 1))))]


def sweep(znodes, steps=30):
    ens = ra.Ensemble(servers=1, tick_ms=100, max_session_timeout_ms=60000)
    ens.start()
    host, port = ens.connect_string().rsplit(":", 1)
    c = ra.ZkClient(servers=[(host, int(port))], session_timeout_ms=40000)
    c.start()
    assert c.wait_connected(15000)
    registration = {
        "domain": "sweep%d.test" % znodes,
        "type": "host",
        "adminIp": "127.0.0.1",
        "hostname": "s0",
        "settleMs": 0,
        "aliases": ["a%05d.sweep%d.test" % (i, znodes) for i in range(znodes - 1)],
    }
    prep = ra.PreparedRegistration(json.dumps(registration))
    reg_times, rtts = [], []
    for i in range(steps + 3):
        t0 = time.perf_counter()
        rc, err, zn = prep.register_(c)
        reg_dt = time.perf_counter() - t0
        assert rc == 0, err
        rc, rtt_us = c.heartbeat(zn)
        assert rc == 0
        if i >= 3:  # warmup
            reg_times.append(reg_dt)
            rtts.append(rtt_us)
    c.close()
    ens.stop()
    return {
        "znodes": znodes,
        "steps": steps,
        "register_ms_p50": round(percentile(reg_times, 50) * 1000, 3),
        "register_ms_p99": round(percentile(reg_times, 99) * 1000, 3),
        "regs_per_sec_p50": round(znodes / percentile(reg_times, 50), 1),
        "heartbeat_rtt_ms_p50": round(percentile(rtts, 50) / 1000, 3),
        "heartbeat_rtt_ms_p99": round(percentile(rtts, 99) / 1000, 3),
    }


if __name__ == "__main__":
    results = [sweep(n) for n in (100, 1000, 10000)]
    print(json.dumps({"latency_sweep": results}, indent=2))
