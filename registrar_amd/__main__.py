"""python -m registrar_amd — operational CLI over the native core.

Subcommands:
  daemon   -f CONFIG [-v] [--exit-on-expiry]   run a registrar (same contract
                                               as the registrard binary)
  ensemble [-n N] [-p PORT ...]                run a synthetic ZK ensemble
  tree     --servers HOST:PORT[,...] [PATH]    dump a ZK subtree (Binder-style
                                               reader; works against the
                                               synthetic ensemble or real ZK)
  gpus                                         show KFD GPU topology/xGMI ranks
  check    -f CONFIG                           validate a config file
  binder   --servers HOST:PORT[,...] [--port N]  serve DNS A/SRV answers from
                                               the registration tree
                                               (binder_lite)
  verify   -f CONFIG                           pre-deployment smoke: connect,
                                               register, heartbeat, unregister
"""
import argparse
import json
import signal
import sys
import time


def cmd_daemon(args):
    import registrar_amd as ra

    with open(args.file) as f:
        cfg = json.load(f)
    if args.exit_on_expiry:
        cfg["exitOnExpiry"] = True
    level = "info"
    if cfg.get("logLevel"):
        level = cfg["logLevel"]
    for _ in range(args.verbose or 0):
        order = ["fatal", "error", "warn", "info", "debug", "trace"]
        level = order[min(order.index(level) + 1, len(order) - 1)] if level in order else "debug"
    orch = ra.Orchestrator(json.dumps(cfg), log_level=level)
    stop = []
    signal.signal(signal.SIGINT, lambda *_: stop.append(1))
    signal.signal(signal.SIGTERM, lambda *_: stop.append(1))
    orch.start()
    def drain():
        for ev in orch.poll_events():
            print(json.dumps({"event": ev["type"], "detail": ev["detail"], "znodes": ev["znodes"]}),
                  flush=True)

    try:
        while not stop:
            drain()
            if orch.expired():
                print(json.dumps({"event": "sessionExpired", "detail": "exiting (exitOnExpiry)"}), flush=True)
                return 1
            time.sleep(0.1)
    finally:
        orch.stop()
        drain()
    return 0


def cmd_ensemble(args):
    import registrar_amd as ra

    ens = ra.Ensemble(servers=args.n, ports=args.port or [], log_level="info" if args.verbose else "warn")
    ens.start()
    print(json.dumps({"ports": ens.ports(), "connect": ens.connect_string()}), flush=True)
    stop = []
    signal.signal(signal.SIGINT, lambda *_: stop.append(1))
    signal.signal(signal.SIGTERM, lambda *_: stop.append(1))
    try:
        while not stop:
            time.sleep(0.1)
    finally:
        ens.stop()
    return 0


def _parse_servers(spec):
    servers = []
    for hp in spec.split(","):
        host, port = hp.rsplit(":", 1)
        servers.append((host, int(port)))
    return servers


def cmd_tree(args):
    import registrar_amd as ra

    c = ra.ZkClient(servers=_parse_servers(args.servers), connect_max_attempts=3)
    c.start()
    if not c.wait_connected(10000):
        print("error: could not connect to %s" % args.servers, file=sys.stderr)
        return 1

    def walk(path, depth):
        rc, data, st = c.get(path)
        if rc != 0:
            return
        label = path if depth == 0 else path.rsplit("/", 1)[1]
        eph = " [ephemeral 0x%x]" % st["ephemeralOwner"] if st["ephemeralOwner"] else ""
        body = ""
        if data:
            try:
                body = " " + json.dumps(json.loads(data))
            except ValueError:
                body = " <%d bytes>" % len(data)
        print("%s%s%s%s" % ("  " * depth, label, eph, body))
        rc, children = c.get_children(path)
        if rc == 0:
            base = path.rstrip("/")
            for ch in children:
                walk("%s/%s" % (base, ch), depth + 1)

    walk(args.path, 0)
    c.close()
    return 0


def cmd_gpus(_args):
    import registrar_amd as ra

    gpus = ra.discover_gpus("")
    if not gpus:
        print("no GPUs visible through KFD")
        return 0
    for g in gpus:
        print("gpu %d: kfd_node=%d hive=0x%x xgmi_rank=%d name=%s uuid=%s"
              % (g["device_index"], g["kfd_node"], g["hive_id"], g["xgmi_rank"], g["name"], g["uuid"]))
        print("  health command: %s" % ra.gpu_health_command(g["device_index"]))
    return 0


def cmd_binder(args):
    from registrar_amd.binder_lite import BinderLite

    b = BinderLite(_parse_servers(args.servers), host=args.host, port=args.port)
    b.start()
    print(json.dumps({"dns": "%s:%d" % b.address}), flush=True)
    stop = []
    signal.signal(signal.SIGINT, lambda *_: stop.append(1))
    signal.signal(signal.SIGTERM, lambda *_: stop.append(1))
    try:
        while not stop:
            time.sleep(0.1)
    finally:
        b.stop()
    return 0


def cmd_verify(args):
    """Register → heartbeat → unregister once against the real ensemble in
    the config, then exit. Safe: leaves no state behind (its ephemerals are
    removed explicitly and die with the session regardless)."""
    import registrar_amd as ra

    with open(args.file) as f:
        cfg = json.load(f)
    try:
        ra.Orchestrator(json.dumps(cfg))  # schema validation up front
    except RuntimeError as e:
        print(json.dumps({"verify": "invalid-config", "error": str(e)}), file=sys.stderr)
        return 1
    zk = cfg["zookeeper"]
    client = ra.ZkClient(
        servers=[(s["host"], int(s["port"])) for s in zk["servers"]],
        session_timeout_ms=int(zk.get("timeout", 30000)),
        connect_timeout_ms=int(zk.get("connectTimeout", 4000)),
        connect_max_attempts=args.attempts,
    )
    client.start()
    t0 = time.monotonic()
    if not client.wait_connected(args.timeout * 1000):
        print(json.dumps({"verify": "connect-failed"}), file=sys.stderr)
        client.close()
        return 1
    connect_ms = round((time.monotonic() - t0) * 1000, 1)
    reg = dict(cfg["registration"])
    if cfg.get("adminIp") and not reg.get("adminIp"):
        reg["adminIp"] = cfg["adminIp"]
    rc, err, znodes = ra.register_node(client, json.dumps(reg))
    if rc != 0:
        print(json.dumps({"verify": "register-failed", "error": err}), file=sys.stderr)
        client.close()
        return 1
    hb_rc, rtt_us = client.heartbeat(znodes)
    un_rc = ra.unregister_node(client, znodes)
    client.close()
    ok = hb_rc == 0 and un_rc == 0
    print(json.dumps({
        "verify": "ok" if ok else "failed",
        "connect_ms": connect_ms,
        "znodes": znodes,
        "heartbeat_rtt_ms": round(rtt_us / 1000.0, 3),
        "heartbeat": ra.error_name(hb_rc),
        "unregister": ra.error_name(un_rc),
    }))
    return 0 if ok else 1


def cmd_check(args):
    import registrar_amd as ra

    with open(args.file) as f:
        cfg = json.load(f)
    try:
        ra.Orchestrator(json.dumps(cfg))  # constructor validates (§2.5 schema)
    except RuntimeError as e:
        print("invalid: %s" % e, file=sys.stderr)
        return 1
    print("ok: %s" % args.file)
    return 0


def main(argv=None):
    ap = argparse.ArgumentParser(prog="python -m registrar_amd")
    sub = ap.add_subparsers(dest="cmd", required=True)

    d = sub.add_parser("daemon", help="run a registrar daemon")
    d.add_argument("-f", "--file", required=True)
    d.add_argument("-v", "--verbose", action="count", default=0)
    d.add_argument("--exit-on-expiry", action="store_true")
    d.set_defaults(fn=cmd_daemon)

    e = sub.add_parser("ensemble", help="run a synthetic ZK ensemble")
    e.add_argument("-n", type=int, default=1)
    e.add_argument("-p", "--port", type=int, action="append")
    e.add_argument("-v", "--verbose", action="store_true")
    e.set_defaults(fn=cmd_ensemble)

    t = sub.add_parser("tree", help="dump a ZK subtree")
    t.add_argument("--servers", required=True, help="host:port[,host:port...]")
    t.add_argument("path", nargs="?", default="/")
    t.set_defaults(fn=cmd_tree)

    g = sub.add_parser("gpus", help="show GPU/xGMI topology")
    g.set_defaults(fn=cmd_gpus)

    ck = sub.add_parser("check", help="validate a config file")
    ck.add_argument("-f", "--file", required=True)
    ck.set_defaults(fn=cmd_check)

    v = sub.add_parser("verify", help="pre-deployment registration smoke test")
    v.add_argument("-f", "--file", required=True)
    v.add_argument("--timeout", type=int, default=30, help="connect wait seconds")
    v.add_argument("--attempts", type=int, default=5, help="connect attempts")
    v.set_defaults(fn=cmd_verify)

    b = sub.add_parser("binder", help="serve DNS A/SRV answers from the registration tree")
    b.add_argument("--servers", required=True, help="ZK host:port[,host:port...]")
    b.add_argument("--host", default="127.0.0.1")
    b.add_argument("--port", type=int, default=5353)
    b.set_defaults(fn=cmd_binder)

    args = ap.parse_args(argv)
    return args.fn(args)


if __name__ == "__main__":
    sys.exit(main())
