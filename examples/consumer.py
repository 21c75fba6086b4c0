#!/usr/bin/env python3
"""A Binder-style consumer: discover live instances of a service by reading
the records registrar writes, and follow changes with watches.

Usage:
    python examples/consumer.py --servers HOST:PORT[,HOST:PORT...] DOMAIN

Example (against a demo ensemble + daemon):
    ./bin/zkensembled -n 1           # note the port
    ./bin/registrard -f etc/config.example.json &
    python examples/consumer.py --servers 127.0.0.1:PORT test.coal.example.com
"""
import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import registrar_amd as ra  # noqa: E402


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--servers", required=True)
    ap.add_argument("domain")
    ap.add_argument("--follow", action="store_true", help="keep watching for changes")
    args = ap.parse_args()

    servers = []
    for hp in args.servers.split(","):
        host, port = hp.rsplit(":", 1)
        servers.append((host, int(port)))
    path = ra.domain_to_path(args.domain)

    c = ra.ZkClient(servers=servers, connect_max_attempts=5)
    c.start()
    if not c.wait_connected(15000):
        print("could not connect", file=sys.stderr)
        return 1

    def show():
        rc, data, st = c.get(path)
        if rc == 0 and data:
            rec = json.loads(data)
            if rec.get("type") == "service":
                svc = rec["service"]["service"]
                print("SRV %s.%s.%s port=%s ttl=%s"
                      % (svc["srvce"], svc["proto"], args.domain, svc["port"], svc["ttl"]))
        rc, children = c.get_children(path, watch=args.follow)
        if rc != 0:
            print("(no such domain path: %s)" % path)
            return
        live = []
        for ch in children:
            rc, data, st = c.get("%s/%s" % (path, ch))
            if rc != 0:
                continue
            rec = json.loads(data)
            addr = rec.get("address")
            gpu = rec.get(rec.get("type", ""), {}).get("gpu")
            live.append((ch, addr, st["ephemeralOwner"], gpu))
        print("%d live instance(s) under %s:" % (len(live), path))
        for name, addr, owner, gpu in live:
            extra = " xgmiRank=%s gpu=%s" % (gpu["xgmiRank"], gpu["index"]) if gpu else ""
            print("  %-24s %-15s session=0x%x%s" % (name, addr, owner, extra))

    show()
    if args.follow:
        print("watching for membership changes (ctrl-c to stop)...")
        try:
            while True:
                for ev in c.poll_watches():
                    print("-- change: %s %s" % (ev["type"], ev["path"]))
                    show()
                time.sleep(0.2)
        except KeyboardInterrupt:
            pass
    c.close()
    return 0


if __name__ == "__main__":
    sys.exit(main())
