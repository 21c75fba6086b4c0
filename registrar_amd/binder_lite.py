"""binder_lite — a minimal DNS responder over the registrar's ZooKeeper
records, demonstrating the full Triton-style discovery triangle in this one
repo: registrar writes → (this) reads → DNS answers.

The real Binder is a separate reference repo; this lite implementation
answers A and SRV queries straight from the data contract documented in
docs/data-format.md (reference README.md:441-783):

- A <domain>        → one answer per live host record under the domain path
                      (ephemeral znodes: liveness comes from ZK sessions)
- SRV <domain>      → one answer per host record, port from the host
                      record's ports (else the service record's port),
                      target <child>.<domain>
- TTL precedence    → host-record ttl, else service-record ttl, else 30
                      (reference README.md:670-754's record-over-service
                      precedence)

Wire format is hand-built (no DNS library): standard header, QNAME
compression pointers for answers, A (type 1) and SRV (type 33) RRs.
Unsupported types → empty NOERROR; unknown domains → NXDOMAIN.

Transport: UDP with classic 512-byte truncation (TC bit set, whole RRs only)
plus a TCP listener on the same port (RFC 1035 2-byte length framing) so a
truncated answer set — easy at 1k records per domain — is retried over TCP
and served complete.
"""
import socket
import struct
import threading

import registrar_amd as ra

DEFAULT_TTL = 30
QTYPE_A = 1
QTYPE_SRV = 33
QCLASS_IN = 1
UDP_MAX = 512  # classic DNS/UDP payload limit (no EDNS here)
FLAG_TC = 0x0200


def _encode_name(name):
    out = b""
    for label in name.strip(".").split("."):
        raw = label.encode("ascii")
        out += bytes([len(raw)]) + raw
    return out + b"\x00"


def _decode_name(buf, off):
    labels = []
    while True:
        n = buf[off]
        if n == 0:
            off += 1
            break
        if n & 0xC0:  # compression pointer (queries don't normally use these)
            ptr = struct.unpack(">H", buf[off:off + 2])[0] & 0x3FFF
            inner, _ = _decode_name(buf, ptr)
            labels.append(inner)
            off += 2
            return ".".join(labels), off
        labels.append(buf[off + 1:off + 1 + n].decode("ascii"))
        off += 1 + n
    return ".".join(labels), off


class BinderLite:
    """UDP DNS responder backed by a ZkClient (works against the synthetic
    ensemble or a real ZooKeeper)."""

    def __init__(self, servers, host="127.0.0.1", port=0):
        self._client = ra.ZkClient(servers=servers, session_timeout_ms=30000)
        self._client.start()
        if not self._client.wait_connected(15000):
            raise RuntimeError("binder_lite: cannot connect to ZooKeeper")
        self._sock = socket.socket(socket.AF_INET, socket.SOCK_DGRAM)
        self._sock.bind((host, port))
        self._sock.settimeout(0.2)
        self._addr = self._sock.getsockname()
        # TCP companion on the SAME port (separate protocol namespace):
        # truncated UDP answers retry here and get the full set
        self._tsock = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
        self._tsock.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
        self._tsock.bind(self._addr)
        self._tsock.listen(16)
        self._tsock.settimeout(0.2)
        self._stop = threading.Event()
        self._thread = threading.Thread(target=self._serve, daemon=True)
        self._tthread = threading.Thread(target=self._serve_tcp, daemon=True)

    @property
    def address(self):
        return self._addr

    def start(self):
        self._thread.start()
        self._tthread.start()

    def stop(self):
        self._stop.set()
        self._thread.join()
        self._tthread.join()
        self._sock.close()
        self._tsock.close()
        self._client.close()

    # ---- record lookup (the registrar data contract) ----

    def _lookup(self, domain):
        path = ra.domain_to_path(domain)
        rc, data, _ = self._client.get(path)
        service = None
        if rc == 0 and data:
            try:
                import json

                rec = json.loads(data)
                if rec.get("type") == "service":
                    service = rec["service"]["service"]
            except (ValueError, KeyError, TypeError):
                pass  # malformed service record: answer from host records only
        rc, children = self._client.get_children(path)
        if rc != 0:
            return None, None, None  # NXDOMAIN
        hosts = []
        import json

        for ch in sorted(children):
            rc, data, st = self._client.get("%s/%s" % (path, ch))
            if rc != 0 or not data:
                continue
            try:
                rec = json.loads(data)
            except ValueError:
                continue
            if rec.get("type") == "service":
                continue
            typed = rec.get(rec.get("type", ""), {})
            hosts.append({
                "name": ch,
                "address": rec.get("address"),
                "ttl": rec.get("ttl"),
                "ports": typed.get("ports") or ([service["port"]] if service else []),
            })
        return hosts, service, path

    # ---- DNS wire ----

    def _serve(self):
        while not self._stop.is_set():
            try:
                buf, peer = self._sock.recvfrom(4096)
            except socket.timeout:
                continue
            except OSError:
                break
            try:
                resp = self._handle(buf, udp=True)
            except Exception:
                continue
            if resp:
                self._sock.sendto(resp, peer)

    def _serve_tcp(self):
        while not self._stop.is_set():
            try:
                conn, _ = self._tsock.accept()
            except socket.timeout:
                continue
            except OSError:
                break
            try:
                conn.settimeout(2.0)
                hdr = self._recv_exact(conn, 2)
                if hdr is None:
                    continue
                n = struct.unpack(">H", hdr)[0]
                buf = self._recv_exact(conn, n)
                if buf is None:
                    continue
                resp = self._handle(buf, udp=False)
                if resp:
                    conn.sendall(struct.pack(">H", len(resp)) + resp)
            except Exception:
                pass
            finally:
                conn.close()

    @staticmethod
    def _recv_exact(conn, n):
        buf = b""
        while len(buf) < n:
            chunk = conn.recv(n - len(buf))
            if not chunk:
                return None
            buf += chunk
        return buf

    def _handle(self, buf, udp=True):
        if len(buf) < 12:
            return None
        (txid, flags, qd, _an, _ns, _ar) = struct.unpack(">HHHHHH", buf[:12])
        if qd != 1:
            return None
        qname, off = _decode_name(buf, 12)
        qtype, qclass = struct.unpack(">HH", buf[off:off + 4])
        question = buf[12:off + 4]

        hosts, service, _ = self._lookup(qname)
        rcode = 0
        rrs = []
        if hosts is None:
            rcode = 3  # NXDOMAIN
        elif qclass == QCLASS_IN:
            for h in hosts:
                ttl = h["ttl"] if h["ttl"] is not None else (
                    service["ttl"] if service and service.get("ttl") is not None else DEFAULT_TTL)
                if qtype == QTYPE_A and h["address"]:
                    rdata = socket.inet_aton(h["address"])
                    rrs.append(struct.pack(">HHHIH", 0xC00C, QTYPE_A, QCLASS_IN, ttl, len(rdata)) + rdata)
                elif qtype == QTYPE_SRV and h["ports"]:
                    target = _encode_name("%s.%s" % (h["name"], qname))
                    rdata = struct.pack(">HHH", 0, 10, int(h["ports"][0])) + target
                    rrs.append(struct.pack(">HHHIH", 0xC00C, QTYPE_SRV, QCLASS_IN, ttl, len(rdata)) + rdata)

        # UDP: enforce the classic 512-byte limit — keep only whole RRs that
        # fit and set TC so the resolver retries over TCP (RFC 1035 §4.2.1)
        tc = 0
        count = len(rrs)
        answers = b"".join(rrs)
        if udp and 12 + len(question) + len(answers) > UDP_MAX:
            tc = FLAG_TC
            budget = UDP_MAX - 12 - len(question)
            answers = b""
            count = 0
            for rr in rrs:
                if len(answers) + len(rr) > budget:
                    break
                answers += rr
                count += 1

        flags = 0x8180 | tc | rcode  # QR + RD + RA (+TC)
        header = struct.pack(">HHHHHH", txid, flags, 1, count, 0, 0)
        return header + question + answers
