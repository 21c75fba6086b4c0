"""Protocol robustness: the ensemble must survive hostile/garbage input
without crashing or wedging other sessions."""
import os
import socket
import struct
import time

import pytest

import registrar_amd as ra
from conftest import make_client, wait_for


def raw_conn(ens):
    host, port = ens.connect_string().split(",")[0].rsplit(":", 1)
    s = socket.create_connection((host, int(port)), timeout=5)
    return s


def handshake(sock):
    # minimal valid ConnectRequest
    body = struct.pack(">iqiq", 0, 0, 5000, 0) + struct.pack(">i", 16) + b"\x00" * 16
    sock.sendall(struct.pack(">i", len(body)) + body)
    hdr = sock.recv(4)
    n = struct.unpack(">i", hdr)[0]
    resp = b""
    while len(resp) < n:
        resp += sock.recv(n - len(resp))
    return resp


def ensemble_alive(ens):
    c = make_client(ens)
    rc, _ = c.create("/alive-%d" % time.monotonic_ns(), b"", True)
    ok = rc == ra.ZOK
    c.close()
    return ok


def test_garbage_pre_handshake(ensemble):
    s = raw_conn(ensemble)
    s.sendall(os.urandom(512))
    s.close()
    assert ensemble_alive(ensemble)


def test_oversized_frame_rejected(ensemble):
    s = raw_conn(ensemble)
    s.sendall(struct.pack(">i", 1 << 30))  # absurd frame length
    s.sendall(b"x" * 1024)
    time.sleep(0.1)
    s.close()
    assert ensemble_alive(ensemble)


def test_truncated_frames(ensemble):
    s = raw_conn(ensemble)
    handshake(s)
    # header claims 100 bytes, send 3, then hang up
    s.sendall(struct.pack(">i", 100) + b"abc")
    s.close()
    assert ensemble_alive(ensemble)


def test_garbage_post_handshake(ensemble):
    s = raw_conn(ensemble)
    handshake(s)
    for _ in range(50):
        blob = os.urandom(64)
        s.sendall(struct.pack(">i", len(blob)) + blob)
    time.sleep(0.2)
    s.close()
    assert ensemble_alive(ensemble)


def test_fuzz_random_streams(ensemble):
    import random

    rng = random.Random(1234)
    for _ in range(20):
        s = raw_conn(ensemble)
        try:
            n = rng.randrange(1, 2048)
            s.sendall(bytes(rng.getrandbits(8) for _ in range(n)))
        except (BrokenPipeError, ConnectionResetError):
            pass
        s.close()
    assert ensemble_alive(ensemble)
    # other sessions unaffected throughout
    c = make_client(ensemble)
    rc, _ = c.create("/post-fuzz", b"ok")
    assert rc == ra.ZOK
    c.close()
