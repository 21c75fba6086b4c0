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


def test_malformed_multi_and_negative_lengths(ensemble):
    """Hostile jute bodies post-handshake: negative string lengths, absurd
    ACL counts, truncated multi framing — each closes only the offending
    connection; the ensemble keeps serving."""
    cases = [
        # create with negative path length
        struct.pack(">ii", 1, 1) + struct.pack(">i", -5),
        # create with path length far beyond the frame
        struct.pack(">ii", 1, 1) + struct.pack(">i", 1 << 30) + b"/x",
        # create with absurd ACL vector count
        struct.pack(">ii", 1, 1) + struct.pack(">i", 2) + b"/y"
        + struct.pack(">i", 0) + struct.pack(">i", 1 << 29),
        # multi with op body truncated mid-header
        struct.pack(">ii", 1, 14) + struct.pack(">i", 1),
        # multi that never sends the done terminator (frame just ends)
        struct.pack(">ii", 1, 14) + struct.pack(">i?i", 13, False, -1)
        + struct.pack(">i", 2) + b"/z" + struct.pack(">i", -1),
        # setWatches with a negative vector count
        struct.pack(">ii", -8, 101) + struct.pack(">q", 0) + struct.pack(">i", -3),
    ]
    for body in cases:
        s = raw_conn(ensemble)
        handshake(s)
        s.sendall(struct.pack(">i", len(body)) + body)
        s.settimeout(2)
        try:
            s.recv(64)  # server may reply with an error or just close
        except socket.timeout:
            pass
        s.close()
        assert ensemble_alive(ensemble)


def test_zero_and_negative_frame_lengths(ensemble):
    for framelen in (0, -1, -(1 << 31)):
        s = raw_conn(ensemble)
        handshake(s)
        s.sendall(struct.pack(">i", framelen))
        s.settimeout(2)
        try:
            s.recv(16)
        except socket.timeout:
            pass
        s.close()
        assert ensemble_alive(ensemble)
