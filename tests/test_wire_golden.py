"""Golden wire-format tests: hand-packed jute bytes over a raw socket.

These vectors are built independently with struct.pack from the ZooKeeper
wire layout (big-endian ints, length-prefixed strings/buffers, 4-byte
frames), so they cross-check the native codec rather than testing it
against itself."""
import socket
import struct

import pytest

import registrar_amd as ra


def connect_raw(ens):
    host, port = ens.connect_string().split(",")[0].rsplit(":", 1)
    s = socket.create_connection((host, int(port)), timeout=5)
    return s


def send_frame(s, body):
    s.sendall(struct.pack(">i", len(body)) + body)


def recv_frame(s):
    hdr = b""
    while len(hdr) < 4:
        hdr += s.recv(4 - len(hdr))
    n = struct.unpack(">i", hdr)[0]
    body = b""
    while len(body) < n:
        body += s.recv(n - len(body))
    return body


def zk_string(b):
    return struct.pack(">i", len(b)) + b


def test_connect_request_layout(ensemble):
    s = connect_raw(ensemble)
    # ConnectRequest: protoVersion(i) lastZxid(q) timeout(i) sessionId(q) passwd(buffer)
    body = struct.pack(">iqiq", 0, 0, 5000, 0) + zk_string(b"\x00" * 16)
    assert len(body) == 4 + 8 + 4 + 8 + 4 + 16  # 44 bytes, fixed layout
    send_frame(s, body)
    resp = recv_frame(s)
    # ConnectResponse: protoVersion(i) timeout(i) sessionId(q) passwd(buffer)
    proto, timeout = struct.unpack(">ii", resp[:8])
    session_id = struct.unpack(">q", resp[8:16])[0]
    pwlen = struct.unpack(">i", resp[16:20])[0]
    assert proto == 0
    assert timeout >= 400  # negotiated up to the server's floor
    assert session_id != 0
    assert pwlen == 16
    s.close()


def full_session(ensemble):
    s = connect_raw(ensemble)
    body = struct.pack(">iqiq", 0, 0, 5000, 0) + zk_string(b"\x00" * 16)
    send_frame(s, body)
    recv_frame(s)
    return s


def test_create_exists_delete_roundtrip(ensemble):
    s = full_session(ensemble)
    # create(xid=1): header{xid,type=1} path data acls[world:anyone,perms=31] flags=0
    req = struct.pack(">ii", 1, 1)
    req += zk_string(b"/golden")
    req += zk_string(b"hello")
    req += struct.pack(">i", 1) + struct.pack(">i", 31) + zk_string(b"world") + zk_string(b"anyone")
    req += struct.pack(">i", 0)
    send_frame(s, req)
    resp = recv_frame(s)
    xid, zxid, err = struct.unpack(">iqi", resp[:16])
    assert (xid, err) == (1, 0) and zxid > 0
    created = resp[16:]
    assert created == zk_string(b"/golden")

    # exists(xid=2, type=3): path watch=false → Stat (11 fields, 68 bytes)
    send_frame(s, struct.pack(">ii", 2, 3) + zk_string(b"/golden") + b"\x00")
    resp = recv_frame(s)
    xid, zxid2, err = struct.unpack(">iqi", resp[:16])
    assert (xid, err) == (2, 0)
    stat = resp[16:]
    assert len(stat) == 8 * 5 + 4 * 5 + 8  # czxid mzxid ctime mtime | ver cver aver dlen nchild | eph pzxid
    czxid, mzxid = struct.unpack(">qq", stat[:16])
    version, cversion, aversion = struct.unpack(">iii", stat[32:44])
    eph_owner = struct.unpack(">q", stat[44:52])[0]
    data_len = struct.unpack(">i", stat[52:56])[0]
    assert czxid == zxid and mzxid == zxid
    assert version == 0 and eph_owner == 0 and data_len == 5

    # getData(xid=3, type=4) → data + Stat
    send_frame(s, struct.pack(">ii", 3, 4) + zk_string(b"/golden") + b"\x00")
    resp = recv_frame(s)
    assert struct.unpack(">iqi", resp[:16])[2] == 0
    assert resp[16:16 + 4 + 5] == zk_string(b"hello")

    # delete(xid=4, type=2): path version=-1
    send_frame(s, struct.pack(">ii", 4, 2) + zk_string(b"/golden") + struct.pack(">i", -1))
    resp = recv_frame(s)
    assert struct.unpack(">iqi", resp[:16])[2] == 0

    # exists again → err NoNode(-101), no body
    send_frame(s, struct.pack(">ii", 5, 3) + zk_string(b"/golden") + b"\x00")
    resp = recv_frame(s)
    assert struct.unpack(">iqi", resp[:16])[2] == -101
    assert len(resp) == 16
    s.close()


def test_ping_layout(ensemble):
    s = full_session(ensemble)
    send_frame(s, struct.pack(">ii", -2, 11))  # xid=-2 type=ping, no body
    resp = recv_frame(s)
    xid, zxid, err = struct.unpack(">iqi", resp)
    assert xid == -2 and err == 0
    s.close()


def test_ephemeral_owner_on_wire(ensemble):
    s = full_session(ensemble)
    # re-read our session id via a fresh handshake socket is complex; instead
    # create ephemeral and check ephemeralOwner equals the ConnectResponse id
    s.close()
    s = connect_raw(ensemble)
    send_frame(s, struct.pack(">iqiq", 0, 0, 5000, 0) + zk_string(b"\x00" * 16))
    resp = recv_frame(s)
    session_id = struct.unpack(">q", resp[8:16])[0]
    req = struct.pack(">ii", 1, 1) + zk_string(b"/eph-wire") + zk_string(b"")
    req += struct.pack(">i", 1) + struct.pack(">i", 31) + zk_string(b"world") + zk_string(b"anyone")
    req += struct.pack(">i", 1)  # EPHEMERAL
    send_frame(s, req)
    assert struct.unpack(">iqi", recv_frame(s)[:16])[2] == 0
    send_frame(s, struct.pack(">ii", 2, 3) + zk_string(b"/eph-wire") + b"\x00")
    resp = recv_frame(s)
    stat = resp[16:]
    eph_owner = struct.unpack(">q", stat[44:52])[0]
    assert eph_owner == session_id
    s.close()


def test_get_acl_and_set_acl_wire(ensemble):
    s = full_session(ensemble)
    # create a node
    req = struct.pack(">ii", 1, 1) + zk_string(b"/acl") + zk_string(b"")
    req += struct.pack(">i", 1) + struct.pack(">i", 31) + zk_string(b"world") + zk_string(b"anyone")
    req += struct.pack(">i", 0)
    send_frame(s, req)
    recv_frame(s)
    # getACL (type 6)
    send_frame(s, struct.pack(">ii", 2, 6) + zk_string(b"/acl"))
    resp = recv_frame(s)
    assert struct.unpack(">iqi", resp[:16])[2] == 0
    nacl = struct.unpack(">i", resp[16:20])[0]
    perms = struct.unpack(">i", resp[20:24])[0]
    assert nacl == 1 and perms == 31
    # setACL (type 7) acks and bumps aversion
    acl = struct.pack(">i", 1) + struct.pack(">i", 31) + zk_string(b"world") + zk_string(b"anyone")
    send_frame(s, struct.pack(">ii", 3, 7) + zk_string(b"/acl") + acl + struct.pack(">i", -1))
    resp = recv_frame(s)
    assert struct.unpack(">iqi", resp[:16])[2] == 0
    stat = resp[16:]
    aversion = struct.unpack(">i", stat[40:44])[0]
    assert aversion == 1
    s.close()
