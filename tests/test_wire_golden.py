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


# ---------------------------------------------------------------------------
# round 2: independent byte-level vectors for EVERY opcode the client speaks
# (VERDICT r1 next-round #1 — jute.hpp:33-50). Layout constants below are
# hand-derived from the Apache ZooKeeper jute definitions, not from this
# repo's codec: ReplyHeader = xid(i32) zxid(i64) err(i32); Stat = czxid mzxid
# ctime mtime (4×i64) | version cversion aversion (3×i32) | ephemeralOwner
# (i64) dataLength numChildren (2×i32) pzxid (i64) = 68 bytes.


def reply_header(resp):
    return struct.unpack(">iqi", resp[:16])


def parse_stat(b):
    assert len(b) >= 68
    czxid, mzxid, ctime, mtime = struct.unpack(">qqqq", b[:32])
    version, cversion, aversion = struct.unpack(">iii", b[32:44])
    eph, dlen, nchild = struct.unpack(">qii", b[44:60])
    pzxid = struct.unpack(">q", b[60:68])[0]
    return {"czxid": czxid, "mzxid": mzxid, "ctime": ctime, "mtime": mtime,
            "version": version, "cversion": cversion, "aversion": aversion,
            "ephemeralOwner": eph, "dataLength": dlen, "numChildren": nchild,
            "pzxid": pzxid}


OPEN_ACL = struct.pack(">i", 1) + struct.pack(">i", 31) + zk_string(b"world") + zk_string(b"anyone")


def create_req(xid, path, data=b"", flags=0, op=1):
    return (struct.pack(">ii", xid, op) + zk_string(path) + zk_string(data)
            + OPEN_ACL + struct.pack(">i", flags))


def test_set_data_golden(ensemble):
    s = full_session(ensemble)
    send_frame(s, create_req(1, b"/sd", b"v0"))
    xid, czxid, err = reply_header(recv_frame(s))
    assert err == 0

    # setData (type 5): path data version
    send_frame(s, struct.pack(">ii", 2, 5) + zk_string(b"/sd") + zk_string(b"value1")
               + struct.pack(">i", -1))
    resp = recv_frame(s)
    xid, zxid, err = reply_header(resp)
    assert (xid, err) == (2, 0)
    st = parse_stat(resp[16:])
    assert st["version"] == 1 and st["dataLength"] == 6
    assert st["czxid"] == czxid
    # the reply zxid must be THIS op's commit zxid == the node's new mzxid
    assert zxid == st["mzxid"] > st["czxid"]

    # version conflict → BadVersion(-103), empty body
    send_frame(s, struct.pack(">ii", 3, 5) + zk_string(b"/sd") + zk_string(b"x")
               + struct.pack(">i", 7))
    resp = recv_frame(s)
    assert reply_header(resp)[2] == -103 and len(resp) == 16
    s.close()


def test_get_children_golden(ensemble):
    s = full_session(ensemble)
    for i, p in enumerate([b"/gc", b"/gc/b", b"/gc/a"]):
        send_frame(s, create_req(1 + i, p))
        assert reply_header(recv_frame(s))[2] == 0

    # getChildren (type 8): path watch → vector<string> (sorted)
    send_frame(s, struct.pack(">ii", 4, 8) + zk_string(b"/gc") + b"\x00")
    resp = recv_frame(s)
    assert reply_header(resp)[:1] == (4,) and reply_header(resp)[2] == 0
    body = resp[16:]
    count = struct.unpack(">i", body[:4])[0]
    assert count == 2
    assert body[4:] == zk_string(b"a") + zk_string(b"b")

    # getChildren2 (type 12): children + Stat
    send_frame(s, struct.pack(">ii", 5, 12) + zk_string(b"/gc") + b"\x00")
    resp = recv_frame(s)
    assert reply_header(resp)[2] == 0
    body = resp[16:]
    count = struct.unpack(">i", body[:4])[0]
    assert count == 2 and body[4:4 + 10] == zk_string(b"a") + zk_string(b"b")
    st = parse_stat(body[14:])
    assert st["numChildren"] == 2 and st["cversion"] == 2

    # getChildren on a missing node → NoNode, no body
    send_frame(s, struct.pack(">ii", 6, 8) + zk_string(b"/gc-missing") + b"\x00")
    resp = recv_frame(s)
    assert reply_header(resp)[2] == -101 and len(resp) == 16
    s.close()


def test_create2_golden(ensemble):
    # create2 (type 15): CreateResponse + Stat
    s = full_session(ensemble)
    send_frame(s, create_req(1, b"/c2", b"abc", op=15))
    resp = recv_frame(s)
    xid, zxid, err = reply_header(resp)
    assert (xid, err) == (1, 0)
    body = resp[16:]
    assert body[:4 + 3] == zk_string(b"/c2")
    st = parse_stat(body[7:])
    assert st["czxid"] == st["mzxid"] == zxid
    assert st["version"] == 0 and st["dataLength"] == 3
    s.close()


def test_sync_golden(ensemble):
    # sync (type 9) echoes the path back
    s = full_session(ensemble)
    send_frame(s, struct.pack(">ii", 1, 9) + zk_string(b"/whatever"))
    resp = recv_frame(s)
    assert reply_header(resp)[2] == 0
    assert resp[16:] == zk_string(b"/whatever")
    s.close()


def multi_header(op_type, done=False, err=-1):
    return struct.pack(">i?i", op_type, done, err)


def test_multi_success_golden(ensemble):
    s = full_session(ensemble)
    # multi (type 14): create /m1 → setData /m1 → check /m1 v1 → delete /m1
    body = struct.pack(">ii", 1, 14)
    body += multi_header(1) + zk_string(b"/m1") + zk_string(b"d0") + OPEN_ACL + struct.pack(">i", 0)
    body += multi_header(5) + zk_string(b"/m1") + zk_string(b"d1") + struct.pack(">i", -1)
    body += multi_header(13) + zk_string(b"/m1") + struct.pack(">i", 1)
    body += multi_header(2) + zk_string(b"/m1") + struct.pack(">i", -1)
    body += multi_header(-1, done=True)
    send_frame(s, body)
    resp = recv_frame(s)
    xid, zxid, err = reply_header(resp)
    assert (xid, err) == (1, 0) and zxid > 0
    r = resp[16:]
    # op 0: create result — header{1, false, 0} + path
    t, done, oerr = struct.unpack(">i?i", r[:9]); r = r[9:]
    assert (t, done, oerr) == (1, False, 0)
    assert r[:7] == zk_string(b"/m1"); r = r[7:]
    # op 1: setData result — header{5, false, 0} + Stat(version=1)
    t, done, oerr = struct.unpack(">i?i", r[:9]); r = r[9:]
    assert (t, done, oerr) == (5, False, 0)
    st = parse_stat(r[:68]); r = r[68:]
    assert st["version"] == 1 and st["dataLength"] == 2
    # op 2: check result — bare header{13, false, 0}
    t, done, oerr = struct.unpack(">i?i", r[:9]); r = r[9:]
    assert (t, done, oerr) == (13, False, 0)
    # op 3: delete result — bare header{2, false, 0}
    t, done, oerr = struct.unpack(">i?i", r[:9]); r = r[9:]
    assert (t, done, oerr) == (2, False, 0)
    # terminator: {-1, true, -1}
    t, done, oerr = struct.unpack(">i?i", r[:9]); r = r[9:]
    assert (t, done, oerr) == (-1, True, -1) and r == b""
    # the delete really applied
    send_frame(s, struct.pack(">ii", 2, 3) + zk_string(b"/m1") + b"\x00")
    assert reply_header(recv_frame(s))[2] == -101
    s.close()


def test_multi_rollback_golden(ensemble):
    s = full_session(ensemble)
    # create would succeed, delete of a missing node fails ⇒ whole txn rolls
    # back; every result is framed as an ERROR record: header{-1, false, err}
    # + ErrorResult(int err); non-failing ops report RuntimeInconsistency(-2)
    body = struct.pack(">ii", 1, 14)
    body += multi_header(1) + zk_string(b"/mr1") + zk_string(b"") + OPEN_ACL + struct.pack(">i", 0)
    body += multi_header(2) + zk_string(b"/mr-missing") + struct.pack(">i", -1)
    body += multi_header(-1, done=True)
    send_frame(s, body)
    resp = recv_frame(s)
    xid, zxid, err = reply_header(resp)
    assert xid == 1 and err == -101  # first failure code surfaces in the header
    r = resp[16:]
    t, done, oerr = struct.unpack(">i?i", r[:9])
    assert (t, done, oerr) == (-1, False, -2)
    assert struct.unpack(">i", r[9:13])[0] == -2
    r = r[13:]
    t, done, oerr = struct.unpack(">i?i", r[:9])
    assert (t, done, oerr) == (-1, False, -101)
    assert struct.unpack(">i", r[9:13])[0] == -101
    r = r[13:]
    assert struct.unpack(">i?i", r[:9]) == (-1, True, -1) and r[9:] == b""
    # rollback: /mr1 must NOT exist
    send_frame(s, struct.pack(">ii", 2, 3) + zk_string(b"/mr1") + b"\x00")
    assert reply_header(recv_frame(s))[2] == -101
    s.close()


def test_close_session_golden(ensemble):
    s = connect_raw(ensemble)
    send_frame(s, struct.pack(">iqiq", 0, 0, 5000, 0) + zk_string(b"\x00" * 16))
    recv_frame(s)
    send_frame(s, create_req(1, b"/cs-eph", flags=1))  # EPHEMERAL
    assert reply_header(recv_frame(s))[2] == 0
    # closeSession (type -11, empty body): server acks then closes the conn
    send_frame(s, struct.pack(">ii", 2, -11))
    resp = recv_frame(s)
    assert reply_header(resp)[:1] == (2,) and reply_header(resp)[2] == 0
    assert s.recv(1) == b""  # server-side close
    s.close()
    # the ephemeral dies WITH the closed session (no timeout wait)
    s2 = full_session(ensemble)
    send_frame(s2, struct.pack(">ii", 1, 3) + zk_string(b"/cs-eph") + b"\x00")
    assert reply_header(recv_frame(s2))[2] == -101
    s2.close()


def test_reconnect_same_session_golden(ensemble):
    s = connect_raw(ensemble)
    send_frame(s, struct.pack(">iqiq", 0, 0, 5000, 0) + zk_string(b"\x00" * 16))
    resp = recv_frame(s)
    sid = struct.unpack(">q", resp[8:16])[0]
    passwd = resp[20:36]
    assert len(passwd) == 16
    send_frame(s, create_req(1, b"/rc-eph", flags=1))
    assert reply_header(recv_frame(s))[2] == 0
    s.close()  # abrupt drop — NOT closeSession

    # reconnect with the same sessionId + passwd: session (and ephemeral)
    # survive; the ConnectResponse echoes the same id
    s2 = connect_raw(ensemble)
    send_frame(s2, struct.pack(">iqiq", 0, 0, 5000, sid) + zk_string(passwd))
    resp = recv_frame(s2)
    assert struct.unpack(">q", resp[8:16])[0] == sid
    assert struct.unpack(">i", resp[4:8])[0] > 0  # negotiated timeout
    send_frame(s2, struct.pack(">ii", 1, 3) + zk_string(b"/rc-eph") + b"\x00")
    resp = recv_frame(s2)
    assert reply_header(resp)[2] == 0
    assert parse_stat(resp[16:])["ephemeralOwner"] == sid
    s2.close()


def test_reconnect_bad_passwd_golden(ensemble):
    s = connect_raw(ensemble)
    send_frame(s, struct.pack(">iqiq", 0, 0, 5000, 0) + zk_string(b"\x00" * 16))
    sid = struct.unpack(">q", recv_frame(s)[8:16])[0]
    s.close()
    # wrong passwd ⇒ the canonical "expired" handshake: sessionId=0, timeOut=0
    s2 = connect_raw(ensemble)
    send_frame(s2, struct.pack(">iqiq", 0, 0, 5000, sid) + zk_string(b"\xff" * 16))
    resp = recv_frame(s2)
    assert struct.unpack(">i", resp[4:8])[0] == 0
    assert struct.unpack(">q", resp[8:16])[0] == 0
    s2.close()


def test_watcher_event_delivery_golden(ensemble):
    a = full_session(ensemble)
    b = full_session(ensemble)
    # A arms an exist-watch on a nonexistent node (watch=1 even on NoNode)
    send_frame(a, struct.pack(">ii", 1, 3) + zk_string(b"/we1") + b"\x01")
    assert reply_header(recv_frame(a))[2] == -101
    # B creates it → A gets an unsolicited frame: ReplyHeader{xid=-1, _, 0} +
    # WatcherEvent{type=1 NodeCreated, state=3 SyncConnected, path}
    send_frame(b, create_req(1, b"/we1"))
    assert reply_header(recv_frame(b))[2] == 0
    ev = recv_frame(a)
    xid, zxid, err = reply_header(ev)
    assert (xid, err) == (-1, 0)
    etype, estate = struct.unpack(">ii", ev[16:24])
    assert (etype, estate) == (1, 3)
    assert ev[24:] == zk_string(b"/we1")

    # data watch fires on setData with type=3 NodeDataChanged
    send_frame(a, struct.pack(">ii", 2, 4) + zk_string(b"/we1") + b"\x01")  # getData+watch
    assert reply_header(recv_frame(a))[2] == 0
    send_frame(b, struct.pack(">ii", 2, 5) + zk_string(b"/we1") + zk_string(b"x") + struct.pack(">i", -1))
    assert reply_header(recv_frame(b))[2] == 0
    ev = recv_frame(a)
    assert reply_header(ev)[0] == -1
    assert struct.unpack(">ii", ev[16:24]) == (3, 3)
    assert ev[24:] == zk_string(b"/we1")

    # child watch fires on child create with type=4 NodeChildrenChanged
    send_frame(a, struct.pack(">ii", 3, 8) + zk_string(b"/we1") + b"\x01")  # getChildren+watch
    assert reply_header(recv_frame(a))[2] == 0
    send_frame(b, create_req(3, b"/we1/kid"))
    assert reply_header(recv_frame(b))[2] == 0
    ev = recv_frame(a)
    assert reply_header(ev)[0] == -1
    assert struct.unpack(">ii", ev[16:24]) == (4, 3)
    assert ev[24:] == zk_string(b"/we1")
    a.close()
    b.close()


def test_set_watches_golden(ensemble):
    # watches do not survive a disconnect server-side; setWatches (op 101,
    # xid -8) re-arms them and synthesizes events for changes missed while
    # away (mzxid > relativeZxid)
    a = connect_raw(ensemble)
    send_frame(a, struct.pack(">iqiq", 0, 0, 5000, 0) + zk_string(b"\x00" * 16))
    resp = recv_frame(a)
    sid = struct.unpack(">q", resp[8:16])[0]
    passwd = resp[20:36]
    send_frame(a, create_req(1, b"/sw1", b"v0"))
    assert reply_header(recv_frame(a))[2] == 0
    send_frame(a, struct.pack(">ii", 2, 4) + zk_string(b"/sw1") + b"\x01")  # getData+watch
    resp = recv_frame(a)
    xid, last_zxid, err = reply_header(resp)
    assert err == 0
    a.close()  # drop; the armed watch will fire into the void

    b = full_session(ensemble)
    send_frame(b, struct.pack(">ii", 1, 5) + zk_string(b"/sw1") + zk_string(b"v1") + struct.pack(">i", -1))
    assert reply_header(recv_frame(b))[2] == 0

    a2 = connect_raw(ensemble)
    send_frame(a2, struct.pack(">iqiq", 0, last_zxid, 5000, sid) + zk_string(passwd))
    assert struct.unpack(">q", recv_frame(a2)[8:16])[0] == sid
    # setWatches: relativeZxid, dataWatches=[/sw1], existWatches=[], childWatches=[]
    body = struct.pack(">ii", -8, 101) + struct.pack(">q", last_zxid)
    body += struct.pack(">i", 1) + zk_string(b"/sw1")
    body += struct.pack(">i", 0) + struct.pack(">i", 0)
    send_frame(a2, body)
    # two frames arrive: the synthetic NodeDataChanged event and the -8 reply
    # (order unspecified — classify by xid)
    frames = [recv_frame(a2), recv_frame(a2)]
    by_xid = {reply_header(f)[0]: f for f in frames}
    assert set(by_xid) == {-1, -8}
    ev = by_xid[-1]
    assert struct.unpack(">ii", ev[16:24]) == (3, 3)  # NodeDataChanged
    assert ev[24:] == zk_string(b"/sw1")
    assert reply_header(by_xid[-8])[2] == 0

    # re-arm via setWatches with a CURRENT zxid: no synthetic event, but the
    # watch is live — a later setData from B fires it for real
    send_frame(a2, struct.pack(">ii", 3, 4) + zk_string(b"/sw1") + b"\x00")
    resp = recv_frame(a2)
    cur_zxid = reply_header(resp)[1]
    body = struct.pack(">ii", -8, 101) + struct.pack(">q", cur_zxid)
    body += struct.pack(">i", 1) + zk_string(b"/sw1")
    body += struct.pack(">i", 0) + struct.pack(">i", 0)
    send_frame(a2, body)
    assert reply_header(recv_frame(a2))[:1] == (-8,)
    send_frame(b, struct.pack(">ii", 2, 5) + zk_string(b"/sw1") + zk_string(b"v2") + struct.pack(">i", -1))
    assert reply_header(recv_frame(b))[2] == 0
    ev = recv_frame(a2)
    assert reply_header(ev)[0] == -1
    assert struct.unpack(">ii", ev[16:24]) == (3, 3)
    a2.close()
    b.close()


def test_self_watch_event_precedes_triggering_reply(ensemble):
    """ZooKeeper ordering guarantee: when a client's own operation fires its
    own watch, the WatcherEvent frame is delivered BEFORE the reply to the
    operation that triggered it (both ride the same ordered outbox)."""
    s = full_session(ensemble)
    send_frame(s, create_req(1, b"/selfw", b"v0"))
    assert reply_header(recv_frame(s))[2] == 0
    send_frame(s, struct.pack(">ii", 2, 4) + zk_string(b"/selfw") + b"\x01")  # getData+watch
    assert reply_header(recv_frame(s))[2] == 0
    send_frame(s, struct.pack(">ii", 3, 5) + zk_string(b"/selfw") + zk_string(b"v1")
               + struct.pack(">i", -1))
    first = recv_frame(s)
    second = recv_frame(s)
    assert reply_header(first)[0] == -1, "watch event must precede the setData reply"
    assert struct.unpack(">ii", first[16:24]) == (3, 3)  # NodeDataChanged
    assert reply_header(second)[:1] == (3,) and reply_header(second)[2] == 0
    s.close()


def test_session_timeout_negotiation_golden(ensemble):
    # requested timeout is clamped into [min, max] and echoed in the
    # ConnectResponse (real ZK: 2×tick .. 20×tick; ours: configurable floor)
    s = connect_raw(ensemble)
    send_frame(s, struct.pack(">iqiq", 0, 0, 1, 0) + zk_string(b"\x00" * 16))  # absurdly small
    resp = recv_frame(s)
    lo = struct.unpack(">i", resp[4:8])[0]
    assert lo > 1  # negotiated UP to the server floor
    s.close()
    s = connect_raw(ensemble)
    send_frame(s, struct.pack(">iqiq", 0, 0, 10**9, 0) + zk_string(b"\x00" * 16))  # absurdly large
    resp = recv_frame(s)
    hi = struct.unpack(">i", resp[4:8])[0]
    assert hi <= 60000  # capped at the server max
    s.close()


def test_sequence_create_golden(ensemble):
    """SEQUENCE creates append a zero-padded 10-digit counter (parent
    cversion), like real ZooKeeper's %010d suffix."""
    s = full_session(ensemble)
    send_frame(s, create_req(1, b"/seqparent"))
    assert reply_header(recv_frame(s))[2] == 0
    send_frame(s, create_req(2, b"/seqparent/n-", flags=2))  # SEQUENCE
    resp = recv_frame(s)
    assert reply_header(resp)[2] == 0
    # counter = parent cversion at create time; a fresh parent starts at 0
    assert resp[16:] == zk_string(b"/seqparent/n-0000000000")
    send_frame(s, create_req(3, b"/seqparent/n-", flags=2))
    resp = recv_frame(s)
    assert resp[16:] == zk_string(b"/seqparent/n-0000000001")
    s.close()


def test_auth_packet_acked(ensemble):
    """Real clients send an auth packet (xid -4, op 100) right after the
    handshake; the open-ACL ensemble must ack it, not error the session."""
    s = full_session(ensemble)
    body = struct.pack(">ii", -4, 100) + struct.pack(">i", 0) + zk_string(b"digest") + zk_string(b"u:p")
    send_frame(s, body)
    resp = recv_frame(s)
    assert reply_header(resp)[:1] == (-4,) and reply_header(resp)[2] == 0
    # session still fully functional
    send_frame(s, create_req(1, b"/after-auth"))
    assert reply_header(recv_frame(s))[2] == 0
    s.close()
