"""One-shot watches: the Binder-side read pattern (registrar's consumers set
watches on the znodes registrar writes). Verifies watch registration via
exists/get/get_children and delivery on create/change/delete/child events."""
import time

import pytest

import registrar_amd as ra
from conftest import make_client, wait_for


def watches(c):
    acc = []

    def drain():
        acc.extend(c.poll_watches())
        return acc

    return drain


def test_exists_watch_fires_on_create(ensemble):
    c = make_client(ensemble)
    drain = watches(c)
    rc, _ = c.exists("/watched", watch=True)
    assert rc == ra.ZNONODE  # watch registers even on a missing node
    c.create("/watched", b"x")
    assert wait_for(lambda: any(w["type"] == "created" and w["path"] == "/watched" for w in drain()), 5)
    c.close()


def test_data_watch_fires_on_change_and_delete(ensemble):
    c = make_client(ensemble)
    drain = watches(c)
    c.create("/d", b"v0")
    c.get("/d", watch=True)
    c.set("/d", b"v1")
    assert wait_for(lambda: any(w["type"] == "changed" and w["path"] == "/d" for w in drain()), 5)
    # one-shot: re-arm for the delete
    c.get("/d", watch=True)
    c.delete_("/d")
    assert wait_for(lambda: any(w["type"] == "deleted" and w["path"] == "/d" for w in drain()), 5)
    c.close()


def test_watch_is_one_shot(ensemble):
    c = make_client(ensemble)
    drain = watches(c)
    c.create("/once", b"0")
    c.get("/once", watch=True)
    c.set("/once", b"1")
    assert wait_for(lambda: len([w for w in drain() if w["path"] == "/once"]) == 1, 5)
    c.set("/once", b"2")  # no watch armed now
    time.sleep(0.2)
    assert len([w for w in drain() if w["path"] == "/once"]) == 1
    c.close()


def test_child_watch_binder_pattern(ensemble):
    """A Binder-style reader: watch a service path's children, see host
    records come and go as another session registers/expires."""
    import json

    reader = make_client(ensemble)
    writer = make_client(ensemble)
    drain = watches(reader)

    reader.mkdirp("/svc/web")
    rc, ch = reader.get_children("/svc/web", watch=True)
    assert rc == ra.ZOK and ch == []

    writer.create("/svc/web/host-a", json.dumps({"type": "host", "address": "10.0.0.1"}).encode(), True)
    assert wait_for(lambda: any(w["type"] == "child" and w["path"] == "/svc/web" for w in drain()), 5)

    # re-arm, then the writer's session dies ⇒ ephemeral vanishes ⇒ child event
    rc, ch = reader.get_children("/svc/web", watch=True)
    assert ch == ["host-a"]
    ensemble.expire_session(writer.session_id())
    assert wait_for(lambda: len([w for w in drain() if w["type"] == "child"]) >= 2, 5)
    rc, ch = reader.get_children("/svc/web")
    assert ch == []
    reader.close()
    writer.close()


def test_watch_survives_across_requests(ensemble):
    # watches registered by another session don't leak to this one
    c1 = make_client(ensemble)
    c2 = make_client(ensemble)
    d1, d2 = watches(c1), watches(c2)
    c1.create("/iso", b"")
    c1.get("/iso", watch=True)
    c2.set("/iso", b"x")
    assert wait_for(lambda: any(w["path"] == "/iso" for w in d1()), 5)
    time.sleep(0.2)
    assert not any(w["path"] == "/iso" for w in d2())
    c1.close()
    c2.close()


def test_watches_survive_reconnect(ensemble3):
    """setWatches: a same-session reconnect re-arms watches; changes made
    while disconnected fire synthetic events on reconnect."""
    c = make_client(ensemble3)
    other = make_client(ensemble3)
    drain = watches(c)

    other.mkdirp("/sw")
    other.create("/sw/live", b"v0")
    c.get("/sw/live", watch=True)       # data watch
    rc, _ = c.exists("/sw/coming", watch=True)  # exist watch on missing node
    assert rc == ra.ZNONODE
    c.get_children("/sw", watch=True)   # child watch

    # find and kill the server c is attached to: kill servers until c drops,
    # but keep one up; do the changes while c is reconnecting
    sid = c.session_id()
    ensemble3.kill_server(0)
    ensemble3.kill_server(1)
    other2 = None
    # `other` may have been on a killed server too; use a fresh client for
    # the mutations (server 2 is still up)
    other.close()
    other2 = make_client(ensemble3)
    other2.set("/sw/live", b"v1")           # missed data change
    other2.create("/sw/coming", b"here")    # missed creation (child of /sw too)

    # c reconnects (same session) to server 2 and must see all three events
    assert wait_for(lambda: c.session_id() == sid and c.state() == "connected", timeout=10)
    assert wait_for(lambda: any(w["type"] == "changed" and w["path"] == "/sw/live" for w in drain()), 10)
    assert wait_for(lambda: any(w["type"] == "created" and w["path"] == "/sw/coming" for w in drain()), 10)
    assert wait_for(lambda: any(w["type"] == "child" and w["path"] == "/sw" for w in drain()), 10)
    other2.close()
    c.close()
    ensemble3.restart_server(0)
    ensemble3.restart_server(1)
