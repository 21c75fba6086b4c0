"""Health checker: exec, timeout, regex, flap damping.

Mirrors the reference's test/health.test.js cases (ok, ignoreExitStatus,
exit-fail, timeout-fail, stdout-regex fail, threshold mark-down transition)
plus the fixed behaviors: invert honored, true sliding window, recovery
resets the window."""
import json
import time

import pytest

import registrar_amd as ra


def hc(config, **kw):
    config = dict(config)
    return ra.HealthCheck(json.dumps(config), **kw)


def test_exec_ok():
    res = ra.exec_with_timeout("true", 1000)
    assert res["exit_status"] == 0 and not res["timed_out"]


def test_exec_capture_stdout():
    res = ra.exec_with_timeout("echo hello world", 1000)
    assert res["stdout"] == b"hello world\n"


def test_exec_exit_status():
    assert ra.exec_with_timeout("exit 3", 1000)["exit_status"] == 3


def test_exec_timeout():
    t0 = time.monotonic()
    res = ra.exec_with_timeout("sleep 5", 200)
    assert res["timed_out"]
    assert time.monotonic() - t0 < 3.0


def test_exec_timeout_kills_stubborn():
    # a shell that ignores SIGTERM must be SIGKILLed
    t0 = time.monotonic()
    res = ra.exec_with_timeout("trap '' TERM; sleep 5", 200)
    assert res["timed_out"]
    assert time.monotonic() - t0 < 4.0


def test_check_ok():
    # reference test/health.test.js:29-52
    h = hc({"command": "true", "interval": 60000})
    rec = h.check_once()
    assert rec["type"] == "ok" and not rec["isDown"]


def test_ignore_exit_status():
    # reference test/health.test.js:56-80
    h = hc({"command": "false", "ignoreExitStatus": True})
    assert h.check_once()["type"] == "ok"


def test_exit_failure():
    # reference test/health.test.js:83-112
    h = hc({"command": "false", "threshold": 5})
    rec = h.check_once()
    assert rec["type"] == "fail" and rec["failures"] == 1 and not rec["isDown"]


def test_timeout_failure():
    # reference test/health.test.js:115-145 (timeout 10ms vs sleep 2)
    h = hc({"command": "sleep 2", "timeout": 100})
    rec = h.check_once()
    assert rec["type"] == "fail"
    assert "timed out" in rec["error"]


def test_stdout_match_failure():
    # reference test/health.test.js:148-180
    h = hc({"command": "echo bad", "stdoutMatch": {"pattern": "^good$"}})
    rec = h.check_once()
    assert rec["type"] == "fail"
    assert "stdout match" in rec["error"]
    assert rec["exit_status"] == -1  # code -1 on regex failure (lib/health.js:107)


def test_stdout_match_multiline_ok():
    h = hc({"command": "printf 'a\\ngood\\nb\\n'", "stdoutMatch": {"pattern": "good"}})
    assert h.check_once()["type"] == "ok"


def test_stdout_match_flags_icase():
    h = hc({"command": "echo GOOD", "stdoutMatch": {"pattern": "good", "flags": "i"}})
    assert h.check_once()["type"] == "ok"


def test_stdout_match_invert():
    # invert was accepted-but-ignored in the reference; here
    # invert=true means a MATCH is a failure
    h = hc({"command": "echo ERROR: broken", "stdoutMatch": {"pattern": "ERROR", "invert": True}})
    assert h.check_once()["type"] == "fail"
    h2 = hc({"command": "echo all fine", "stdoutMatch": {"pattern": "ERROR", "invert": True}})
    assert h2.check_once()["type"] == "ok"


def test_threshold_markdown_transition():
    # reference test/health.test.js:183-225: isDown flips on the Nth failure
    h = hc({"command": "false", "threshold": 3, "period": 60000})
    r1 = h.check_once()
    r2 = h.check_once()
    assert not r1["isDown"] and not r2["isDown"]
    r3 = h.check_once()
    assert r3["isDown"] and r3["failures"] == 3
    assert h.is_down()


def test_sliding_window_evicts_old_failures():
    # failures outside the trailing period must not count (fix of §2.2.2)
    h = hc({"command": "false", "threshold": 3, "period": 300})
    h.check_once()
    h.check_once()
    time.sleep(0.4)  # both failures age out of the 300ms window
    r3 = h.check_once()
    assert r3["failures"] == 1 and not r3["isDown"]


def test_recovery_resets_window():
    # after ok-while-down, re-marking down needs `threshold` fresh failures
    h = hc({"command": "sh -c 'exit 1'", "threshold": 2, "period": 60000})
    h.check_once()
    r2 = h.check_once()
    assert r2["isDown"]
    ok = hc({"command": "true", "threshold": 2})
    # simulate recovery against the same checker: flip command is not
    # supported, so exercise via the real flow below instead
    del ok
    # use a file-gated command for a true flap sequence
    import os
    import tempfile

    gate = tempfile.NamedTemporaryFile(delete=False)
    gate.close()
    h2 = hc({"command": "test -e %s" % gate.name, "threshold": 2, "period": 60000})
    assert h2.check_once()["type"] == "ok"
    os.unlink(gate.name)
    h2.check_once()
    assert h2.check_once()["isDown"]
    open(gate.name, "w").close()
    assert h2.check_once()["type"] == "ok"
    assert not h2.is_down()
    os.unlink(gate.name)
    r = h2.check_once()
    assert r["failures"] == 1 and not r["isDown"]  # window was reset
    os.path.exists(gate.name) and os.unlink(gate.name)


def test_periodic_runner():
    h = hc({"command": "true", "interval": 50})
    h.start()
    time.sleep(0.4)
    h.stop()
    recs = h.poll_records()
    assert len(recs) >= 3
    assert all(r["type"] == "ok" for r in recs)


def test_config_validation():
    with pytest.raises(RuntimeError, match="command"):
        ra.HealthCheck(json.dumps({"interval": 10}))


def test_stdout_match_flags_multiline():
    # 'm' flag: ^/$ match per line (ECMAScript multiline)
    h = hc({"command": "printf 'noise\\nOK\\nnoise\\n'", "stdoutMatch": {"pattern": "^OK$", "flags": "m"}})
    assert h.check_once()["type"] == "ok"
    h2 = hc({"command": "printf 'noise\\nOK\\nnoise\\n'", "stdoutMatch": {"pattern": "^OK$"}})
    assert h2.check_once()["type"] == "fail"  # without m, ^$ anchor whole text


# ---- round 2: stdout/stderr separation + regex fidelity hardening ----
# (VERDICT r1 Weak #1 / next-round items 2 and 5)


def test_exec_stderr_captured_separately():
    res = ra.exec_with_timeout("echo out; echo err 1>&2", 1000)
    assert res["stdout"] == b"out\n"
    assert res["stderr"] == b"err\n"


def test_stdout_match_ignores_stderr():
    # the reference matches child.exec's STDOUT arg only
    # (/root/reference/lib/health.js:89-101): a pattern that appears only on
    # stderr must NOT satisfy the match
    h = hc({"command": "echo HEALTHY 1>&2", "stdoutMatch": {"pattern": "HEALTHY"}})
    rec = h.check_once()
    assert rec["type"] == "fail"
    assert "HEALTHY" in rec["stderr"]  # preserved for diagnostics
    # same text on stdout: ok
    h2 = hc({"command": "echo HEALTHY", "stdoutMatch": {"pattern": "HEALTHY"}})
    assert h2.check_once()["type"] == "ok"


def test_stdout_match_invert_ignores_stderr():
    # invert converse: a forbidden pattern on stderr only must not FAIL the
    # check (invert fails when the pattern matches stdout)
    h = hc({"command": "echo BROKEN 1>&2", "stdoutMatch": {"pattern": "BROKEN", "invert": True}})
    assert h.check_once()["type"] == "ok"
    h2 = hc({"command": "echo BROKEN", "stdoutMatch": {"pattern": "BROKEN", "invert": True}})
    assert h2.check_once()["type"] == "fail"


def test_invalid_pattern_rejected_at_parse():
    # fail fast at config time (reference: assert-plus throws at construction,
    # lib/health.js:23-38); previously an invalid pattern silently never matched
    with pytest.raises(RuntimeError, match="invalid regex"):
        hc({"command": "true", "stdoutMatch": {"pattern": "(unclosed"}})


def test_unknown_flags_rejected_at_parse():
    with pytest.raises(RuntimeError, match="unsupported flag"):
        hc({"command": "true", "stdoutMatch": {"pattern": "x", "flags": "s"}})
    # 'g' is a harmless no-op for a single search and is accepted
    h = hc({"command": "echo zz", "stdoutMatch": {"pattern": "z", "flags": "gi"}})
    assert h.check_once()["type"] == "ok"


def test_background_grandchild_does_not_hang():
    # a backgrounded grandchild keeps the pipe write end open after the shell
    # exits; the post-reap drain must be bounded (health.cpp drain)
    t0 = time.monotonic()
    res = ra.exec_with_timeout("echo done; sleep 5 & exit 0", 3000)
    elapsed = time.monotonic() - t0
    assert res["exit_status"] == 0 and not res["timed_out"]
    assert b"done" in res["stdout"]
    assert elapsed < 2.0  # returned long before the grandchild's 5 s sleep


def test_fail_record_carries_stderr_tail():
    h = hc({"command": "echo diagnostic-detail 1>&2; exit 1"})
    rec = h.check_once()
    assert rec["type"] == "fail"
    assert "diagnostic-detail" in rec["stderr"]
