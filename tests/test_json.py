"""JSON codec semantics: the payload byte-shape contract.

The reference serializes payloads with JSON.stringify, whose key behaviors we
must preserve (reference: lib/register.js:140-159 drops `ttl` when undefined;
test/register.test.js:122-153 asserts exact shapes)."""
import json

import pytest

import registrar_amd as ra
from registrar_amd import _core


def roundtrip(s):
    return _core.json_roundtrip(s)


def test_roundtrip_scalars():
    assert roundtrip("null") == "null"
    assert roundtrip("true") == "true"
    assert roundtrip("false") == "false"
    assert roundtrip("42") == "42"
    assert roundtrip("-7") == "-7"
    assert roundtrip('"hi"') == '"hi"'
    assert roundtrip("1.5") == "1.5"


def test_roundtrip_preserves_key_order():
    s = '{"b":1,"a":2,"z":{"y":3,"x":4}}'
    assert roundtrip(s) == s


def test_nested_and_arrays():
    s = '{"a":[1,2,{"b":[]}],"c":{}}'
    assert roundtrip(s) == s


def test_string_escapes():
    assert roundtrip('"a\\nb\\tc\\"d\\\\e"') == '"a\\nb\\tc\\"d\\\\e"'
    # unicode escape becomes UTF-8
    assert json.loads(roundtrip('"\\u00e9"')) == "é"
    # surrogate pair
    assert json.loads(roundtrip('"\\ud83d\\ude00"')) == "\U0001f600"


def test_whitespace_tolerated():
    assert roundtrip(' { "a" : [ 1 , 2 ] } ') == '{"a":[1,2]}'


def test_equality_order_insensitive():
    assert _core.json_equal('{"a":1,"b":2}', '{"b":2,"a":1}')
    assert not _core.json_equal('{"a":1}', '{"a":2}')
    assert _core.json_equal("[1,2]", "[1,2]")
    assert not _core.json_equal("[1,2]", "[2,1]")


@pytest.mark.parametrize("bad", ["", "{", '{"a"}', "[1,", '"unterminated', "tru", "01x", '{"a":1}]'])
def test_parse_errors(bad):
    with pytest.raises(RuntimeError):
        roundtrip(bad)


def test_large_numbers():
    assert roundtrip("1152921504606846976") == "1152921504606846976"  # 2^60


def test_payload_has_no_null_ttl():
    """ttl must be ABSENT (not null) when unset — JSON.stringify semantics."""
    rec = json.loads(ra.build_host_record(json.dumps({"domain": "a.b", "type": "host", "adminIp": "1.2.3.4"})))
    assert "ttl" not in rec


def test_depth_cap_no_stack_overflow():
    """A hostile deeply-nested document must raise a parse error, not
    overflow the recursive-descent stack (previously SIGSEGV at ~100k deep)."""
    deep = "[" * 100000 + "]" * 100000
    with pytest.raises(RuntimeError, match="nesting too deep"):
        roundtrip(deep)
    # legitimate nesting well below the cap still parses
    ok = "[" * 200 + "0" + "]" * 200
    assert roundtrip(ok) == ok
