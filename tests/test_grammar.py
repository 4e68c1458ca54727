"""Byte-level JSON grammar FSM tests."""
import json

import pytest

from aurora_amd.engine.grammar import JsonGrammar, validate_json_bytes


GOOD = [
    '{}', '{"a": 1}', '{"a": {"b": [1, 2, 3]}}', '{"s": "he\\"llo"}',
    '{"n": -1.5e-3}', '{"b": true, "c": false, "d": null}', '{"u": "\\u00e9"}',
    '{"arr": []}', '{ "x" :\n[ {"y": "z"} , 2 ] }',
]
BAD = [
    '{', '{"a" 1}', '{"a": }', "{'a': 1}", '{"a": 1,}', '{"a": 01e}',
    '{"a": tru}', '[1, 2]',  # require_object rejects top-level arrays
    '{"a": [1,]}', '{"a": "\\x"}', '{"a": 1} extra',
]


@pytest.mark.parametrize("s", GOOD)
def test_good(s):
    g = JsonGrammar(compact=False)
    assert g.advance(s.encode()), s
    assert g.is_complete(), s
    json.loads(s)  # sanity: python agrees


@pytest.mark.parametrize("s", BAD)
def test_bad(s):
    g = JsonGrammar(compact=False)
    ok = g.advance(s.encode()) and g.is_complete()
    assert not ok, s


def test_incremental_byte_at_a_time():
    s = b'{"tool":"kubectl","args":{"cmd":"get pods","n":3.5}}'
    g = JsonGrammar()
    for i in range(len(s)):
        assert g.advance(s[i:i+1]), s[:i+1]
        if i < len(s) - 1:
            assert not g.is_complete()
    assert g.is_complete()


def test_fork_independent():
    g = JsonGrammar(compact=False)
    assert g.advance(b'{"a"')
    f = g.fork()
    assert f.advance(b': 1}')
    assert f.is_complete()
    assert not g.is_complete()
    assert g.advance(b': 2}')


def test_validate_non_object_allowed():
    assert validate_json_bytes(b'[1, 2, 3]')
    assert validate_json_bytes(b'"str"')
    assert validate_json_bytes(b'42')
    assert not validate_json_bytes(b'42x')


def test_utf8_in_strings():
    s = '{"k": "héllo ✓"}'.encode()
    assert validate_json_bytes(s)


def test_compact_rejects_whitespace():
    g = JsonGrammar()  # compact by default
    assert not g.fork().advance(b' ')
    assert not g.fork().advance(b'{ ')
    assert g.fork().advance(b'{"a":1}')
    g2 = JsonGrammar()
    assert g2.advance(b'{"a b":" c "}')  # ws inside strings still fine
    assert g2.is_complete()
