"""Property-based fuzzing (hypothesis) for the byte-level JSON grammar
FSM (engine/grammar.py — it guards the tool-call constrained decode
path).  Default config is require_object=True + compact=True (tool calls
are single compact objects); general-value acceptance is fuzzed with
those relaxed."""
import json

from hypothesis import given, settings, strategies as st

from aurora_amd.engine.grammar import JsonGrammar

json_values = st.recursive(
    st.none() | st.booleans()
    | st.integers(min_value=-10**12, max_value=10**12)
    | st.floats(allow_nan=False, allow_infinity=False, width=32)
    | st.text(max_size=40),
    lambda children: st.lists(children, max_size=4)
    | st.dictionaries(st.text(max_size=10), children, max_size=4),
    max_leaves=12)

json_objects = st.dictionaries(st.text(max_size=10), json_values, max_size=4)


@settings(max_examples=150, deadline=None)
@given(json_values)
def test_grammar_accepts_any_value_relaxed(value):
    doc = json.dumps(value)  # default separators include spaces
    g = JsonGrammar(require_object=False, compact=False)
    assert g.advance(doc.encode()), doc
    assert g.is_complete(), doc


@settings(max_examples=150, deadline=None)
@given(json_objects)
def test_grammar_default_accepts_compact_objects(value):
    doc = json.dumps(value, separators=(",", ":"))
    g = JsonGrammar()
    assert g.advance(doc.encode()), doc
    assert g.is_complete(), doc


@settings(max_examples=100, deadline=None)
@given(json_objects, st.integers(min_value=0, max_value=3))
def test_grammar_rejects_trailing_garbage(value, which):
    doc = json.dumps(value, separators=(",", ":"))
    garbage = ["}", "]", ",", "x"][which]
    g = JsonGrammar()
    assert g.advance(doc.encode()), doc
    assert g.is_complete(), doc
    g2 = g.fork()
    assert not g2.advance(garbage.encode()), doc + garbage


@settings(max_examples=100, deadline=None)
@given(json_objects)
def test_grammar_fork_isolation(value):
    """fork() must not mutate the parent (the sampler walks candidates
    by forking per token)."""
    doc = json.dumps(value, separators=(",", ":")).encode()
    g = JsonGrammar()
    half = max(1, len(doc) // 2)
    assert g.advance(doc[:half])
    bad = g.fork()
    bad.advance(b"\x00\xff")  # likely-invalid bytes into the fork only
    g2 = g.fork()
    assert g2.advance(doc[half:]), doc
    assert g2.is_complete()


@settings(max_examples=120, deadline=None)
@given(st.binary(max_size=24))
def test_grammar_random_bytes_never_crash(data):
    """Arbitrary byte soup must never raise — only return False or keep
    a valid in-progress state."""
    g = JsonGrammar()
    ok = g.advance(data)
    assert ok in (True, False)
    if ok:
        g.fork().advance(b"}")  # further feeding must also not raise
