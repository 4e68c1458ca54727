"""Byte-level constrained decoding for tool-call JSON.

The reference gets well-formed tool calls for free from provider APIs
(reference: server/chat/backend/agent/agent.py:871-1047 — the SDK enforces
the schema); an in-process engine must enforce the grammar during sampling
(SURVEY.md §7 hard part 2).  This module is an incremental byte-level JSON
validator; the sampler uses it as a filter: propose top-k candidates, keep
the first whose bytes keep the FSM valid (guided rejection sampling — exact
for a byte tokenizer, a high-probability guide for BPE).
"""
from __future__ import annotations

from typing import List

# parser states
_VALUE = 0        # expecting the start of a value
_STRING = 1
_STR_ESC = 2
_STR_U = 3        # \uXXXX, counts hex digits via _u_left
_NUM_INT = 4      # in integer part
_NUM_FRAC = 5
_NUM_EXP = 6
_LITERAL = 7      # inside true/false/null
_OBJ_KEY_OR_END = 8
_OBJ_KEY = 9      # key string started (shares _STRING via _ctx)
_OBJ_COLON = 10
_OBJ_COMMA_OR_END = 11
_ARR_COMMA_OR_END = 12
_DONE = 13
_OBJ_KEY_REQUIRED = 14  # after ',' in an object: '}' illegal

_WS = b" \t\n\r"
_DIGITS = b"0123456789"


class JsonGrammar:
    """Incremental JSON validator; max_depth bounds container nesting."""

    def __init__(self, max_depth: int = 16, require_object: bool = True,
                 compact: bool = True, soft_close_bytes: int = 0):
        # compact: reject whitespace outside strings — JSON allows unbounded
        # inter-token whitespace, which lets a badly-steered model spin on
        # '\r' forever without completing; generation uses compact=True,
        # validation compact=False.
        # soft_close_bytes > 0: past that many emitted bytes the sampler is
        # steered toward closing the document (allowed_next_bytes narrows to
        # closers) so generation completes before max_tokens instead of
        # truncating mid-string. Legality is unchanged.
        self.compact = compact
        self.max_depth = max_depth
        self.soft_close_bytes = soft_close_bytes
        self.nbytes = 0
        self.stack: List[int] = []  # b'{' or b'[' markers (as ints)
        self.state = _VALUE
        self.lit: bytes = b""
        self.lit_pos = 0
        self.u_left = 0
        self.in_key = False
        self.require_object = require_object
        self.started = False
        self.num_had_digit = False
        self.arr_fresh = False  # directly after '[' (so ']' is legal, ',]' is not)

    def fork(self) -> "JsonGrammar":
        g = JsonGrammar.__new__(JsonGrammar)
        g.__dict__.update(self.__dict__)
        g.stack = list(self.stack)
        return g

    def is_complete(self) -> bool:
        return self.state == _DONE or (
            self.started and not self.stack and self.state in (_NUM_INT, _NUM_FRAC, _NUM_EXP)
            and self.num_had_digit
        )

    def advance(self, data: bytes) -> bool:
        for b in data:
            if not self._advance_byte(b):
                return False
            self.nbytes += 1
        return True

    def allowed_next_bytes(self):
        """The set of legal next bytes (as ``bytes``), or None when the
        state is effectively unconstrained (inside a string). Lets the
        sampler pick from tiny per-byte candidate buckets instead of
        scanning the vocabulary when the FSM pins the continuation.
        Computed for compact generation (no inter-token whitespace)."""
        out = self._allowed_raw()
        if self.soft_close_bytes and self.nbytes >= self.soft_close_bytes:
            if out is None:  # inside a string: steer to the closing quote
                return b'"' if self.state == _STRING else out
            closers = bytes(c for c in out if c in b'"}]')
            if closers:
                return closers
        return out

    def _allowed_raw(self):
        s = self.state
        if s == _DONE:
            return b""
        if s == _VALUE:
            if self.require_object and not self.started:
                return b"{"
            out = b'"{[-0123456789tfn'
            if self.arr_fresh and self.stack and self.stack[-1] == 0x5B:
                out += b"]"
            return out
        if s == _STRING:
            return None  # any byte >= 0x20: unconstrained enough
        if s == _STR_ESC:
            return b'"\\/bfnrtu'
        if s == _STR_U:
            return b"0123456789abcdefABCDEF"
        if s in (_NUM_INT, _NUM_FRAC, _NUM_EXP):
            out = bytes(_DIGITS)
            if s == _NUM_INT and self.num_had_digit:
                out += b".eE"
            elif s == _NUM_FRAC and self.num_had_digit:
                out += b"eE"
            elif s == _NUM_EXP and not self.num_had_digit:
                out += b"+-"
            if self.num_had_digit and self.stack:
                out += b",}" if self.stack[-1] == 0x7B else b",]"
            return out
        if s == _LITERAL:
            return bytes([self.lit[self.lit_pos]])
        if s == _OBJ_KEY_OR_END:
            return b'"}'
        if s == _OBJ_COLON:
            return b":"
        if s == _OBJ_KEY_REQUIRED:
            return b'"'
        if s == _OBJ_COMMA_OR_END:
            return b",}"
        if s == _ARR_COMMA_OR_END:
            return b",]"
        return None

    # -- single byte -------------------------------------------------------

    def _pop_value_done(self) -> bool:
        if not self.stack:
            self.state = _DONE
        elif self.stack[-1] == 0x7B:  # {
            self.state = _OBJ_COMMA_OR_END
        else:
            self.state = _ARR_COMMA_OR_END
        return True

    def _start_value(self, b: int) -> bool:
        if b in _WS:
            return True
        self.started = True
        if b == 0x5D and self.arr_fresh and self.stack and self.stack[-1] == 0x5B:
            # empty array "[]" (']' legal only directly after '[')
            self.arr_fresh = False
            self.stack.pop()
            return self._pop_value_done()
        self.arr_fresh = False
        if b == 0x22:  # "
            self.state = _STRING
            self.in_key = False
            return True
        if b == 0x7B:  # {
            if len(self.stack) >= self.max_depth:
                return False
            self.stack.append(0x7B)
            self.state = _OBJ_KEY_OR_END
            return True
        if b == 0x5B:  # [
            if len(self.stack) >= self.max_depth:
                return False
            self.stack.append(0x5B)
            self.state = _VALUE
            self.arr_fresh = True
            return True
        if b in _DIGITS or b == 0x2D:  # digit or -
            self.state = _NUM_INT
            self.num_had_digit = b != 0x2D
            return True
        for lit in (b"true", b"false", b"null"):
            if b == lit[0]:
                self.lit, self.lit_pos, self.state = lit, 1, _LITERAL
                return True
        return False

    def _advance_byte(self, b: int) -> bool:
        s = self.state
        if self.compact and b in _WS and s not in (_STRING, _STR_ESC, _STR_U):
            return False
        if s == _DONE:
            return b in _WS
        if s == _VALUE:
            if self.require_object and not self.started and b not in _WS:
                if b != 0x7B:
                    return False
            return self._start_value(b)
        if s in (_STRING,):
            if b == 0x22:
                if self.in_key:
                    self.state = _OBJ_COLON
                else:
                    return self._pop_value_done()
                return True
            if b == 0x5C:  # backslash
                self.state = _STR_ESC
                return True
            return b >= 0x20  # control chars illegal; raw utf-8 bytes ok
        if s == _STR_ESC:
            if b == 0x75:  # u
                self.state = _STR_U
                self.u_left = 4
                return True
            if b in b'"\\/bfnrt':
                self.state = _STRING
                return True
            return False
        if s == _STR_U:
            if b in b"0123456789abcdefABCDEF":
                self.u_left -= 1
                if self.u_left == 0:
                    self.state = _STRING
                return True
            return False
        if s in (_NUM_INT, _NUM_FRAC, _NUM_EXP):
            if b in _DIGITS:
                self.num_had_digit = True
                return True
            if s == _NUM_INT and b == 0x2E and self.num_had_digit:  # .
                self.state = _NUM_FRAC
                self.num_had_digit = False
                return True
            if s in (_NUM_INT, _NUM_FRAC) and b in b"eE" and self.num_had_digit:
                self.state = _NUM_EXP
                self.num_had_digit = False
                return True
            if s == _NUM_EXP and b in b"+-" and not self.num_had_digit:
                return True
            if not self.num_had_digit:
                return False
            # number terminated by this byte: re-dispatch in the parent state
            self._pop_value_done()
            return self._advance_byte(b)
        if s == _LITERAL:
            if self.lit_pos < len(self.lit) and b == self.lit[self.lit_pos]:
                self.lit_pos += 1
                if self.lit_pos == len(self.lit):
                    return self._pop_value_done()
                return True
            return False
        if s == _OBJ_KEY_OR_END:
            if b in _WS:
                return True
            if b == 0x7D:  # }
                self.stack.pop()
                return self._pop_value_done()
            if b == 0x22:
                self.state = _STRING
                self.in_key = True
                return True
            return False
        if s == _OBJ_COLON:
            if b in _WS:
                return True
            if b == 0x3A:  # :
                self.state = _VALUE
                self.in_key = False
                return True
            return False
        if s == _OBJ_KEY_REQUIRED:
            if b in _WS:
                return True
            if b == 0x22:
                self.state = _STRING
                self.in_key = True
                return True
            return False
        if s == _OBJ_COMMA_OR_END:
            if b in _WS:
                return True
            if b == 0x2C:  # ,
                self.state = _OBJ_KEY_REQUIRED
                return True
            if b == 0x7D:
                self.stack.pop()
                return self._pop_value_done()
            return False
        if s == _ARR_COMMA_OR_END:
            if b in _WS:
                return True
            if b == 0x2C:
                self.state = _VALUE
                return True
            if b == 0x5D:
                self.stack.pop()
                return self._pop_value_done()
            return False
        return False


def validate_json_bytes(data: bytes) -> bool:
    g = JsonGrammar(require_object=False, compact=False)
    return g.advance(data) and g.is_complete()


class ReActTurnGrammar:
    """One grammar for a whole ReAct turn: free-text *thought* bytes, then
    the literal ``\\nAction: `` transition, then a :class:`ToolCallGrammar`
    tool call. Lets the agent decode thought+action in a SINGLE engine
    request (one prefill, one stream) instead of two — the round-1
    workflow re-prefilled the prompt for the action call and paid an extra
    host round-trip per turn.

    The thought phase is unconstrained; past ``thought_bytes`` the sampler
    is steered onto the transition literal (legality unchanged). The
    emitted text is ``<thought>\\nAction: {"tool":...,"args":{...}}``.
    """

    LIT = b"\nAction:"

    def __init__(self, tools: List[str], thought_bytes: int = 400,
                 tool_soft_close: int = 160, max_args_depth: int = 8):
        self.tool = ToolCallGrammar(tools, max_args_depth=max_args_depth,
                                    soft_close_bytes=tool_soft_close)
        self.thought_bytes = thought_bytes
        self.phase = 0      # 0 = thought, 1 = tool call
        self.lit_pos = 0    # progress through LIT (match or forced)
        self.sp_ok = True   # one optional space after the colon
        self.nbytes = 0

    def fork(self) -> "ReActTurnGrammar":
        g = ReActTurnGrammar.__new__(ReActTurnGrammar)
        g.tool = self.tool.fork()
        g.thought_bytes = self.thought_bytes
        g.phase = self.phase
        g.lit_pos = self.lit_pos
        g.sp_ok = self.sp_ok
        g.nbytes = self.nbytes
        return g

    def is_complete(self) -> bool:
        return self.phase == 1 and self.tool.is_complete()

    def tool_name(self) -> str:
        return self.tool.tool_name()

    def advance(self, data: bytes) -> bool:
        for b in data:
            if not self._advance_byte(b):
                return False
            self.nbytes += 1
        return True

    def _advance_byte(self, b: int) -> bool:
        if self.phase == 1:
            if self.sp_ok and b == 0x20:
                self.sp_ok = False
                return True
            self.sp_ok = False
            ok = self.tool._advance_byte(b)
            if ok:
                self.tool.nbytes += 1  # keep its soft-close budget live
            return ok
        # thought phase: every byte is legal; track the transition literal
        if self.nbytes >= self.thought_bytes:
            # over budget the steering is STRICT: a multi-byte token whose
            # tail diverges from the literal must be rejected outright —
            # under a BPE tokenizer a permissive mismatch would keep
            # resetting the match and the turn would never transition
            if b == self.LIT[self.lit_pos]:
                self.lit_pos += 1
                if self.lit_pos == len(self.LIT):
                    self.phase = 1
                return True
            return False
        if b == self.LIT[self.lit_pos]:
            self.lit_pos += 1
            if self.lit_pos == len(self.LIT):
                self.phase = 1
        elif b == self.LIT[0]:
            self.lit_pos = 1
        else:
            self.lit_pos = 0
        return True

    def allowed_next_bytes(self):
        if self.phase == 1:
            inner = self.tool.allowed_next_bytes()
            if self.sp_ok and inner is not None:
                return bytes({0x20} | set(inner))
            return inner
        if self.nbytes >= self.thought_bytes:
            # over the thought budget: steer onto the transition literal
            return bytes([self.LIT[self.lit_pos]])
        return None


class ToolCallGrammar:
    """Constrains decoding to exactly ``{"tool":"<name>","args":{...}}``
    with ``<name>`` drawn from a fixed set — the in-engine analog of the
    reference's forced tool choice (reference: middleware/force_tool.py,
    used agent.py:615-622) plus the provider SDK's schema enforcement.

    Byte-level FSM: literal prefix -> trie over tool names -> literal
    ``","args":`` -> a nested :class:`JsonGrammar` object -> closing ``}``.
    Same interface as JsonGrammar (fork/advance/is_complete), so the
    engine's guided sampler uses it unchanged. With this grammar a model
    can never emit an unknown tool or malformed call — one sampling
    attempt always parses.
    """

    _PREFIX = b'{"tool":"'
    _MID = b'","args":'

    def __init__(self, tools: List[str], max_args_depth: int = 8,
                 soft_close_bytes: int = 160):
        if not tools:
            raise ValueError("ToolCallGrammar needs at least one tool name")
        self.names = sorted(set(t.encode() for t in tools))
        self.phase = 0        # 0 prefix, 1 name, 2 mid, 3 args, 4 done
        self.lit_pos = 0
        self.name_prefix = b""
        self.max_args_depth = max_args_depth
        self.args: JsonGrammar = None  # created lazily at phase 3
        # past this many bytes the sampler is steered toward closing the
        # call (so it completes before the token budget instead of being
        # truncated mid-string into unparseable JSON)
        self.soft_close_bytes = soft_close_bytes
        self.nbytes = 0

    def fork(self) -> "ToolCallGrammar":
        g = ToolCallGrammar.__new__(ToolCallGrammar)
        g.names = self.names  # immutable, shared
        g.phase = self.phase
        g.lit_pos = self.lit_pos
        g.name_prefix = self.name_prefix
        g.max_args_depth = self.max_args_depth
        g.args = self.args.fork() if self.args is not None else None
        g.soft_close_bytes = self.soft_close_bytes
        g.nbytes = self.nbytes
        return g

    def is_complete(self) -> bool:
        return self.phase == 4

    def tool_name(self) -> str:
        """The accepted tool name (valid once phase >= 2)."""
        return self.name_prefix.decode()

    def advance(self, data: bytes) -> bool:
        for b in data:
            if not self._advance_byte(b):
                return False
            self.nbytes += 1
        return True

    def allowed_next_bytes(self):
        """Legal next bytes for the sampler's bucket pick (see JsonGrammar)."""
        if self.phase == 0:
            return bytes([self._PREFIX[self.lit_pos]])
        if self.phase == 1:
            n = len(self.name_prefix)
            out = {m[n] for m in self.names
                   if len(m) > n and m.startswith(self.name_prefix)}
            if self.name_prefix in self.names:
                out.add(0x22)  # closing quote
            return bytes(sorted(out))
        if self.phase == 2:
            return bytes([self._MID[self.lit_pos]])
        if self.phase == 3:
            if self.args.is_complete():
                return b"}"
            inner = self.args.allowed_next_bytes()
            if self.nbytes >= self.soft_close_bytes:
                # over budget: steer toward closing (legality unchanged —
                # this only narrows what the sampler is offered)
                if inner is None:  # inside a string: close it
                    return b'"'
                closers = bytes(c for c in inner if c in b'"}]')
                if closers:
                    return closers
            return inner
        return b""  # phase 4: only EOS

    def advance_bytes_hint(self) -> int:
        return self.nbytes

    def _advance_byte(self, b: int) -> bool:
        if self.phase == 0:
            if b != self._PREFIX[self.lit_pos]:
                return False
            self.lit_pos += 1
            if self.lit_pos == len(self._PREFIX):
                self.phase, self.lit_pos = 1, 0
            return True
        if self.phase == 1:
            if b == 0x22:  # '"' closes the name: must match exactly
                if self.name_prefix in self.names:
                    self.phase, self.lit_pos = 2, 1  # consumed MID[0]='"'
                    return True
                return False
            cand = self.name_prefix + bytes([b])
            if any(n.startswith(cand) for n in self.names):
                self.name_prefix = cand
                return True
            return False
        if self.phase == 2:
            if b != self._MID[self.lit_pos]:
                return False
            self.lit_pos += 1
            if self.lit_pos == len(self._MID):
                self.phase = 3
                self.args = JsonGrammar(max_depth=self.max_args_depth,
                                        require_object=True, compact=True)
            return True
        if self.phase == 3:
            if self.args.is_complete():
                if b == 0x7D:  # closing } of the outer object
                    self.phase = 4
                    return True
                return False
            return self.args._advance_byte(b)
        return False  # phase 4: nothing after the closing brace
