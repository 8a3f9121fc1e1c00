"""Guided decoding: OpenAI ``response_format: {"type": "json_object"}``.

A byte-level JSON pushdown automaton constrains sampling so the generated
text is always a prefix of valid JSON (the reference delegates this to
vLLM's guided-decoding backends; this is a dependency-free equivalent).

Design:
- The PDA tracks a small set of byte-class states plus a container stack.
  Which bytes are legal depends only on a compact SIGNATURE
  (state, stack-top), so per-signature *token* masks — "which vocab ids
  keep the machine alive" — are memoized and shared across requests and
  steps. A token is allowed iff ALL its bytes advance the machine.
- Each request holds one machine; the chosen token's bytes advance it.
- Once the top-level value completes, only whitespace and EOS remain
  legal, so generation terminates naturally.

Token byte strings come from the tokenizer (decoded per id, cached); with
the tokenizer-less byte fallback, id == byte.
"""

from __future__ import annotations

from typing import Dict, List, Optional, Sequence, Tuple

# states
EXPECT_VALUE = 0       # start of any JSON value
IN_STRING = 1
IN_STRING_ESCAPE = 2
IN_NUMBER = 3          # number substate lives in ``lit_pos`` (N_*)
IN_LITERAL = 4         # true/false/null (position tracked separately)
AFTER_VALUE = 5        # value finished; container decides what's next
EXPECT_KEY = 6         # inside object, before a key string
EXPECT_COLON = 7
DONE = 8               # top-level value complete

WS = frozenset(b" \t\n\r")
DIGITS = frozenset(b"0123456789")
NUMBER_BODY = frozenset(b"0123456789+-.eE")
STRING_ESCAPES = frozenset(b'"\\/bfnrtu')
LITERALS = {b"true", b"false", b"null"}

# number substates (stored in lit_pos while state == IN_NUMBER)
N_SIGN = 0        # '-' seen, need first digit
N_ZERO = 1        # leading 0: only '.', 'e' or end may follow
N_INT = 2         # in integer part
N_FRAC_START = 3  # '.' seen, need a digit
N_FRAC = 4        # in fraction digits
N_EXP_START = 5   # 'e' seen, need sign or digit
N_EXP_SIGN = 6    # exponent sign seen, need a digit
N_EXP = 7         # in exponent digits
_NUM_TERMINAL = frozenset((N_ZERO, N_INT, N_FRAC, N_EXP))


class JsonMachine:
    """Byte-at-a-time JSON acceptor with container stack."""

    __slots__ = ("state", "stack", "lit", "lit_pos", "key_mode",
                 "object_only")

    def __init__(self, object_only: bool = True):
        self.state = EXPECT_VALUE
        self.stack: List[int] = []  # ord('{') or ord('[')
        self.lit: bytes = b""
        self.lit_pos = 0
        self.key_mode = False  # the string being read is an object key
        # OpenAI json_object: the top-level value must be an object
        self.object_only = object_only

    def clone(self) -> "JsonMachine":
        m = JsonMachine.__new__(JsonMachine)
        m.object_only = self.object_only
        m.state = self.state
        m.stack = list(self.stack)
        m.lit = self.lit
        m.lit_pos = self.lit_pos
        m.key_mode = self.key_mode
        return m

    def signature(self) -> Tuple:
        top = self.stack[-1] if self.stack else 0
        return (self.state, top, self.lit, self.lit_pos, self.key_mode,
                len(self.stack) == 0, self.object_only)

    # -- transitions ---------------------------------------------------------
    def _finish_value(self) -> None:
        if not self.stack:
            self.state = DONE
        else:
            self.state = AFTER_VALUE

    def advance(self, b: int) -> bool:
        """Consume one byte; False = illegal."""
        s = self.state
        if s == DONE:
            return b in WS
        if s == IN_STRING:
            if b == 0x22:  # closing quote
                if self.key_mode:
                    self.key_mode = False
                    self.state = EXPECT_COLON
                else:
                    self._finish_value()
                return True
            if b == 0x5C:
                self.state = IN_STRING_ESCAPE
                return True
            return b >= 0x20  # control bytes banned; UTF-8 bytes pass
        if s == IN_STRING_ESCAPE:
            if b in STRING_ESCAPES:
                self.state = IN_STRING
                return True
            return False
        if s == IN_NUMBER:
            ns = self.lit_pos
            if b in DIGITS:
                if ns == N_ZERO:
                    return False  # no leading zeros
                self.lit_pos = {
                    N_SIGN: N_ZERO if b == 0x30 else N_INT,
                    N_INT: N_INT,
                    N_FRAC_START: N_FRAC,
                    N_FRAC: N_FRAC,
                    N_EXP_START: N_EXP,
                    N_EXP_SIGN: N_EXP,
                    N_EXP: N_EXP,
                }[ns]
                return True
            if b == 0x2E:  # .
                if ns in (N_ZERO, N_INT):
                    self.lit_pos = N_FRAC_START
                    return True
                return False
            if b in (0x65, 0x45):  # e / E
                if ns in (N_ZERO, N_INT, N_FRAC):
                    self.lit_pos = N_EXP_START
                    return True
                return False
            if b in (0x2B, 0x2D):  # + / -
                if ns == N_EXP_START:
                    self.lit_pos = N_EXP_SIGN
                    return True
                return False
            if ns not in _NUM_TERMINAL:
                return False  # dangling '-', '.', or exponent
            # number ended; the byte belongs to the enclosing context
            self._finish_value()
            return self.advance(b)
        if s == IN_LITERAL:
            if self.lit_pos < len(self.lit) and b == self.lit[self.lit_pos]:
                self.lit_pos += 1
                if self.lit_pos == len(self.lit):
                    self._finish_value()
                return True
            return False
        if s == EXPECT_VALUE:
            if b in WS:
                return True
            if self.object_only and not self.stack and b != 0x7B:
                return False  # top level must open an object
            if b == 0x7B:  # {
                self.stack.append(0x7B)
                self.state = EXPECT_KEY
                return True
            if b == 0x5B:  # [
                self.stack.append(0x5B)
                self.state = EXPECT_VALUE
                return True
            if b == 0x22:
                self.state = IN_STRING
                self.key_mode = False
                return True
            if b in DIGITS or b == 0x2D:  # digit or -
                self.state = IN_NUMBER
                self.lit_pos = (
                    N_SIGN if b == 0x2D
                    else (N_ZERO if b == 0x30 else N_INT)
                )
                return True
            for lit in LITERALS:
                if b == lit[0]:
                    self.state = IN_LITERAL
                    self.lit = lit
                    self.lit_pos = 1
                    return True
            if b == 0x5D and self.stack and self.stack[-1] == 0x5B:
                # empty array "[]"
                self.stack.pop()
                self._finish_value()
                return True
            return False
        if s == EXPECT_KEY:
            if b in WS:
                return True
            if b == 0x22:
                self.state = IN_STRING
                self.key_mode = True
                return True
            if b == 0x7D and self.stack and self.stack[-1] == 0x7B:
                # empty object "{}"
                self.stack.pop()
                self._finish_value()
                return True
            return False
        if s == EXPECT_COLON:
            if b in WS:
                return True
            if b == 0x3A:  # :
                self.state = EXPECT_VALUE
                return True
            return False
        if s == AFTER_VALUE:
            if b in WS:
                return True
            top = self.stack[-1] if self.stack else 0
            if b == 0x2C:  # ,
                self.state = EXPECT_KEY if top == 0x7B else EXPECT_VALUE
                return True
            if b == 0x7D and top == 0x7B:
                self.stack.pop()
                self._finish_value()
                return True
            if b == 0x5D and top == 0x5B:
                self.stack.pop()
                self._finish_value()
                return True
            return False
        return False

    def accepts(self, data: bytes) -> bool:
        for b in data:
            if not self.advance(b):
                return False
        return True

    @property
    def complete(self) -> bool:
        return self.state == DONE


class GuidedJsonProcessor:
    """Per-vocabulary guided-JSON masks with signature memoization."""

    def __init__(self, token_bytes: Sequence[bytes], eos_token_id: Optional[int]):
        self.token_bytes = token_bytes
        self.eos_token_id = eos_token_id
        self._mask_cache: Dict[Tuple, List[int]] = {}

    @classmethod
    def from_tokenizer(cls, tokenizer, vocab_size: int,
                       eos_token_id: Optional[int]) -> "GuidedJsonProcessor":
        toks: List[bytes] = []
        for i in range(vocab_size):
            if tokenizer is None:
                toks.append(bytes([i]) if i < 256 else b"")
            else:
                try:
                    toks.append(tokenizer.decode([i]).encode("utf-8"))
                except Exception:
                    toks.append(b"")
        return cls(toks, eos_token_id)

    def allowed_tokens(self, machine: JsonMachine) -> List[int]:
        sig = machine.signature()
        cached = self._mask_cache.get(sig)
        if cached is not None:
            return cached
        allowed: List[int] = []
        for tid, tb in enumerate(self.token_bytes):
            if not tb:
                continue
            if machine.clone().accepts(tb):
                allowed.append(tid)
        if machine.complete and self.eos_token_id is not None:
            allowed.append(self.eos_token_id)
        self._mask_cache[sig] = allowed
        return allowed

    def advance(self, machine: JsonMachine, token_id: int) -> bool:
        return machine.accepts(self.token_bytes[token_id])


class ChoiceMachine:
    """Constrain output to one of a fixed set of byte strings (vLLM
    guided_choice). State = the set of candidate indices still live plus
    the byte position; legal bytes are the candidates' next bytes."""

    __slots__ = ("choices", "live", "pos", "done")

    def __init__(self, choices: Sequence[bytes]):
        self.choices = [bytes(c) for c in choices]
        self.live = list(range(len(self.choices)))
        self.pos = 0
        self.done = False

    def clone(self) -> "ChoiceMachine":
        m = ChoiceMachine.__new__(ChoiceMachine)
        m.choices = self.choices
        m.live = list(self.live)
        m.pos = self.pos
        m.done = self.done
        return m

    def signature(self) -> Tuple:
        return ("choice", tuple(self.live), self.pos, self.done)

    @property
    def complete(self) -> bool:
        return self.done

    def advance(self, b: int) -> bool:
        if self.done:
            return False
        nxt = [
            i for i in self.live
            if self.pos < len(self.choices[i]) and self.choices[i][self.pos] == b
        ]
        if not nxt:
            return False
        self.live = nxt
        self.pos += 1
        if any(len(self.choices[i]) == self.pos for i in self.live):
            # a candidate is fully matched; prefer finishing exactly when
            # no longer candidate shares the prefix
            if all(len(self.choices[i]) == self.pos for i in self.live):
                self.done = True
        return True

    def accepts(self, data: bytes) -> bool:
        for b in data:
            if not self.advance(b):
                return False
        return True


# ---------------------------------------------------------------------------
# JSON-Schema-constrained machine (OpenAI response_format json_schema).
#
# Strict structured outputs over the practical schema subset: objects with
# typed properties (emitted in schema order, compact serialization — no
# interstitial whitespace, matching outlines/OpenAI strict mode), strings,
# enums (any scalar constants), integers/numbers, booleans, null, arrays
# with minItems/maxItems, nested combinations, and nullable via
# type: [T, "null"] (byte-time branch). Same machine protocol as
# JsonMachine (clone/signature/advance/complete), so GuidedJsonProcessor's
# signature-memoized masks apply unchanged.
# ---------------------------------------------------------------------------

import json as _json

_CONSUME = 0
_POP_CONSUME = 1
_POP_REDISPATCH = 2
_ILLEGAL = 3


class _FixedFrame:
    __slots__ = ("data", "pos")

    def __init__(self, data: bytes, pos: int = 0):
        self.data = data
        self.pos = pos

    def clone(self):
        return _FixedFrame(self.data, self.pos)

    def sig(self):
        return ("fx", self.data, self.pos)

    def feed(self, b: int):
        if self.data[self.pos] != b:
            return _ILLEGAL
        self.pos += 1
        return _POP_CONSUME if self.pos == len(self.data) else _CONSUME


class _StringFrame:
    """'"' body '"' with escape handling; any UTF-8 content."""

    __slots__ = ("state",)  # 0=open quote, 1=body, 2=escape

    def __init__(self, state: int = 0):
        self.state = state

    def clone(self):
        return _StringFrame(self.state)

    def sig(self):
        return ("str", self.state)

    def feed(self, b: int):
        if self.state == 0:
            if b == 0x22:
                self.state = 1
                return _CONSUME
            return _ILLEGAL
        if self.state == 1:
            if b == 0x22:
                return _POP_CONSUME
            if b == 0x5C:
                self.state = 2
                return _CONSUME
            return _CONSUME if b >= 0x20 else _ILLEGAL
        if b in STRING_ESCAPES:
            self.state = 1
            return _CONSUME
        return _ILLEGAL


class _NumberFrame:
    __slots__ = ("ns", "integer", "started")

    def __init__(self, integer: bool, ns: int = -1):
        self.ns = ns  # -1 = nothing consumed yet
        self.integer = integer
        self.started = ns >= 0

    def clone(self):
        f = _NumberFrame(self.integer, self.ns)
        f.started = self.started
        return f

    def sig(self):
        return ("num", self.ns, self.integer)

    @property
    def terminal(self) -> bool:
        return self.ns in _NUM_TERMINAL

    def feed(self, b: int):
        ns = self.ns
        if ns == -1:
            if b == 0x2D:
                self.ns = N_SIGN
                return _CONSUME
            if b in DIGITS:
                self.ns = N_ZERO if b == 0x30 else N_INT
                return _CONSUME
            return _ILLEGAL
        if b in DIGITS:
            if ns == N_ZERO:
                return _ILLEGAL
            self.ns = {
                N_SIGN: N_ZERO if b == 0x30 else N_INT,
                N_INT: N_INT,
                N_FRAC_START: N_FRAC,
                N_FRAC: N_FRAC,
                N_EXP_START: N_EXP,
                N_EXP_SIGN: N_EXP,
                N_EXP: N_EXP,
            }[ns]
            return _CONSUME
        if not self.integer:
            if b == 0x2E and ns in (N_ZERO, N_INT):
                self.ns = N_FRAC_START
                return _CONSUME
            if b in (0x65, 0x45) and ns in (N_ZERO, N_INT, N_FRAC):
                self.ns = N_EXP_START
                return _CONSUME
            if b in (0x2B, 0x2D) and ns == N_EXP_START:
                self.ns = N_EXP_SIGN
                return _CONSUME
        if self.terminal:
            return _POP_REDISPATCH  # delimiter belongs to the parent
        return _ILLEGAL


class _EnumFrame:
    """One of a fixed set of serialized JSON constants."""

    __slots__ = ("choices", "live", "pos")

    def __init__(self, choices, live=None, pos: int = 0):
        self.choices = choices
        self.live = live if live is not None else list(range(len(choices)))
        self.pos = pos

    def clone(self):
        return _EnumFrame(self.choices, list(self.live), self.pos)

    def sig(self):
        return ("enum", id(self.choices), tuple(self.live), self.pos)

    def feed(self, b: int):
        nxt = [
            i for i in self.live
            if self.pos < len(self.choices[i]) and self.choices[i][self.pos] == b
        ]
        if not nxt:
            return _ILLEGAL
        self.live = nxt
        self.pos += 1
        if all(len(self.choices[i]) == self.pos for i in self.live):
            return _POP_CONSUME
        return _CONSUME


class _ArrayFrame:
    """'[' item (',' item)* ']' with [minItems, maxItems] bounds.
    Pushes item frames; between items decides ',' vs ']'."""

    __slots__ = ("item_schema", "count", "min_items", "max_items", "state")
    # state: 0=expect '[', 1=first item or ']', 2=',' or ']'

    def __init__(self, item_schema, min_items: int, max_items, count=0,
                 state=0):
        self.item_schema = item_schema
        self.min_items = min_items
        self.max_items = max_items
        self.count = count
        self.state = state

    def clone(self):
        return _ArrayFrame(self.item_schema, self.min_items, self.max_items,
                           self.count, self.state)

    def sig(self):
        return ("arr", id(self.item_schema), self.count, self.state)

    def feed(self, b: int):
        if self.state == 0:
            if b != 0x5B:
                return _ILLEGAL
            self.state = 1
            return _CONSUME
        if self.state == 1:
            if b == 0x5D and self.min_items <= 0:
                return _POP_CONSUME
            if self.max_items is not None and self.count >= self.max_items:
                return _ILLEGAL
            # start of the first item: push its frames, redispatch b
            self.count += 1
            self.state = 2
            return ("push", _frames_for(self.item_schema))
        # state 2: after an item
        if b == 0x2C:  # ,
            if self.max_items is not None and self.count >= self.max_items:
                return _ILLEGAL
            self.count += 1
            return ("push_consume", _frames_for(self.item_schema))
        if b == 0x5D and self.count >= self.min_items:
            return _POP_CONSUME
        return _ILLEGAL


class _BranchFrame:
    """Byte-time union (nullable / anyOf of scalars): live sub-machines
    advance in lockstep; legal while any survives."""

    __slots__ = ("subs",)

    def __init__(self, subs):
        self.subs = subs  # list of SchemaMachine

    def clone(self):
        return _BranchFrame([m.clone() for m in self.subs])

    def sig(self):
        return ("br", tuple(m.signature() for m in self.subs))

    def feed(self, b: int):
        survivors = []
        for m in self.subs:
            mm = m.clone()
            if mm.advance(b):
                survivors.append(mm)
        if not survivors:
            # a completed sub-value means the byte belongs to the parent
            if any(m.complete for m in self.subs):
                return _POP_REDISPATCH
            return _ILLEGAL
        self.subs = survivors
        if all(m.complete and not m._can_continue() for m in survivors):
            return _POP_CONSUME
        return _CONSUME


class SchemaUnsupported(ValueError):
    pass


def _frames_for(schema: dict):
    """Compile a schema node into a frame list (top of stack first)."""
    if "enum" in schema:
        choices = [
            _json.dumps(v, separators=(",", ":")).encode() for v in schema["enum"]
        ]
        return [_EnumFrame(choices)]
    if "const" in schema:
        return [_FixedFrame(
            _json.dumps(schema["const"], separators=(",", ":")).encode()
        )]
    t = schema.get("type")
    if isinstance(t, list):
        subs = [
            SchemaMachine({**schema, "type": tt}) for tt in t
        ]
        return [_BranchFrame(subs)]
    if t == "string":
        return [_StringFrame()]
    if t == "integer":
        return [_NumberFrame(integer=True)]
    if t == "number":
        return [_NumberFrame(integer=False)]
    if t == "boolean":
        return [_EnumFrame([b"true", b"false"])]
    if t == "null":
        return [_FixedFrame(b"null")]
    if t == "array":
        return [
            _ArrayFrame(
                schema.get("items", {"type": "string"}),
                schema.get("minItems", 0),
                schema.get("maxItems"),
            )
        ]
    if t == "object" or "properties" in schema:
        props = schema.get("properties", {})
        if not props:
            raise SchemaUnsupported(
                "object schemas need properties (free-form objects: use "
                "response_format json_object)"
            )
        # strict serialization: every property, schema order, compact
        frames = []
        keys = list(props.keys())
        prefix = b"{"
        for i, k in enumerate(keys):
            key_bytes = _json.dumps(k).encode() + b":"
            frames.append(_FixedFrame(prefix + key_bytes))
            frames.extend(_frames_for(props[k]))
            prefix = b","
        frames.append(_FixedFrame(b"}"))
        return frames
    raise SchemaUnsupported(f"unsupported schema node: {schema!r}")


class SchemaMachine:
    """Byte-level acceptor for one JSON document matching ``schema``."""

    __slots__ = ("frames", "_schema")

    def __init__(self, schema: dict, _frames=None):
        self._schema = schema
        self.frames = (
            _frames if _frames is not None else list(reversed(_frames_for(schema)))
        )  # stack: top = last element

    def clone(self) -> "SchemaMachine":
        return SchemaMachine(
            self._schema, _frames=[f.clone() for f in self.frames]
        )

    def signature(self) -> Tuple:
        return ("schema", tuple(f.sig() for f in self.frames[-3:]),
                len(self.frames))

    def _can_continue(self) -> bool:
        """True when a trailing number frame could still accept digits."""
        return bool(self.frames)

    @property
    def complete(self) -> bool:
        if not self.frames:
            return True
        # a single terminal number frame at top level completes via EOS
        return (
            len(self.frames) == 1
            and isinstance(self.frames[-1], _NumberFrame)
            and self.frames[-1].terminal
        )

    def advance(self, b: int) -> bool:
        while True:
            if not self.frames:
                return False  # document finished; nothing may follow
            top = self.frames[-1]
            r = top.feed(b)
            if r == _CONSUME:
                return True
            if r == _POP_CONSUME:
                self.frames.pop()
                return True
            if r == _POP_REDISPATCH:
                self.frames.pop()
                continue
            if isinstance(r, tuple):
                action, new_frames = r
                self.frames.extend(reversed(new_frames))
                if action == "push_consume":
                    return True
                continue  # "push": redispatch b to the new top
            return False

    def accepts(self, data: bytes) -> bool:
        for b in data:
            if not self.advance(b):
                return False
        return True
