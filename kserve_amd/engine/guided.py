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
