"""Byte-level JSON grammar automaton for constrained decoding.

With the byte tokenizer (one token per UTF-8 byte, id = byte + 4) a JSON
pushdown automaton over BYTES gives exact grammar masking: at every decode
step the sampler only sees logits of bytes that keep the output valid
JSON, and a remaining-token budget filter only admits bytes from which the
value can still be CLOSED within the budget — so max_tokens can never
truncate mid-object.  (HF multi-byte tokenizers need token-level grammar
compilation — round-2.)

Reference seam: the upstream framework's structured output is
prompt+validate (agent_ai.py:335-341 response_format); this is the
stronger guarantee the in-process engine can make.
"""
from __future__ import annotations

BYTE_OFFSET = 4   # ByteTokenizer: byte b <-> token id b + 4
EOS_ID = 2

# states
VAL = 0            # a value must start here
OBJ_KEY_OR_END = 1  # after '{': '"' or '}'
OBJ_KEY = 2        # after ',' in object: '"' only
OBJ_COLON = 3      # after key string: ':'
STR = 4            # inside a string
STR_ESC = 5        # after backslash
NUM_INT_START = 6  # after '-'
NUM_INT = 7        # in integer digits (may end)
NUM_ZERO = 8       # leading 0 (only '.', 'e', or end may follow)
NUM_FRAC_START = 9  # after '.'
NUM_FRAC = 10      # in fraction digits (may end)
NUM_EXP_START = 11  # after 'e'/'E': sign or digit
NUM_EXP_SIGN = 12  # after exponent sign: digit
NUM_EXP = 13       # in exponent digits (may end)
AFTER_VAL = 14     # value complete; ',' / closer / end
LIT = 15           # inside true/false/null (remaining in self.lit)
STR_U = 16         # \uXXXX, remaining hex count in self.hex_left
ARR_VAL_OR_END = 17  # just after '[': value or immediate ']'

_WS = b" \t\n\r"
_DIGITS = b"0123456789"
_VALUE_START = b"{[\"-0123456789tfn"
_ESCAPABLE = b'"\\/bfnrtu'
_HEX = b"0123456789abcdefABCDEF"
_NUM_END_STATES = (NUM_INT, NUM_ZERO, NUM_FRAC, NUM_EXP)


# UTF-8 lead-byte table: lead -> (n_continuations, first-cont lo, hi);
# continuations after the first are always 0x80-0xBF.  The restricted
# first-continuation ranges reject overlong encodings and surrogates, so
# masked output is always decodable UTF-8 (what json.loads requires).
def _utf8_lead(b: int):
    if 0xC2 <= b <= 0xDF:
        return (1, 0x80, 0xBF)
    if b == 0xE0:
        return (2, 0xA0, 0xBF)
    if b == 0xED:
        return (2, 0x80, 0x9F)
    if 0xE1 <= b <= 0xEF:
        return (2, 0x80, 0xBF)
    if b == 0xF0:
        return (3, 0x90, 0xBF)
    if b == 0xF4:
        return (3, 0x80, 0x8F)
    if 0xF1 <= b <= 0xF3:
        return (3, 0x80, 0xBF)
    return None  # ASCII handled elsewhere; C0/C1/F5+ invalid


_UTF8_LEADS = bytes(b for b in range(0xC2, 0xF5) if _utf8_lead(b))
_STR_ASCII = bytes(b for b in range(0x20, 0x80) if b not in b'"\\')


class JsonFSM:
    __slots__ = ("state", "stack", "key_str", "lit", "hex_left",
                 "utf_left", "utf_lo", "utf_hi")

    def __init__(self):
        self.state = VAL
        self.stack = []       # b'{' / b'[' context markers (as ints)
        self.key_str = False  # current STR is an object key
        self.lit = b""
        self.hex_left = 0
        self.utf_left = 0     # pending UTF-8 continuation bytes
        self.utf_lo = 0x80
        self.utf_hi = 0xBF

    def clone(self) -> "JsonFSM":
        f = JsonFSM.__new__(JsonFSM)
        f.state = self.state
        f.stack = list(self.stack)
        f.key_str = self.key_str
        f.lit = self.lit
        f.hex_left = self.hex_left
        f.utf_left = self.utf_left
        f.utf_lo = self.utf_lo
        f.utf_hi = self.utf_hi
        return f

    # ---------------------------------------------------------- grammar
    def _allowed_raw(self) -> bytes:
        s = self.state
        if s == VAL:
            return _VALUE_START + _WS
        if s == ARR_VAL_OR_END:
            return _VALUE_START + b"]" + _WS
        if s == OBJ_KEY_OR_END:
            return b'"}' + _WS
        if s == OBJ_KEY:
            return b'"' + _WS
        if s == OBJ_COLON:
            return b":" + _WS
        if s == STR:
            if self.utf_left:
                return bytes(range(self.utf_lo, self.utf_hi + 1))
            return b'"\\' + _STR_ASCII + _UTF8_LEADS
        if s == STR_ESC:
            return _ESCAPABLE
        if s == STR_U:
            return _HEX
        if s == NUM_INT_START:
            return _DIGITS
        if s == NUM_ZERO:
            return b".eE" + self._terminators()
        if s == NUM_INT:
            return _DIGITS + b".eE" + self._terminators()
        if s == NUM_FRAC_START:
            return _DIGITS
        if s == NUM_FRAC:
            return _DIGITS + b"eE" + self._terminators()
        if s == NUM_EXP_START:
            return b"+-" + _DIGITS
        if s == NUM_EXP_SIGN:
            return _DIGITS
        if s == NUM_EXP:
            return _DIGITS + self._terminators()
        if s == LIT:
            return self.lit[:1]
        if s == AFTER_VAL:
            return self._terminators()
        raise AssertionError(f"bad state {s}")

    def _terminators(self) -> bytes:
        """Bytes legal where a value just ended (incl. trailing ws)."""
        if not self.stack:
            return bytes(_WS)
        top = self.stack[-1]
        return (b",}" if top == ord("{") else b",]") + _WS

    def advance(self, b: int) -> None:
        """Consume one byte (must be in allowed set)."""
        s = self.state
        c = bytes([b])
        if s in _NUM_END_STATES and c not in self._num_continue(s):
            self.state = AFTER_VAL  # number ended; reprocess terminator
            return self.advance(b)
        if s == ARR_VAL_OR_END:
            if c in _WS:
                return
            if b == ord("]"):
                self.stack.pop()          # empty array
                self.state = AFTER_VAL
                return
            self.state = VAL              # a value starts; reprocess byte
            return self.advance(b)
        if s == VAL:
            if c in _WS:
                return
            if b == ord("{"):
                self.stack.append(b)
                self.state = OBJ_KEY_OR_END
            elif b == ord("["):
                self.stack.append(b)
                self.state = ARR_VAL_OR_END
            elif b == ord('"'):
                self.key_str = False
                self.state = STR
            elif b == ord("-"):
                self.state = NUM_INT_START
            elif b == ord("0"):
                self.state = NUM_ZERO
            elif c in _DIGITS:
                self.state = NUM_INT
            elif b == ord("t"):
                self.lit, self.state = b"rue", LIT
            elif b == ord("f"):
                self.lit, self.state = b"alse", LIT
            elif b == ord("n"):
                self.lit, self.state = b"ull", LIT
            else:
                raise ValueError(f"byte {c!r} invalid in VAL")
            return
        if s == OBJ_KEY_OR_END:
            if c in _WS:
                return
            if b == ord('"'):
                self.key_str = True
                self.state = STR
            elif b == ord("}"):
                self.stack.pop()
                self.state = AFTER_VAL
            else:
                raise ValueError(f"byte {c!r} invalid after '{{'")
            return
        if s == OBJ_KEY:
            if c in _WS:
                return
            if b == ord('"'):
                self.key_str = True
                self.state = STR
            else:
                raise ValueError("expected object key")
            return
        if s == OBJ_COLON:
            if c in _WS:
                return
            if b == ord(":"):
                self.state = VAL
            else:
                raise ValueError("expected ':'")
            return
        # NOTE: advance() must be STRICT — the token-level grammar walk
        # (token_grammar.py) discovers legality by calling advance() on a
        # clone and catching ValueError, so any byte advance() silently
        # accepts becomes a byte multi-byte tokens may smuggle into the
        # output (found on GPU: raw control bytes inside strings).
        if s == STR:
            if self.utf_left:
                if not (self.utf_lo <= b <= self.utf_hi):
                    raise ValueError("invalid UTF-8 continuation")
                self.utf_left -= 1
                self.utf_lo, self.utf_hi = 0x80, 0xBF
                return
            if b == ord('"'):
                self.state = OBJ_COLON if self.key_str else AFTER_VAL
                self.key_str = False
            elif b == ord("\\"):
                self.state = STR_ESC
            elif b < 0x20:
                raise ValueError("raw control byte in string")
            elif b >= 0x80:
                lead = _utf8_lead(b)
                if lead is None:
                    raise ValueError("invalid UTF-8 lead byte")
                self.utf_left, self.utf_lo, self.utf_hi = lead
            return
        if s == STR_ESC:
            if b == ord("u"):
                self.hex_left, self.state = 4, STR_U
            elif c in _ESCAPABLE:
                self.state = STR
            else:
                raise ValueError(f"invalid escape {c!r}")
            return
        if s == STR_U:
            if c not in _HEX:
                raise ValueError(f"invalid \\u hex digit {c!r}")
            self.hex_left -= 1
            if self.hex_left == 0:
                self.state = STR
            return
        if s == NUM_INT_START:
            if c not in _DIGITS:
                raise ValueError("digit required after '-'")
            self.state = NUM_ZERO if b == ord("0") else NUM_INT
            return
        if s in (NUM_INT, NUM_ZERO):
            if b == ord("."):
                self.state = NUM_FRAC_START
            elif c in b"eE":
                self.state = NUM_EXP_START
            # NUM_INT digit: stay
            return
        if s == NUM_FRAC_START:
            if c not in _DIGITS:
                raise ValueError("digit required after '.'")
            self.state = NUM_FRAC
            return
        if s == NUM_FRAC:
            if c in b"eE":
                self.state = NUM_EXP_START
            return
        if s == NUM_EXP_START:
            if c not in b"+-" + _DIGITS:
                raise ValueError("sign or digit required after exponent")
            self.state = NUM_EXP_SIGN if c in b"+-" else NUM_EXP
            return
        if s == NUM_EXP_SIGN:
            if c not in _DIGITS:
                raise ValueError("digit required after exponent sign")
            self.state = NUM_EXP
            return
        if s == NUM_EXP:
            return  # digits stay
        if s == LIT:
            if c != self.lit[:1]:
                raise ValueError(f"literal expects {self.lit[:1]!r}")
            self.lit = self.lit[1:]
            if not self.lit:
                self.state = AFTER_VAL
            return
        if s == AFTER_VAL:
            if c in _WS:
                return
            if b == ord(","):
                if not self.stack:
                    raise ValueError("',' outside any container")
                top = self.stack[-1]
                self.state = OBJ_KEY if top == ord("{") else VAL
            elif b == ord("}") and self.stack and self.stack[-1] == ord("{"):
                self.stack.pop()
            elif b == ord("]") and self.stack and self.stack[-1] == ord("["):
                self.stack.pop()
            else:
                raise ValueError(f"byte {c!r} invalid after value")
            return
        raise AssertionError(f"bad state {s}")

    @staticmethod
    def _num_continue(s: int) -> bytes:
        if s == NUM_INT:
            return _DIGITS + b".eE"
        if s == NUM_ZERO:
            return b".eE"
        if s == NUM_FRAC:
            return _DIGITS + b"eE"
        return _DIGITS  # NUM_EXP

    # ------------------------------------------------------- completion
    def complete(self) -> bool:
        """A full top-level JSON value has been produced."""
        return not self.stack and (self.state == AFTER_VAL or
                                   self.state in _NUM_END_STATES)

    def min_close(self) -> int:
        """Fewest additional bytes to reach a complete value."""
        s = self.state
        closers = len(self.stack)
        if s == AFTER_VAL or s in _NUM_END_STATES:
            return closers
        if s == VAL:
            return 1 + closers
        if s in (OBJ_KEY_OR_END, ARR_VAL_OR_END):
            return closers  # the closer is counted in `closers`
        if s == OBJ_KEY:
            return 4 + closers          # "":0  -> '"','"',':','0'
        if s == OBJ_COLON:
            return 2 + closers          # ':','0'
        if s == STR:
            extra = 3 if self.key_str else 0  # '"' then ':','0'
            return self.utf_left + 1 + extra + closers
        if s == STR_ESC:
            extra = 3 if self.key_str else 0
            return 2 + extra + closers
        if s == STR_U:
            extra = 3 if self.key_str else 0
            return self.hex_left + 1 + extra + closers
        if s in (NUM_INT_START, NUM_FRAC_START, NUM_EXP_START, NUM_EXP_SIGN):
            return 1 + closers
        if s == LIT:
            return len(self.lit) + closers
        raise AssertionError(f"bad state {s}")

    # -------------------------------------------------------- masking
    def allowed_token_ids(self, remaining: int) -> list[int]:
        """Token ids legal for the NEXT generated token, keeping the value
        completable within `remaining` tokens.  Includes EOS exactly when
        the value is already complete."""
        out = [EOS_ID] if self.complete() else []
        for b in set(self._allowed_raw()):
            nxt = self.clone()
            try:
                nxt.advance(b)
            except (ValueError, AssertionError):
                continue
            if nxt.min_close() <= remaining - 1:
                out.append(b + BYTE_OFFSET)
        return out
