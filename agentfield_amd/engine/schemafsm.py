"""Schema-constrained JSON decoding: a byte automaton that enforces a
JSON-Schema subset ON TOP of syntactic validity, so `ai(schema=...)`
GUARANTEES the output parses AND matches the schema (keys, types,
enums) — strictly stronger than the reference's prompt-and-validate
seam, and stronger than grammar-only json_mode.

Supported subset (anything else falls back to unconstrained JSON for
that subtree via an embedded JsonFSM):
  object  — properties (byte-trie over UNSEEN keys), required enforced,
            additionalProperties never offered; optional keys may be
            omitted
  array   — items schema; may be empty
  string / integer / number / boolean / null
  enum    — JSON literals (prefix-tracked; number enums end lazily)
  no type — any JSON value (embedded JsonFSM)

Same surface as JsonFSM (clone/advance/complete/min_close/
allowed_token_ids) so the byte-mask path and the token-trie walk
(token_grammar.py) drive it unchanged; `sig()` keys the token-mask
cache.  Lazy termination: values with no closing byte (numbers, enum
numbers, embedded any-values) stay on the stack until a byte they
cannot consume arrives, then pop and hand that byte to the parent.
"""
from __future__ import annotations

from .jsonfsm import (_DIGITS, _ESCAPABLE, _HEX, _UTF8_LEADS, _WS,
                      BYTE_OFFSET, EOS_ID, JsonFSM, _STR_ASCII, _utf8_lead)


def _lit_bytes(v) -> bytes:
    import json as _json
    return _json.dumps(v).encode()


class SchemaSpec:
    """Compiled schema node tree (shared, immutable)."""

    __slots__ = ("kind", "properties", "required", "items", "enum",
                 "prop_order", "_min_close", "sid", "source")
    _next_id = [0]

    def __init__(self, schema):
        SchemaSpec._next_id[0] += 1
        self.sid = SchemaSpec._next_id[0]
        self.source = schema  # original JSON (wire codec re-serializes it)
        self.properties: dict[str, "SchemaSpec"] = {}
        self.prop_order: list[str] = []
        self.required: frozenset = frozenset()
        self.items: SchemaSpec | None = None
        self.enum: tuple[bytes, ...] | None = None
        self._min_close = None
        if not isinstance(schema, dict):
            self.kind = "any"
            return
        if isinstance(schema.get("enum"), list) and schema["enum"]:
            self.kind = "enum"
            self.enum = tuple(_lit_bytes(v) for v in schema["enum"])
            return
        t = schema.get("type")
        if t == "object" and isinstance(schema.get("properties"), dict) \
                and schema["properties"]:
            self.kind = "object"
            self.properties = {k: SchemaSpec(v)
                               for k, v in schema["properties"].items()}
            self.prop_order = list(self.properties)
            req = schema.get("required")
            self.required = frozenset(k for k in (req or [])
                                      if k in self.properties)
        elif t == "array":
            self.kind = "array"
            self.items = SchemaSpec(schema.get("items"))
        elif t in ("string", "integer", "number", "boolean", "null"):
            self.kind = t
        else:
            self.kind = "any"

    def min_close(self) -> int:
        """Fewest bytes a fresh value of this schema needs."""
        if self._min_close is not None:
            return self._min_close
        self._min_close = 64  # recursion guard (self-referential schemas)
        k = self.kind
        if k == "enum":
            out = min(len(b) for b in self.enum)
        elif k == "string":
            out = 2
        elif k in ("integer", "number", "any"):
            out = 1
        elif k in ("boolean", "null"):
            out = 4
        elif k == "array":
            out = 2
        else:  # object
            out = 2
            for name in self.required:
                # "name": <value> (+ ',' separators, minus the last)
                out += len(name) + 3 + self.properties[name].min_close() + 1
            if self.required:
                out -= 1
        self._min_close = out
        return out


# ---------------------------------------------------------------- frames
class _F:
    VALUE = 0   # expect a value of .spec
    OBJ = 1
    ARR = 2
    STR = 3
    NUM = 4
    LIT = 5     # fixed remaining byte tail (true/false/null)
    ANY = 6     # embedded JsonFSM
    ENUM = 7    # literal set, .tail = matched prefix


# object sub-states
_O_FIRST, _O_KEY_OPEN, _O_KEY, _O_COLON, _O_AFTER = range(5)
# array sub-states
_A_FIRST, _A_AFTER, _A_VALUE = range(3)
# number sub-states
(_N_START, _N_INT, _N_ZERO, _N_FRACS, _N_FRAC, _N_EXPS, _N_EXPSIGN,
 _N_EXP) = range(8)
_N_END = (_N_INT, _N_ZERO, _N_FRAC, _N_EXP)
# string sub-states
_S_BODY, _S_ESC, _S_HEX = range(3)


class _Frame:
    __slots__ = ("kind", "spec", "sub", "seen", "tail", "fsm", "esc")

    def __init__(self, kind, spec=None, sub=0, seen=None, tail=b"",
                 fsm=None, esc=0):
        self.kind = kind
        self.spec = spec
        self.sub = sub
        self.seen = seen
        self.tail = tail
        self.fsm = fsm
        self.esc = esc

    def clone(self):
        return _Frame(self.kind, self.spec, self.sub,
                      set(self.seen) if self.seen is not None else None,
                      self.tail, self.fsm.clone() if self.fsm else None,
                      self.esc)


class SchemaFSM:
    __slots__ = ("spec", "stack", "done")

    def __init__(self, spec: SchemaSpec):
        self.spec = spec
        self.stack: list[_Frame] = [_Frame(_F.VALUE, spec)]
        self.done = False

    def clone(self) -> "SchemaFSM":
        f = SchemaFSM.__new__(SchemaFSM)
        f.spec = self.spec
        f.stack = [fr.clone() for fr in self.stack]
        f.done = self.done
        return f

    def sig(self) -> tuple:
        out: list = [self.spec.sid, self.done]
        for fr in self.stack:
            out.append((fr.kind, fr.spec.sid if fr.spec else 0, fr.sub,
                        frozenset(fr.seen) if fr.seen is not None else None,
                        fr.tail,
                        (fr.fsm.state, tuple(fr.fsm.stack), fr.fsm.key_str,
                         fr.fsm.lit, fr.fsm.hex_left, fr.fsm.utf_left,
                         fr.fsm.utf_lo, fr.fsm.utf_hi) if fr.fsm else None,
                        fr.esc))
        return tuple(out)

    # ------------------------------------------------------------ helpers
    @staticmethod
    def _candidates(fr: _Frame) -> list[bytes]:
        return [k.encode() for k in fr.spec.prop_order if k not in fr.seen]

    def _frame_closable(self, fr: _Frame) -> bool:
        """This frame's value is complete as-is (lazy-end frames)."""
        if fr.kind == _F.NUM:
            return fr.sub in _N_END
        if fr.kind == _F.ANY:
            return fr.fsm.complete()
        if fr.kind == _F.ENUM:
            return fr.tail in fr.spec.enum
        return False

    def complete(self) -> bool:
        if self.done:
            return True
        return len(self.stack) == 1 and self._frame_closable(self.stack[0])

    # ------------------------------------------------------------ advance
    def advance(self, b: int) -> None:
        c = bytes([b])
        if self.done:
            if c in _WS:
                return
            raise ValueError("value already complete")
        fr = self.stack[-1]
        k = fr.kind

        # lazy-end frames: hand bytes they cannot consume to the parent
        if k == _F.NUM:
            cont = self._num_continue(fr)
            if c not in cont:
                if fr.sub in _N_END:
                    self._pop_value()
                    return self.advance(b)
                raise ValueError(f"{c!r} invalid in number")
            self._num_advance(fr, b, c)
            return
        if k == _F.ANY:
            try:
                fr.fsm.advance(b)
            except (ValueError, AssertionError):
                if fr.fsm.complete():
                    self._pop_value()
                    return self.advance(b)
                raise
            return
        if k == _F.ENUM:
            nt = fr.tail + c
            exts = [e for e in fr.spec.enum if e.startswith(nt)]
            if not exts:
                if fr.tail in fr.spec.enum:  # lazy end (number enums)
                    self._pop_value()
                    return self.advance(b)
                raise ValueError(f"no enum literal starts with {nt!r}")
            fr.tail = nt
            # eager pop when exactly matched and nothing extends it
            if nt in fr.spec.enum and len(exts) == 1 and exts[0] == nt:
                self._pop_value()
            return

        if k == _F.VALUE:
            if c in _WS:
                return
            self._start_value(fr, b, c)
            return
        if k == _F.LIT:
            if c != fr.tail[:1]:
                raise ValueError(f"expected {fr.tail[:1]!r}")
            fr.tail = fr.tail[1:]
            if not fr.tail:
                self._pop_value()
            return
        if k == _F.STR:
            if fr.sub == _S_ESC:
                if b == ord("u"):
                    fr.sub, fr.esc = _S_HEX, 4
                elif c in _ESCAPABLE:
                    fr.sub = _S_BODY
                else:
                    raise ValueError(f"invalid escape {c!r}")
            elif fr.sub == _S_HEX:
                if c not in _HEX:
                    raise ValueError("invalid \\u hex digit")
                fr.esc -= 1
                if fr.esc == 0:
                    fr.sub = _S_BODY
            elif fr.tail:  # pending UTF-8 continuations (left, lo, hi)
                left, lo, hi = fr.tail
                if not (lo <= b <= hi):
                    raise ValueError("invalid UTF-8 continuation")
                fr.tail = bytes([left - 1, 0x80, 0xBF]) if left > 1 else b""
            elif b == ord('"'):
                self._pop_value()
            elif b == ord("\\"):
                fr.sub = _S_ESC
            elif b < 0x20:
                raise ValueError("raw control byte in string")
            elif b >= 0x80:
                lead = _utf8_lead(b)
                if lead is None:
                    raise ValueError("invalid UTF-8 lead byte")
                fr.tail = bytes(lead)
            return
        if k == _F.OBJ:
            self._obj_advance(fr, b, c)
            return
        if k == _F.ARR:
            self._arr_advance(fr, b, c)
            return
        raise AssertionError(k)

    def _start_value(self, fr: _Frame, b: int, c: bytes) -> None:
        spec = fr.spec
        k = spec.kind
        if k == "any":
            fsm = JsonFSM()
            fsm.advance(b)
            fr.kind, fr.fsm = _F.ANY, fsm
            return
        if k == "enum":
            if not any(e[:1] == c for e in spec.enum):
                raise ValueError(f"{c!r} starts no enum literal")
            fr.kind, fr.tail = _F.ENUM, c
            exts = [e for e in spec.enum if e.startswith(c)]
            if c in spec.enum and len(exts) == 1:
                self._pop_value()
            return
        if k == "object":
            if b != ord("{"):
                raise ValueError("'{' required by schema")
            fr.kind, fr.sub, fr.seen = _F.OBJ, _O_FIRST, set()
            return
        if k == "array":
            if b != ord("["):
                raise ValueError("'[' required by schema")
            fr.kind, fr.sub = _F.ARR, _A_FIRST
            return
        if k == "string":
            if b != ord('"'):
                raise ValueError("string required by schema")
            fr.kind, fr.sub = _F.STR, _S_BODY
            return
        if k in ("integer", "number"):
            if b == ord("-"):
                fr.kind, fr.sub = _F.NUM, _N_START
            elif b == ord("0"):
                fr.kind, fr.sub = _F.NUM, _N_ZERO
            elif c in _DIGITS:
                fr.kind, fr.sub = _F.NUM, _N_INT
            else:
                raise ValueError("number required by schema")
            return
        if k == "boolean":
            if b == ord("t"):
                fr.kind, fr.tail = _F.LIT, b"rue"
            elif b == ord("f"):
                fr.kind, fr.tail = _F.LIT, b"alse"
            else:
                raise ValueError("boolean required by schema")
            return
        if k == "null":
            if b != ord("n"):
                raise ValueError("null required by schema")
            fr.kind, fr.tail = _F.LIT, b"ull"
            return
        raise AssertionError(k)

    def _num_continue(self, fr: _Frame) -> bytes:
        s = fr.sub
        frac = fr.spec.kind == "number"  # integers: digits only
        if s == _N_START:
            return _DIGITS
        if s == _N_INT:
            return _DIGITS + (b".eE" if frac else b"")
        if s == _N_ZERO:
            return b".eE" if frac else b""
        if s == _N_FRACS:
            return _DIGITS
        if s == _N_FRAC:
            return _DIGITS + b"eE"
        if s == _N_EXPS:
            return b"+-" + _DIGITS
        if s == _N_EXPSIGN:
            return _DIGITS
        return _DIGITS  # _N_EXP

    @staticmethod
    def _num_advance(fr: _Frame, b: int, c: bytes) -> None:
        s = fr.sub
        if s == _N_START:
            fr.sub = _N_ZERO if b == ord("0") else _N_INT
        elif s in (_N_INT, _N_ZERO):
            if b == ord("."):
                fr.sub = _N_FRACS
            elif c in b"eE":
                fr.sub = _N_EXPS
        elif s == _N_FRACS:
            fr.sub = _N_FRAC
        elif s == _N_FRAC:
            if c in b"eE":
                fr.sub = _N_EXPS
        elif s == _N_EXPS:
            fr.sub = _N_EXPSIGN if c in b"+-" else _N_EXP
        elif s == _N_EXPSIGN:
            fr.sub = _N_EXP

    def _obj_advance(self, fr: _Frame, b: int, c: bytes) -> None:
        s = fr.sub
        if s != _O_KEY and c in _WS:
            return
        if s in (_O_FIRST, _O_KEY_OPEN):
            if b == ord('"') and self._candidates(fr):
                fr.sub, fr.tail = _O_KEY, b""
            elif b == ord("}") and s == _O_FIRST and not fr.spec.required:
                self._pop_value()
            else:
                raise ValueError("expected a schema key"
                                 if s == _O_KEY_OPEN or fr.spec.required
                                 else "expected a schema key or '}'")
            return
        if s == _O_KEY:
            if b == ord('"'):
                name = fr.tail.decode("utf-8", errors="strict")
                if name not in fr.spec.properties or name in fr.seen:
                    raise ValueError(f"key {name!r} not allowed here")
                fr.seen.add(name)
                fr.sub = _O_COLON
                return
            nt = fr.tail + c
            if not any(k.startswith(nt) for k in self._candidates(fr)):
                raise ValueError(f"no schema key starts with {nt!r}")
            fr.tail = nt
            return
        if s == _O_COLON:
            if b != ord(":"):
                raise ValueError("':' expected")
            fr.sub = _O_AFTER  # state seen again when the value pops
            prop = fr.spec.properties[fr.tail.decode()]
            self.stack.append(_Frame(_F.VALUE, prop))
            return
        if s == _O_AFTER:
            if b == ord(",") and self._candidates(fr):
                fr.sub = _O_KEY_OPEN
            elif b == ord("}") and fr.spec.required <= fr.seen:
                self._pop_value()
            else:
                raise ValueError("',' (unseen keys) or '}' expected")
            return
        raise AssertionError(s)

    def _arr_advance(self, fr: _Frame, b: int, c: bytes) -> None:
        s = fr.sub
        if c in _WS:
            return
        if s in (_A_FIRST, _A_VALUE):
            if b == ord("]") and s == _A_FIRST:
                self._pop_value()
                return
            fr.sub = _A_AFTER
            self.stack.append(_Frame(_F.VALUE, fr.spec.items))
            self.advance(b)  # reprocess inside the new VALUE frame
            return
        if s == _A_AFTER:
            if b == ord(","):
                fr.sub = _A_VALUE
            elif b == ord("]"):
                self._pop_value()
            else:
                raise ValueError("',' or ']' expected")
            return
        raise AssertionError(s)

    def _pop_value(self) -> None:
        self.stack.pop()
        if not self.stack:
            self.done = True

    # ------------------------------------------------------------ allowed
    def _parent_allowed(self) -> bytes:
        """Bytes the PARENT accepts once the (closable) top frame pops."""
        nxt = self.clone()
        nxt._pop_value()
        if nxt.done:
            return bytes(_WS)
        return nxt._allowed_raw()

    def _allowed_raw(self) -> bytes:
        if self.done:
            return bytes(_WS)
        fr = self.stack[-1]
        k = fr.kind
        if k == _F.VALUE:
            return self._value_starts(fr.spec) + _WS
        if k == _F.ANY:
            base = fr.fsm._allowed_raw()
            if fr.fsm.complete():
                base += self._parent_allowed()
            return base
        if k == _F.NUM:
            base = self._num_continue(fr)
            if fr.sub in _N_END:
                base += self._parent_allowed()
            return base
        if k == _F.ENUM:
            nxt = {e[len(fr.tail)] for e in fr.spec.enum
                   if e.startswith(fr.tail) and len(e) > len(fr.tail)}
            base = bytes(sorted(nxt))
            if fr.tail in fr.spec.enum:
                base += self._parent_allowed()
            return base
        if k == _F.LIT:
            return fr.tail[:1]
        if k == _F.STR:
            if fr.sub == _S_ESC:
                return _ESCAPABLE
            if fr.sub == _S_HEX:
                return _HEX
            if fr.tail:
                return bytes(range(fr.tail[1], fr.tail[2] + 1))
            return b'"\\' + _STR_ASCII + _UTF8_LEADS
        if k == _F.OBJ:
            if fr.sub == _O_FIRST:
                close = b"}" if not fr.spec.required else b""
                key = b'"' if self._candidates(fr) else b""
                return key + close + _WS
            if fr.sub == _O_KEY_OPEN:
                return b'"' + _WS
            if fr.sub == _O_KEY:
                cands = [key for key in self._candidates(fr)
                         if key.startswith(fr.tail)]
                nxt = {key[len(fr.tail)] for key in cands
                       if len(key) > len(fr.tail)}
                end = b'"' if fr.tail in cands else b""
                return bytes(sorted(nxt)) + end
            if fr.sub == _O_COLON:
                return b":" + _WS
            more = b"," if self._candidates(fr) else b""
            close = b"}" if fr.spec.required <= fr.seen else b""
            return more + close + _WS
        if k == _F.ARR:
            if fr.sub == _A_FIRST:
                return self._value_starts(fr.spec.items) + b"]" + _WS
            if fr.sub == _A_VALUE:
                return self._value_starts(fr.spec.items) + _WS
            return b",]" + _WS
        raise AssertionError(k)

    @staticmethod
    def _value_starts(spec: SchemaSpec) -> bytes:
        k = spec.kind
        if k == "any":
            return b"{[\"-0123456789tfn"
        if k == "enum":
            return bytes(sorted({e[0] for e in spec.enum}))
        if k == "object":
            return b"{"
        if k == "array":
            return b"["
        if k == "string":
            return b'"'
        if k in ("integer", "number"):
            return b"-" + _DIGITS
        if k == "boolean":
            return b"tf"
        return b"n"

    # --------------------------------------------------------- completion
    def min_close(self) -> int:
        total = 0
        for fr in reversed(self.stack):
            k = fr.kind
            if k == _F.VALUE:
                total += fr.spec.min_close()
            elif k == _F.ANY:
                total += fr.fsm.min_close()
            elif k == _F.NUM:
                total += 0 if fr.sub in _N_END else 1
            elif k == _F.ENUM:
                if fr.tail in fr.spec.enum:
                    total += 0
                else:
                    total += min(len(e) - len(fr.tail)
                                 for e in fr.spec.enum
                                 if e.startswith(fr.tail))
            elif k == _F.LIT:
                total += len(fr.tail)
            elif k == _F.STR:
                total += 1 + (1 if fr.sub == _S_ESC else 0) + \
                    (fr.esc if fr.sub == _S_HEX else 0) + \
                    (fr.tail[0] if fr.sub == _S_BODY and fr.tail else 0)
            elif k == _F.ARR:
                total += 1 + (fr.spec.items.min_close()
                              if fr.sub == _A_VALUE else 0)
            elif k == _F.OBJ:
                # costs: a member appended after an existing one is
                # ',"name":<v>' = len+4+mc; the '}' closes the frame.
                props = fr.spec.properties

                def mcost(name: str) -> int:
                    return props[name].min_close()

                def members(names) -> int:
                    return sum(len(p) + 4 + mcost(p) for p in names)

                missing = [p for p in fr.spec.required if p not in fr.seen]
                if fr.sub in (_O_FIRST, _O_AFTER):
                    t = 1 + members(missing)
                    if missing and fr.sub == _O_FIRST:
                        t -= 1  # first member needs no comma
                elif fr.sub == _O_KEY_OPEN:
                    # a separator was consumed: SOME candidate key must
                    # be emitted now ('"key":<v>' = len+3+mc)
                    t = 1 + min(
                        len(key) + 3 + mcost(key.decode()) +
                        members([p for p in missing if p != key.decode()])
                        for key in self._candidates(fr))
                elif fr.sub == _O_KEY:
                    t = 1 + min(
                        (len(key) - len(fr.tail)) + 2 +
                        mcost(key.decode()) +
                        members([p for p in missing if p != key.decode()])
                        for key in self._candidates(fr)
                        if key.startswith(fr.tail))
                else:  # _O_COLON
                    name = fr.tail.decode()
                    t = 2 + mcost(name) + \
                        members([p for p in missing if p != name])
                total += t
            else:
                raise AssertionError(k)
        return total

    # --------------------------------------------------------- byte masks
    def allowed_token_ids(self, remaining: int) -> list[int]:
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


class MultiFSM:
    """Root-level `anyOf`: an NFA of SchemaFSM alternatives.  A byte is
    legal if ANY alive alternative accepts it; alternatives that reject
    die.  Same surface as SchemaFSM/JsonFSM so the byte-mask path and
    the token-trie walk drive it unchanged.  (Nested anyOf inside a
    schema is not supported — compile fails loudly via make_fsm.)"""

    __slots__ = ("alts",)

    def __init__(self, alts):
        self.alts = list(alts)
        if not self.alts:
            raise ValueError("anyOf needs at least one alternative")

    def clone(self) -> "MultiFSM":
        m = MultiFSM.__new__(MultiFSM)
        m.alts = [a.clone() for a in self.alts]
        return m

    def sig(self) -> tuple:
        return ("anyof",) + tuple(a.sig() for a in self.alts)

    def complete(self) -> bool:
        return any(a.complete() for a in self.alts)

    def min_close(self) -> int:
        return min(a.min_close() for a in self.alts)

    def advance(self, b: int) -> None:
        alive = []
        for a in self.alts:
            try:
                c = a.clone()
                c.advance(b)
                alive.append(c)
            except ValueError:
                pass
        if not alive:
            raise ValueError(f"byte {bytes([b])!r} rejected by every "
                             "anyOf alternative")
        self.alts = alive

    def _allowed_bytes(self) -> set:
        out: set = set()
        for a in self.alts:
            out.update(a._allowed_raw())  # candidates; advance() re-checks
        return out

    def allowed_token_ids(self, remaining: int) -> list[int]:
        from .jsonfsm import BYTE_OFFSET, EOS_ID
        out = []
        if self.complete():
            out.append(EOS_ID)
        budget = remaining - 1
        for b in sorted(self._allowed_bytes()):
            m = self.clone()
            try:
                m.advance(b)
            except ValueError:
                continue
            if m.min_close() <= budget:
                out.append(b + BYTE_OFFSET)
        return out


def make_fsm(schema):
    """Compile a JSON value into the right automaton: a root-level anyOf
    becomes a MultiFSM of alternatives, anything else a SchemaFSM."""
    if isinstance(schema, SchemaSpec):
        return SchemaFSM(schema)
    if isinstance(schema, dict) and isinstance(schema.get("anyOf"), list):
        return MultiFSM(SchemaFSM(SchemaSpec(s)) for s in schema["anyOf"])
    return SchemaFSM(SchemaSpec(schema))
