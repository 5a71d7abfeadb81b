"""Grammar-constrained decoding for tool-call JSON.

The reference gets syntactically valid tool calls for free from the remote
provider; an in-process engine must *enforce* them.  This module implements
a byte-level pushdown automaton that constrains sampling to

    {"name": "<one of the offered tool names>", "arguments": <JSON object>}

per tool call.  At every decode step the automaton exposes the set of legal
next bytes; the sampler masks the logits to that set (plus EOT when the
automaton is in an accepting state).  With the byte-level tokenizer one
token = one byte, so the automaton advances one state per sampled token.

States use an explicit stack so arbitrarily nested argument objects are
legal up to ``max_depth``; ``max_len`` bounds runaway generations.
"""
from __future__ import annotations

import json
from typing import Dict, Iterable, List, Optional, Set, Tuple

from .tokenizer import EOT

WS = b" \t\n\r"
DIGITS = b"0123456789"
HEX = b"0123456789abcdefABCDEF"
# printable ASCII minus '"' and '\': generation stays single-byte-valid UTF-8
# (multi-byte sequences would need a UTF-8 sub-automaton to stay decodable)
STRING_SAFE = bytes(b for b in range(0x20, 0x7F) if b not in (0x22, 0x5C))
# UTF-8 multi-byte sequences inside generated strings: lead bytes by
# continuation count (conservative well-formed ranges) + continuations
UTF8_LEAD = {1: bytes(range(0xC2, 0xE0)), 2: bytes(range(0xE0, 0xF0)),
             3: bytes(range(0xF0, 0xF5))}
UTF8_CONT = bytes(range(0x80, 0xC0))
UTF8_LEADS = UTF8_LEAD[1] + UTF8_LEAD[2] + UTF8_LEAD[3]


def _utf8_cont_count(b: int) -> int:
    if 0xC2 <= b < 0xE0:
        return 1
    if 0xE0 <= b < 0xF0:
        return 2
    if 0xF0 <= b < 0xF5:
        return 3
    return 0


def _utf8_first_range(lead: int):
    """(n_continuations, first_lo, first_hi): the FIRST continuation range
    depends on the lead byte (overlong/surrogate/out-of-range encodings
    are illegal UTF-8 and must never be generated)."""
    if 0xC2 <= lead < 0xE0:
        return 1, 0x80, 0xBF
    if lead == 0xE0:
        return 2, 0xA0, 0xBF
    if lead == 0xED:
        return 2, 0x80, 0x9F
    if 0xE0 <= lead < 0xF0:
        return 2, 0x80, 0xBF
    if lead == 0xF0:
        return 3, 0x90, 0xBF
    if lead == 0xF4:
        return 3, 0x80, 0x8F
    if 0xF0 <= lead < 0xF5:
        return 3, 0x80, 0xBF
    return 0, 0, 0


def _bs(*parts: Iterable[int]) -> Set[int]:
    out: Set[int] = set()
    for p in parts:
        out.update(p)
    return out


class JsonValueMachine:
    """Byte-level PDA accepting one JSON value.  ``allowed()`` returns legal
    next bytes; ``done`` is True in an accepting state (value complete)."""

    def __init__(self, max_depth: int = 8, max_len: int = 4096, root_object: bool = False):
        self.stack: List[str] = []
        self.state = "value"          # expecting a value
        self.max_depth = max_depth
        self.max_len = max_len
        self.count = 0
        self.done = False
        self.root_object = root_object

    # states:
    #   value        expecting start of a value
    #   string       inside a string
    #   str_escape   after backslash in string
    #   str_u{n}     expecting n more hex digits
    #   num_*        number sub-states
    #   lit:<rest>   finishing a literal (true/false/null)
    #   obj_key      expecting '"' of a key (or '}' right after '{')
    #   obj_colon    expecting ':'
    #   obj_next     expecting ',' or '}'
    #   arr_next     expecting ',' or ']'
    # stack frames: 'O' (in object, after value), 'A' (in array, after value)

    def clone(self) -> "JsonValueMachine":
        c = object.__new__(JsonValueMachine)
        c.stack = list(self.stack)
        c.state = self.state
        c.max_depth = self.max_depth
        c.max_len = self.max_len
        c.count = self.count
        c.done = self.done
        c.root_object = self.root_object
        return c

    def allowed(self) -> Set[int]:
        if self.count >= self.max_len:
            return self._closing_allowed()
        s = self.state
        deep = len(self.stack) >= self.max_depth
        if s == "value":
            if self.root_object and not self.stack:
                return _bs(b"{")  # the root value must be an object
            opts = _bs(WS, b'"', DIGITS, b"-", b"tfn")
            if not deep:
                opts |= _bs(b"{[")
            return opts
        if s == "string":
            opts = _bs(STRING_SAFE, b'"', b"\\")
            if self.count + 4 < self.max_len:
                # a codepoint's continuations must fit inside the budget
                opts |= _bs(UTF8_LEADS)
            return opts
        if s.startswith("str_c"):
            _, lo, hi = s.split(":")
            return _bs(range(int(lo), int(hi) + 1))
        if s == "str_escape":
            return _bs(b'"\\/bfnrtu')
        if s.startswith("str_u"):
            return _bs(HEX)
        if s == "num_int":
            # after first nonzero digits
            return _bs(DIGITS, b".eE") | self._terminators()
        if s == "num_zero":
            # leading zero: no further digits (JSON forbids 06)
            return _bs(b".eE") | self._terminators()
        if s == "num_start":
            return _bs(DIGITS)
        if s == "num_frac_first":
            return _bs(DIGITS)
        if s == "num_frac":
            return _bs(DIGITS, b"eE") | self._terminators()
        if s == "num_exp_sign":
            return _bs(DIGITS, b"+-")
        if s == "num_exp_first":
            return _bs(DIGITS)
        if s == "num_exp":
            return _bs(DIGITS) | self._terminators()
        if s.startswith("lit:"):
            return {s[4:].encode()[0]}
        if s == "obj_key_first":
            return _bs(WS, b'"', b"}")
        if s == "obj_key":
            return _bs(WS, b'"')
        if s == "obj_key_str":
            return _bs(STRING_SAFE, b'"', b"\\")
        if s == "obj_key_escape":
            # no \u in generated keys (obj_key_escape has no hex sub-states)
            return _bs(b'"\\/bfnrt')
        if s == "obj_colon":
            return _bs(WS, b":")
        if s == "obj_next":
            return _bs(WS, b",}")
        if s == "arr_next":
            return _bs(WS, b",]")
        if s == "arr_first":
            opts = _bs(WS, b'"', DIGITS, b"-", b"tfn", b"]")
            if len(self.stack) < self.max_depth:
                opts |= _bs(b"{[")
            return opts
        if s == "done":
            return set()
        raise AssertionError(f"bad state {s}")

    def _closing_allowed(self) -> Set[int]:
        """Over the length budget: only bytes on the shortest path to an
        accepting state, so generation always terminates in valid JSON."""
        s = self.state
        if s.startswith("str_c"):
            # mid-codepoint: the only legal path to acceptance FINISHES the
            # UTF-8 sequence (truncating it would leave an undecodable buf)
            _, lo, hi = s.split(":")
            return _bs(range(int(lo), int(hi) + 1))
        if s in ("string", "obj_key_str"):
            return {0x22}  # '"'
        if s in ("str_escape", "obj_key_escape"):
            return {ord("n")}
        if s.startswith("str_u"):
            return {ord("0")}
        if s in ("num_start", "num_frac_first", "num_exp_first", "num_exp_sign"):
            return {ord("0")}
        if s in ("num_int", "num_zero", "num_frac", "num_exp"):
            t = self._terminators() - _bs(WS, b",")
            return t or {ord("0")}
        if s.startswith("lit:"):
            return {s[4:].encode()[0]}
        if s in ("value",):
            return {ord("0")}
        if s in ("obj_key_first", "obj_next"):
            return {ord("}")}
        if s == "obj_key":
            return {0x22}
        if s == "obj_colon":
            return {ord(":")}
        if s in ("arr_first", "arr_next"):
            return {ord("]")}
        return set()

    def _terminators(self) -> Set[int]:
        """Bytes that may legally follow a number given the stack."""
        if not self.stack:
            return set()
        if self.stack[-1] == "O":
            return _bs(WS, b",}")
        return _bs(WS, b",]")

    def advance(self, b: int) -> None:
        self.count += 1
        s = self.state
        c = bytes([b])
        if s == "value":
            if c in (b" ", b"\t", b"\n", b"\r"):
                self.count -= 0
                return
            if c == b'"':
                self.state = "string"
            elif c == b"{":
                self.stack.append("O")
                self.state = "obj_key_first"
            elif c == b"[":
                self.stack.append("A")
                self.state = "arr_value_or_end"
                # treat immediately: allow value or ']' — model as value with ']' option
                self.state = "arr_first"
            elif c == b"-":
                self.state = "num_start"
            elif c == b"0":
                self.state = "num_zero"
            elif b in DIGITS:
                self.state = "num_int"
            elif c == b"t":
                self.state = "lit:rue"
            elif c == b"f":
                self.state = "lit:alse"
            elif c == b"n":
                self.state = "lit:ull"
            else:
                raise ValueError(f"illegal byte {c!r} in state {s}")
            return
        if s == "arr_first":
            if c in (b" ", b"\t", b"\n", b"\r"):
                return
            if c == b"]":
                self.stack.pop()
                self._value_done()
                return
            # else it's the start of a value — re-dispatch through 'value'
            self.state = "value"
            self.count -= 1
            return self.advance(b)
        if s == "string":
            n, lo, hi = _utf8_first_range(b)
            if n:
                self.state = f"str_c{n}:{lo}:{hi}"
            elif c == b'"':
                self._value_done()
            elif c == b"\\":
                self.state = "str_escape"
            return
        if s == "str_escape":
            if c == b"u":
                self.state = "str_u4"
            else:
                self.state = "string"
            return
        if s.startswith("str_c"):
            n = int(s.split(":")[0][5:])
            self.state = "string" if n == 1 else f"str_c{n - 1}:128:191"
            return
        if s.startswith("str_u"):
            n = int(s[5:])
            self.state = "string" if n == 1 else f"str_u{n - 1}"
            return
        if s in ("num_start", "num_frac_first", "num_exp_first", "num_exp_sign"):
            if s == "num_start":
                self.state = "num_zero" if c == b"0" else "num_int"
            elif s == "num_frac_first":
                self.state = "num_frac"
            elif s == "num_exp_sign" and c in b"+-":
                self.state = "num_exp_first"
            else:
                self.state = "num_exp"
            return
        if s in ("num_int", "num_zero", "num_frac", "num_exp"):
            if b in DIGITS and s != "num_zero":
                return
            if c == b"." and s in ("num_int", "num_zero"):
                self.state = "num_frac_first"
                return
            if c in b"eE" and s in ("num_int", "num_zero", "num_frac"):
                self.state = "num_exp_sign"
                return
            # terminator: the number is done; re-dispatch
            self._value_done()
            self.count -= 1
            return self.advance(b)
        if s.startswith("lit:"):
            rest = s[4:]
            if c != rest[:1].encode():
                raise ValueError(f"illegal literal byte {c!r}")
            self.state = f"lit:{rest[1:]}" if len(rest) > 1 else "__lit_done__"
            if self.state == "__lit_done__":
                self._value_done()
            return
        if s in ("obj_key", "obj_key_first"):
            if c in (b" ", b"\t", b"\n", b"\r"):
                return
            if c == b'"':
                self.state = "obj_key_str"
            elif c == b"}" and s == "obj_key_first":
                self.stack.pop()
                self._value_done()
            return
        if s == "obj_key_str":
            if c == b'"':
                self.state = "obj_colon"
            elif c == b"\\":
                self.state = "obj_key_escape"
            return
        if s == "obj_key_escape":
            self.state = "obj_key_str"
            return
        if s == "obj_colon":
            if c == b":":
                self.state = "value"
            return
        if s == "obj_next":
            if c == b",":
                self.state = "obj_key"
            elif c == b"}":
                self.stack.pop()
                self._value_done()
            return
        if s == "arr_next":
            if c == b",":
                self.state = "value"
            elif c == b"]":
                self.stack.pop()
                self._value_done()
            return
        raise ValueError(f"illegal byte {c!r} in state {s}")

    def _value_done(self) -> None:
        if not self.stack:
            self.state = "done"
            self.done = True
        elif self.stack[-1] == "O":
            self.state = "obj_next"
        else:
            self.state = "arr_next"


class SchemaArgsMachine:
    """Arguments object constrained to a simple JSON schema: the required
    properties are emitted in order with type-correct values, so the tool
    call is not just parseable but *executable* (e.g. delegate_to_agent's
    required ``message`` string, calculator's numeric ``a``/``b``).

    Supports property types string / number / integer / boolean; anything
    richer falls back to the free JsonValueMachine at the value position.
    """

    def __init__(self, schema: dict, max_len: int = 256):
        props = (schema or {}).get("properties", {}) or {}
        required = (schema or {}).get("required", []) or []
        keys = [k for k in required if k in props] or list(props.keys())[:1]
        self.fields = [(k, (props.get(k) or {}).get("type", "string")) for k in keys]
        self.max_len = max_len
        self.count = 0
        self.done = False
        # build the program: list of ("lit", bytes) | ("str",) | ("num",) |
        # ("bool",) | ("free",)
        prog: List[tuple] = [("lit", b"{")]
        for i, (k, t) in enumerate(self.fields):
            sep = b", " if i else b""
            # json.dumps escapes quotes/backslashes/control bytes in the
            # property name (a raw k.encode() would emit invalid JSON)
            prog.append(("lit", sep + json.dumps(k).encode() + b": "))
            if t == "string":
                prog.append(("str",))
            elif t in ("number", "integer"):
                prog.append(("num",))
            elif t == "boolean":
                prog.append(("bool",))
            else:
                prog.append(("free",))
        prog.append(("lit", b"}"))
        self.prog = prog
        self.pi = 0       # program counter
        self.off = 0      # offset within a literal / value state
        self.sub: Optional[JsonValueMachine] = None
        self._bool_rest = b""
        self.u8 = 0  # pending UTF-8 continuation bytes in a string value
        self.u8lo, self.u8hi = 0x80, 0xBF

    def _cur(self):
        return self.prog[self.pi] if self.pi < len(self.prog) else None

    def clone(self) -> "SchemaArgsMachine":
        c = object.__new__(SchemaArgsMachine)
        c.fields = self.fields      # immutable after build
        c.max_len = self.max_len
        c.count = self.count
        c.done = self.done
        c.prog = self.prog          # immutable after build
        c.pi = self.pi
        c.off = self.off
        c.sub = self.sub.clone() if self.sub is not None else None
        c._bool_rest = self._bool_rest
        c.u8, c.u8lo, c.u8hi = self.u8, self.u8lo, self.u8hi
        return c

    def allowed(self) -> Set[int]:
        cur = self._cur()
        if cur is None:
            return set()
        kind = cur[0]
        if kind == "lit":
            return {cur[1][self.off]}
        budget_left = self.count < self.max_len
        if kind == "str":
            if self.u8:
                return _bs(range(self.u8lo, self.u8hi + 1))
            if self.off == 0:
                return {0x22}
            opts = {0x22}
            if budget_left:
                opts |= set(STRING_SAFE)
                # a multi-byte codepoint needs budget for its continuations
                if self.count + 4 < self.max_len:
                    opts |= set(UTF8_LEADS)
            return opts
        if kind == "num":
            if self.off == 0:
                return _bs(b"-123456789") | {ord("0")}
            if self.off == -2:  # after '-': a first digit is mandatory
                return _bs(DIGITS)
            if self.off == -1:  # after a leading zero: number is complete
                return self._next_lit_first()
            return _bs(DIGITS) | self._next_lit_first() if budget_left else self._next_lit_first()
        if kind == "bool":
            if self.off == 0:
                return _bs(b"tf")
            return {self._bool_rest[0]}
        if kind == "free":
            if self.sub is None:
                # object-rooted: a root-level scalar has no terminator of its
                # own, so untyped/object-typed fields emit a JSON object
                self.sub = JsonValueMachine(
                    max_len=max(16, self.max_len - self.count), root_object=True
                )
            opts = set(self.sub.allowed())
            if self.sub.done:
                opts |= self._next_lit_first()
            return opts
        raise AssertionError(kind)

    def _next_lit_first(self) -> Set[int]:
        nxt = self.prog[self.pi + 1]
        return {nxt[1][0]}

    def advance(self, b: int) -> None:
        self.count += 1
        cur = self._cur()
        kind = cur[0]
        if kind == "lit":
            self.off += 1
            if self.off == len(cur[1]):
                self.pi += 1
                self.off = 0
                if self.pi == len(self.prog):
                    self.done = True
            return
        if kind == "str":
            if self.u8:
                self.u8 -= 1
                self.u8lo, self.u8hi = 0x80, 0xBF
                self.off += 1
                return
            n, lo, hi = _utf8_first_range(b)
            if n:
                self.u8, self.u8lo, self.u8hi = n, lo, hi
                self.off += 1
                return
            if self.off > 0 and b == 0x22:
                self.pi += 1
                self.off = 0
                return
            self.off += 1
            return
        if kind == "num":
            if (self.off > 0 or self.off == -1) and b not in DIGITS:
                # terminator: belongs to the next literal
                self.pi += 1
                self.off = 0
                self.count -= 1
                return self.advance(b)
            if self.off == 0:
                self.off = -2 if b == ord("-") else (-1 if b == ord("0") else 1)
                return
            if self.off == -2:
                self.off = -1 if b == ord("0") else 1
                return
            self.off += 1
            return
        if kind == "bool":
            if self.off == 0:
                self._bool_rest = b"rue" if b == ord("t") else b"alse"
                self.off = 1
                return
            self._bool_rest = self._bool_rest[1:]
            if not self._bool_rest:
                self.pi += 1
                self.off = 0
            return
        if kind == "free":
            if self.sub.done and b in self._next_lit_first():
                self.pi += 1
                self.off = 0
                self.sub = None
                self.count -= 1
                return self.advance(b)
            self.sub.advance(b)
            return
        raise AssertionError(kind)


class ToolCallGrammar:
    """Constrains one tool call:
    ``{"name": "<tool>", "arguments": {…}}`` then EOT.

    ``allowed_tokens()`` returns legal token ids (bytes plus EOT in the
    accepting state); ``advance(token)`` consumes the sampled token.
    ``parse()`` returns the finished (name, arguments_json) pair.
    """

    PRE = b'{"name": "'
    MID = b'", "arguments": '

    def __init__(self, tool_names: Optional[List[str]] = None, max_args_len: int = 2048,
                 tools: Optional[List[dict]] = None, pre_in_prompt: bool = False):
        self.schemas: Dict[str, dict] = {}
        if tools:
            names = []
            for t in tools:
                fn = t.get("function", {}) or {}
                if fn.get("name"):
                    names.append(fn["name"])
                    self.schemas[fn["name"]] = fn.get("parameters") or {}
            tool_names = names
        if not tool_names:
            raise ValueError("no tools to constrain to")
        self.names = sorted(set(tool_names))
        # trie operates on the JSON-escaped byte form (json.dumps minus the
        # surrounding quotes): a name containing '"' or '\\' would
        # otherwise emit invalid JSON
        self._name_bytes = [json.dumps(n).encode()[1:-1] for n in self.names]
        if pre_in_prompt:
            # the engine emitted PRE as prompt tokens (one prefill chunk):
            # parsing must still see the full JSON, so pre-seed the buffer
            self.buf = bytearray(self.PRE)
            self.phase = "name"
        else:
            self.buf = bytearray()
            self.phase = "pre"   # pre → name → mid → args → done
        self.pos = 0
        self.max_args_len = max_args_len
        self.name_prefix = b""
        self.args = JsonValueMachine(max_len=max_args_len, root_object=True)
        self.finished = False

    def _select_args_machine(self, name: str) -> None:
        """Once the tool is known, constrain the arguments to its parameter
        schema so the produced call is executable, not merely parseable."""
        schema = self.schemas.get(name) or {}
        if schema.get("properties"):
            self.args = SchemaArgsMachine(schema, max_len=self.max_args_len)
        else:
            self.args = JsonValueMachine(max_len=self.max_args_len, root_object=True)

    def clone(self) -> "ToolCallGrammar":
        c = object.__new__(ToolCallGrammar)
        c.schemas = self.schemas            # shared, read-only
        c.names = self.names
        c._name_bytes = self._name_bytes
        c.buf = bytearray(self.buf)
        c.phase = self.phase
        c.pos = self.pos
        c.max_args_len = self.max_args_len
        c.name_prefix = self.name_prefix    # bytes, immutable
        c.args = self.args.clone()
        c.finished = self.finished
        return c

    def forced_run(self, cap: int = 64) -> List[int]:
        """The upcoming run of bytes with exactly one legal choice — the
        scaffolding the engine can fold into one prefill chunk instead of
        one masked decode step per byte (MID is 16 forced bytes; schema
        literals and a single matching tool name add more).  Guarded by
        cheap phase checks so unconstrained positions cost nothing; EOT
        never appears in a forced run."""
        # fast bail: states that always have >1 legal byte
        if self.phase in ("done",):
            return []
        if self.phase == "args":
            a = self.args
            if isinstance(a, JsonValueMachine):
                return []  # free JSON: almost never singleton for long
            cur = a._cur()
            if cur is None or cur[0] != "lit":
                return []
        out: List[int] = []
        g = None
        while len(out) < cap:
            src = g if g is not None else self
            al = src.allowed_tokens()
            if len(al) != 1:
                break
            b = next(iter(al))
            if b == EOT or b > 255:
                break
            if g is None:
                g = self.clone()
            g.advance(b)
            out.append(b)
        return out

    def allowed_tokens(self) -> Set[int]:
        if self.phase == "pre":
            return {self.PRE[self.pos]}
        if self.phase == "name":
            nexts: Set[int] = set()
            for nb in self._name_bytes:
                if nb.startswith(self.name_prefix):
                    if len(nb) > len(self.name_prefix):
                        nexts.add(nb[len(self.name_prefix)])
                    else:
                        nexts.add(ord('"'))  # exact match may close
            return nexts
        if self.phase == "mid":
            return {self.MID[self.pos]}
        if self.phase == "args":
            allowed = set(self.args.allowed())
            if self.args.done:
                allowed.add(ord("}"))
            return allowed
        if self.phase == "close":
            return {ord("}")}
        return {EOT}

    def advance(self, token: int) -> None:
        if token == EOT and self.phase == "done":
            self.finished = True
            return
        b = token
        self.buf.append(b)
        if self.phase == "pre":
            self.pos += 1
            if self.pos == len(self.PRE):
                self.phase = "name"
            return
        if self.phase == "name":
            if b == ord('"') and self.name_prefix in self._name_bytes:
                # unescape the trie form back to the raw tool name
                raw = json.loads('"' + self.name_prefix.decode() + '"')
                self._select_args_machine(raw)
                self.phase = "mid"
                self.pos = 1  # the '"' consumed is MID[0]... MID starts with '"'
                return
            self.name_prefix += bytes([b])
            return
        if self.phase == "mid":
            self.pos += 1
            if self.pos == len(self.MID):
                self.phase = "args"
            return
        if self.phase == "args":
            if self.args.done and b == ord("}"):
                self.phase = "done"
                return
            self.args.advance(b)
            return
        raise ValueError(f"unexpected token {token} in phase {self.phase}")

    @property
    def accepting(self) -> bool:
        return self.phase == "done"

    def parse(self) -> Tuple[str, str]:
        obj = json.loads(bytes(self.buf).decode("utf-8"))
        return obj["name"], json.dumps(obj["arguments"])
