"""JSON-mode constrained decoding (OpenAI `response_format`).

The reference gets JSON mode for free by passing `response_format`
through to vLLM (api/openai/v1/chat_completions.go:350-515); the
in-house engine enforces it at the sampler: a streaming JSON-prefix
automaton vets each sampled token's decoded text, the runner masks and
resamples tokens that would break JSON validity, holds EOS until the
top-level value closes, and force-stops once it has
(runner.execute -> _constrain_json).

The automaton accepts exactly the prefixes of valid JSON documents
whose top-level value is an object (OpenAI json_object semantics),
processing text incrementally with O(len(piece)) work per token.
"""
from __future__ import annotations

_WS = " \t\n\r"
_LITERALS = ("true", "false", "null")


class JsonPrefixValidator:
    """Streaming acceptor for prefixes of a single top-level JSON object.

    States are kept in plain attributes so snapshot/restore (for
    rejection sampling) is a cheap tuple copy.
    """

    def __init__(self):
        # stack entries: 'o' in-object, 'a' in-array
        self.stack: list[str] = []
        # mode: what we're lexing right now
        #  'start'  - before the top-level '{'
        #  'value'  - expecting a value
        #  'string' - inside a string (self.in_key says which kind)
        #  'escape' - after backslash in a string
        #  'u' + n  - unicode escape, n hex digits consumed (0-3)
        #  'number' - inside a number
        #  'literal'- inside true/false/null (self.lit, self.lit_pos)
        #  'after'  - after a complete value (expect , } ] or end)
        #  'key'    - in an object, expecting a key string or '}'
        #  'colon'  - after a key, expecting ':'
        #  'done'   - top-level object closed (only whitespace after)
        self.mode = "start"
        self.in_key = False
        self.lit = ""
        self.lit_pos = 0
        self.num_state = ""  # tracks number grammar position
        self.ubuf = ""  # hex digits of an in-flight \u escape

    # -------------------------------------------------------------- copy
    def snapshot(self):
        return (
            tuple(self.stack), self.mode, self.in_key, self.lit,
            self.lit_pos, self.num_state, self.ubuf,
        )

    def restore(self, snap) -> None:
        (stack, self.mode, self.in_key, self.lit, self.lit_pos,
         self.num_state, self.ubuf) = snap
        self.stack = list(stack)

    @property
    def complete(self) -> bool:
        return self.mode == "done"

    # -------------------------------------------------------------- feed
    def feed(self, text: str) -> bool:
        """Consume text; False (state unspecified) if it breaks validity —
        callers snapshot() first and restore() on rejection."""
        for ch in text:
            if not self._feed_char(ch):
                return False
        return True

    def _close_value(self) -> None:
        """A value just finished: what comes next?"""
        if not self.stack:
            self.mode = "done"
        else:
            self.mode = "after"

    def _feed_char(self, ch: str) -> bool:  # noqa: C901 (explicit automaton)
        m = self.mode
        if m == "start":
            if ch in _WS:
                return True
            if ch == "{":
                if not self._enter_object():
                    return False
                self.stack.append("o")
                self.mode = "key"
                return True
            return False
        if m == "done":
            return ch in _WS
        if m == "string":
            if ch == "\\":
                if not self._escape_ok():
                    return False
                self.mode = "escape"
                return True
            if ch == '"':
                if self.in_key:
                    if not self._end_key():
                        return False
                    self.mode = "colon"
                    self.in_key = False
                else:
                    if not self._end_string_value():
                        return False
                    self._close_value()
                return True
            if ch in "\n\r":  # control chars invalid in strings
                return False
            return self._str_char(ch)
        if m == "escape":
            if ch in '"\\/bfnrt':
                self.mode = "string"
                dec = {"b": "\b", "f": "\f", "n": "\n", "r": "\r",
                       "t": "\t"}.get(ch, ch)
                return self._str_char(dec)
            if ch == "u":
                if not self._escape_u_ok():
                    return False
                self.mode = "u0"
                self.ubuf = ""
                return True
            return False
        if m.startswith("u"):
            if ch in "0123456789abcdefABCDEF":
                n = int(m[1]) + 1
                self.ubuf += ch
                if n == 4:
                    self.mode = "string"
                    # decoded escape flows into key/enum buffers too
                    return self._str_char(chr(int(self.ubuf, 16)))
                self.mode = f"u{n}"
                return True
            return False
        if m == "literal":
            if self.lit_pos < len(self.lit) and ch == self.lit[self.lit_pos]:
                self.lit_pos += 1
                if self.lit_pos == len(self.lit):
                    self._close_value()
                return True
            return False  # noqa: TRY300
        if m == "number":
            ns = self.num_state
            if ch.isdigit():
                self.num_state = {
                    "-": "int", "int": "int", "0": "badzero", "frac.": "frac",
                    "frac": "frac", "e": "edig", "esign": "edig",
                    "edig": "edig",
                }.get(ns, "int") if ns != "0" else "badzero"
                if ns == "0":
                    return False  # leading zero followed by digit
                return True
            if ch == "." and ns in ("int", "0"):
                if not self._number_frac_ok():
                    return False
                self.num_state = "frac."
                return True
            if ch in "eE" and ns in ("int", "0", "frac"):
                if not self._number_frac_ok():
                    return False
                self.num_state = "e"
                return True
            if ch in "+-" and ns == "e":
                self.num_state = "esign"
                return True
            # number ends; re-dispatch ch in the after-value state
            if ns in ("int", "0", "frac", "edig"):
                if not self._end_number(ns):
                    return False
                self._close_value()
                return self._feed_char(ch)
            return False
        if m == "key":
            if ch in _WS:
                return True
            if ch == '"':
                self.mode = "string"
                self.in_key = True
                self._begin_key()
                return True
            if ch == "}" and self.stack and self.stack[-1] == "o":
                if not self._exit_object():
                    return False
                self.stack.pop()
                self._close_value()
                return True
            return False
        if m == "colon":
            if ch in _WS:
                return True
            if ch == ":":
                self.mode = "value"
                return True
            return False
        if m == "value":
            if ch in _WS:
                return True
            if ch == '"':
                if not self._begin_value("string"):
                    return False
                self.mode = "string"
                self.in_key = False
                return True
            if ch == "{":
                if not self._enter_object():
                    return False
                self.stack.append("o")
                self.mode = "key"
                return True
            if ch == "[":
                if not self._enter_array():
                    return False
                self.stack.append("a")
                return True  # stay in 'value'; ']' handled below
            if ch == "]" and self.stack and self.stack[-1] == "a":
                self._exit_array()
                self.stack.pop()  # empty array
                self._close_value()
                return True
            if ch == "-":
                if not self._begin_value("number"):
                    return False
                self.mode = "number"
                self.num_state = "-"
                return True
            if ch == "0":
                if not self._begin_value("number"):
                    return False
                self.mode = "number"
                self.num_state = "0"
                return True
            if ch.isdigit():
                if not self._begin_value("number"):
                    return False
                self.mode = "number"
                self.num_state = "int"
                return True
            for lit in _LITERALS:
                if ch == lit[0]:
                    kind = "null" if lit == "null" else "boolean"
                    if not self._begin_value(kind):
                        return False
                    self.mode = "literal"
                    self.lit = lit
                    self.lit_pos = 1
                    return True
            return False
        if m == "after":
            if ch in _WS:
                return True
            top = self.stack[-1] if self.stack else ""
            if ch == ",":
                if top == "o":
                    self.mode = "key"
                else:
                    self._next_array_item()
                    self.mode = "value"
                return True
            if ch == "}" and top == "o":
                if not self._exit_object():
                    return False
                self.stack.pop()
                self._close_value()
                return True
            if ch == "]" and top == "a":
                self._exit_array()
                self.stack.pop()
                self._close_value()
                return True
            return False
        return False

    # ------------------------------------------------- schema hook points
    # No-ops here; SchemaValidator overrides them to enforce a JSON
    # schema on top of the same lexer.
    def _enter_object(self) -> bool:
        return True

    def _exit_object(self) -> bool:
        return True

    def _enter_array(self) -> bool:
        return True

    def _exit_array(self) -> None:
        pass

    def _next_array_item(self) -> None:
        pass

    def _begin_key(self) -> None:
        pass

    def _end_key(self) -> bool:
        return True

    def _begin_value(self, kind: str) -> bool:
        return True

    def _end_string_value(self) -> bool:
        return True

    def _end_number(self, num_state: str) -> bool:
        return True

    def _number_frac_ok(self) -> bool:
        return True

    def _escape_u_ok(self) -> bool:
        return True

    def _escape_ok(self) -> bool:
        return True

    def _str_char(self, ch: str) -> bool:
        return True


def _resolve_ref(schema, root):
    """Follow a local $ref ("#/$defs/x" / "#/definitions/x") chain."""
    seen = 0
    while isinstance(schema, dict) and "$ref" in schema and seen < 16:
        ref = schema["$ref"]
        if not isinstance(ref, str) or not ref.startswith("#/"):
            return {}
        node = root
        for part in ref[2:].split("/"):
            if not isinstance(node, dict) or part not in node:
                return {}
            node = node[part]
        schema = node
        seen += 1
    return schema if isinstance(schema, dict) else {}


_KIND_OF_TYPE = {
    "string": "string", "number": "number", "integer": "number",
    "boolean": "boolean", "null": "null", "object": "object",
    "array": "array",
}


class SchemaValidator(JsonPrefixValidator):
    """JSON-prefix acceptor constrained by a JSON schema subset.

    Enforced: type (incl. unions via list), object properties /
    required / additionalProperties:false, nested objects, array items,
    enum/const of strings (with streaming prefix rejection), integer
    (no fraction/exponent), local $ref ($defs / definitions).
    Not enforced (documented; well-formedness still applies): numeric
    ranges, min/maxItems, patterns, string formats, anyOf/oneOf.

    Reference analog: vLLM guided_json via response_format json_schema,
    which the reference forwards verbatim (chat_completions.go).
    """

    def __init__(self, schema):
        super().__init__()
        self.root_doc = schema if isinstance(schema, dict) else {}
        self.pending = _resolve_ref(self.root_doc, self.root_doc)
        self.frames: tuple = ()  # immutable stack of frame tuples
        self.keybuf = ""
        self.strbuf = ""
        self.enum: tuple = ()  # candidate enum strings for current value

    # frames: ("o", schema, frozenset(seen)) | ("a", items_schema)
    # ------------------------------------------------------------- copy
    def snapshot(self):
        return (
            super().snapshot(), self.pending, self.frames, self.keybuf,
            self.strbuf, self.enum,
        )

    def restore(self, snap) -> None:
        base, self.pending, self.frames, self.keybuf, self.strbuf, \
            self.enum = snap
        super().restore(base)

    # ------------------------------------------------------------ helpers
    def _sch(self):
        return self.pending if isinstance(self.pending, dict) else {}

    def _type_ok(self, kind: str) -> bool:
        sch = self._sch()
        t = sch.get("type")
        if t is None:
            return True
        types = t if isinstance(t, list) else [t]
        return any(_KIND_OF_TYPE.get(x) == kind for x in types)

    # ------------------------------------------------------------- hooks
    def _enter_object(self) -> bool:
        if not self._begin_value("object"):  # enum/type checks included
            return False
        sch = self._sch()
        self.frames = self.frames + (("o", sch, frozenset()),)
        return True

    def _exit_object(self) -> bool:
        kind, sch, seen = self.frames[-1]
        req = sch.get("required") or []
        if not set(req) <= seen:
            return False  # hold '}' until every required key appears
        self.frames = self.frames[:-1]
        return True

    def _enter_array(self) -> bool:
        if not self._begin_value("array"):
            return False
        sch = self._sch()
        items = _resolve_ref(sch.get("items") or {}, self.root_doc)
        self.frames = self.frames + (("a", items, frozenset()),)
        self.pending = items
        return True

    def _exit_array(self) -> None:
        self.frames = self.frames[:-1]

    def _next_array_item(self) -> None:
        self.pending = self.frames[-1][1]

    def _begin_key(self) -> None:
        self.keybuf = ""

    def _end_key(self) -> bool:
        kind, sch, seen = self.frames[-1]
        props = sch.get("properties")
        sub = None
        if isinstance(props, dict):
            sub = props.get(self.keybuf)
        if sub is None:
            addl = sch.get("additionalProperties", True)
            if addl is False and isinstance(props, dict):
                return False  # unknown key under a closed schema
            sub = addl if isinstance(addl, dict) else {}
        self.frames = self.frames[:-1] + (
            ("o", sch, seen | {self.keybuf}),
        )
        self.pending = _resolve_ref(sub, self.root_doc)
        return True

    def _begin_value(self, kind: str) -> bool:
        sch = self._sch()
        enum = sch.get("enum")
        if enum is None and "const" in sch:
            enum = [sch["const"]]
        self.enum = ()
        if enum is not None:
            # streaming enum support covers string members; a value of a
            # different kind must match some member's kind
            kinds = set()
            strs = []
            for e in enum:
                if isinstance(e, str):
                    kinds.add("string")
                    strs.append(e)
                elif isinstance(e, bool):
                    kinds.add("boolean")
                elif e is None:
                    kinds.add("null")
                elif isinstance(e, (int, float)):
                    kinds.add("number")
                elif isinstance(e, dict):
                    kinds.add("object")
                elif isinstance(e, list):
                    kinds.add("array")
            if kind not in kinds:
                return False
            self.enum = tuple(strs)
        if not self._type_ok(kind):
            return False
        if kind == "string":
            self.strbuf = ""
        return True

    def _str_char(self, ch: str) -> bool:
        if self.in_key:
            self.keybuf += ch
            # closed object: the key must be a prefix of some declared
            # property — rejecting early keeps the sampler's candidate
            # walk from wandering inside an unterminated key string
            kind, sch, _seen = self.frames[-1]
            props = sch.get("properties")
            if (
                isinstance(props, dict)
                and sch.get("additionalProperties", True) is False
            ):
                return any(p.startswith(self.keybuf) for p in props)
            return True
        self.strbuf += ch
        if self.enum:
            return any(e.startswith(self.strbuf) for e in self.enum)
        return True

    def _end_string_value(self) -> bool:
        if self.enum:
            return self.strbuf in self.enum
        return True

    def _end_number(self, num_state: str) -> bool:
        if not self._number_frac_ok():
            return num_state in ("int", "0")
        return True

    def _number_frac_ok(self) -> bool:
        sch = self._sch()
        t = sch.get("type")
        types = t if isinstance(t, list) else [t] if t else []
        if types and "number" not in types and "integer" in types:
            return False  # integer-only: no '.' or exponent
        return True

    def _allowed_next_chars(self):
        """Set of next chars that keep the active prefix check alive, or
        None when unconstrained."""
        if self.in_key:
            kind, sch, _seen = self.frames[-1]
            props = sch.get("properties")
            if (
                isinstance(props, dict)
                and sch.get("additionalProperties", True) is False
            ):
                n = len(self.keybuf)
                return {p[n] for p in props
                        if p.startswith(self.keybuf) and len(p) > n}
            return None
        if self.enum:
            n = len(self.strbuf)
            return {e[n] for e in self.enum
                    if e.startswith(self.strbuf) and len(e) > n}
        return None

    def _escape_ok(self) -> bool:
        allowed = self._allowed_next_chars()
        if allowed is None:
            return True
        # chars reachable through simple escapes
        esc = {'"', chr(92), "/", chr(8), chr(12), chr(10), chr(13),
               chr(9)}
        return bool(allowed & esc)

    def _escape_u_ok(self) -> bool:
        r"""\uXXXX escapes inside prefix-checked strings are rejected at
        the 'u': the four hex digits would be accepted blindly and then
        the decoded char rejected -- a dead-end state the sampler cannot
        escape (every continuation invalid). Literal characters cover
        the same strings."""
        if self.in_key:
            kind, sch, _seen = self.frames[-1]
            props = sch.get("properties")
            return not (
                isinstance(props, dict)
                and sch.get("additionalProperties", True) is False
            )
        return not self.enum
