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

    # -------------------------------------------------------------- copy
    def snapshot(self):
        return (
            tuple(self.stack), self.mode, self.in_key, self.lit,
            self.lit_pos, self.num_state,
        )

    def restore(self, snap) -> None:
        (stack, self.mode, self.in_key, self.lit, self.lit_pos,
         self.num_state) = snap
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
                self.stack.append("o")
                self.mode = "key"
                return True
            return False
        if m == "done":
            return ch in _WS
        if m == "string":
            if ch == "\\":
                self.mode = "escape"
                return True
            if ch == '"':
                if self.in_key:
                    self.mode = "colon"
                    self.in_key = False
                else:
                    self._close_value()
                return True
            return ch not in "\n\r"  # control chars invalid in strings
        if m == "escape":
            if ch in '"\\/bfnrt':
                self.mode = "string"
                return True
            if ch == "u":
                self.mode = "u0"
                return True
            return False
        if m.startswith("u"):
            if ch in "0123456789abcdefABCDEF":
                n = int(m[1]) + 1
                self.mode = "string" if n == 4 else f"u{n}"
                return True
            return False
        if m == "literal":
            if self.lit_pos < len(self.lit) and ch == self.lit[self.lit_pos]:
                self.lit_pos += 1
                if self.lit_pos == len(self.lit):
                    self._close_value()
                return True
            return False
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
                self.num_state = "frac."
                return True
            if ch in "eE" and ns in ("int", "0", "frac"):
                self.num_state = "e"
                return True
            if ch in "+-" and ns == "e":
                self.num_state = "esign"
                return True
            # number ends; re-dispatch ch in the after-value state
            if ns in ("int", "0", "frac", "edig"):
                self._close_value()
                return self._feed_char(ch)
            return False
        if m == "key":
            if ch in _WS:
                return True
            if ch == '"':
                self.mode = "string"
                self.in_key = True
                return True
            if ch == "}" and self.stack and self.stack[-1] == "o":
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
                self.mode = "string"
                self.in_key = False
                return True
            if ch == "{":
                self.stack.append("o")
                self.mode = "key"
                return True
            if ch == "[":
                self.stack.append("a")
                return True  # stay in 'value'; ']' handled below
            if ch == "]" and self.stack and self.stack[-1] == "a":
                self.stack.pop()  # empty array
                self._close_value()
                return True
            if ch == "-":
                self.mode = "number"
                self.num_state = "-"
                return True
            if ch == "0":
                self.mode = "number"
                self.num_state = "0"
                return True
            if ch.isdigit():
                self.mode = "number"
                self.num_state = "int"
                return True
            for lit in _LITERALS:
                if ch == lit[0]:
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
                self.mode = "key" if top == "o" else "value"
                return True
            if ch == "}" and top == "o":
                self.stack.pop()
                self._close_value()
                return True
            if ch == "]" and top == "a":
                self.stack.pop()
                self._close_value()
                return True
            return False
        return False
