"""Guided decoding: OpenAI `response_format` JSON mode.

The reference stack delegates structured outputs to vLLM's guided
decoding; this is the engine-native counterpart. A character-level
push-down automaton accepts exactly the prefixes of valid JSON texts;
at each decode step the sampler masks the logits to tokens whose decoded
text keeps the output a valid JSON prefix, and once a complete top-level
value has been produced only EOS remains legal.

`response_format: {"type": "json_object"}` guarantees syntactically
valid JSON. `{"type": "json_schema", ...}`: flat object schemas
(`type: object` + `properties` + `additionalProperties: false`)
compile to an ENFORCED template (SchemaGuide — exact keys in
declaration order, typed value regions); looser schemas use the
generic grammar with end-of-request validation, surfacing failures as
`finish_reason: "error_json_schema"` rather than silently returning a
non-conforming object. vLLM-style `guided_choice` / `guided_regex`
ride the same masking machinery through `regex` partial matching
(RegexGuide).

Masking strategy: candidate tokens are taken from the top-K logits
(K=64, widening x8 on miss up to the full vocab); each candidate's text
is checked against a copy of the automaton. Token texts are cached per
tokenizer. This is the vLLM-v0 style lazy approach — no vocab-wide FSM
precompilation — which fits serving: a handful of candidates almost
always contains many grammar-legal tokens.
"""

from __future__ import annotations

import json
from typing import List, Optional

# ---------------------------------------------------------------------------
# character-level JSON prefix automaton

_WS = " \t\n\r"
_DIGITS = "0123456789"


class JsonPrefixValidator:
    """Incremental validator: feed() characters one at a time; returns
    False (and consumes nothing) when the character cannot extend any
    valid JSON text. `done` is True once a complete top-level value has
    been read (trailing whitespace stays legal)."""

    # frame kinds on the stack
    OBJ_START, OBJ_KEY, OBJ_COLON, OBJ_VALUE, OBJ_NEXT = range(5)
    ARR_START, ARR_VALUE, ARR_NEXT = 5, 6, 7

    def __init__(self) -> None:
        self.stack: List[int] = []
        # mode: 'value' awaiting a value, 'string'/'stringkey' inside a
        # string, 'esc'/'esckey' after backslash, 'u{n}'/'ukey{n}' inside
        # \uXXXX, 'num' inside a number, 'lit' inside true/false/null,
        # 'post' after a finished value, 'end' complete
        self.mode = "value"
        self.num = ""
        self.lit_target = ""
        self.lit_pos = 0
        self.done = False

    def copy(self) -> "JsonPrefixValidator":
        c = JsonPrefixValidator.__new__(JsonPrefixValidator)
        c.stack = list(self.stack)
        c.mode = self.mode
        c.num = self.num
        c.lit_target = self.lit_target
        c.lit_pos = self.lit_pos
        c.done = self.done
        return c

    # -- helpers -----------------------------------------------------------
    def _value_done(self) -> None:
        """A complete value was just read; pop into the container state."""
        if not self.stack:
            self.mode = "end"
            self.done = True
            return
        top = self.stack[-1]
        if top in (self.OBJ_COLON,):
            self.stack[-1] = self.OBJ_NEXT
        elif top in (self.ARR_START, self.ARR_NEXT):
            self.stack[-1] = self.ARR_VALUE
        self.mode = "post"

    def _num_could_end(self) -> bool:
        n = self.num
        return bool(n) and n[-1] in _DIGITS

    def feed(self, ch: str) -> bool:
        m = self.mode
        if m == "end":
            return ch in _WS
        if m in ("string", "stringkey"):
            key = m == "stringkey"
            if ch == "\\":
                self.mode = "esckey" if key else "esc"
                return True
            if ch == '"':
                if key:
                    self.stack[-1] = self.OBJ_KEY
                    self.mode = "post"
                else:
                    self._value_done()
                return True
            return ch >= " "
        if m in ("esc", "esckey"):
            key = m == "esckey"
            if ch in '"\\/bfnrt':
                self.mode = "stringkey" if key else "string"
                return True
            if ch == "u":
                self.mode = ("ukey0" if key else "u0")
                return True
            return False
        if m.startswith("u") and (m[1:].isdigit() or m.startswith("ukey")):
            key = m.startswith("ukey")
            n = int(m[4:] if key else m[1:])
            if ch in "0123456789abcdefABCDEF":
                if n == 3:
                    self.mode = "stringkey" if key else "string"
                else:
                    self.mode = (f"ukey{n + 1}" if key else f"u{n + 1}")
                return True
            return False
        if m == "lit":
            if self.lit_pos < len(self.lit_target) and \
                    ch == self.lit_target[self.lit_pos]:
                self.lit_pos += 1
                if self.lit_pos == len(self.lit_target):
                    self._value_done()
                return True
            return False
        if m == "num":
            n = self.num
            if ch in _DIGITS:
                if n in ("-0", "0"):
                    return False  # no leading zeros
                self.num += ch
                return True
            if ch == "." and n and n[-1] in _DIGITS and "." not in n \
                    and "e" not in n and "E" not in n:
                self.num += ch
                return True
            if ch in "eE" and n and n[-1] in _DIGITS and "e" not in n \
                    and "E" not in n:
                self.num += ch
                return True
            if ch in "+-" and n and n[-1] in "eE":
                self.num += ch
                return True
            if self._num_could_end():
                self._value_done()
                return self.feed(ch)
            return False
        if m == "value":
            if ch in _WS:
                return True
            if ch == '"':
                self.mode = "string"
                return True
            if ch == "{":
                self.stack.append(self.OBJ_START)
                self.mode = "post"
                return True
            if ch == "[":
                self.stack.append(self.ARR_START)
                self.mode = "value"
                # "]" must close an empty array: handled in post? no —
                # array start awaits value OR ']': special-case below
                self.mode = "post_arr_start"
                return True
            if ch == "-" or ch in _DIGITS:
                self.mode = "num"
                self.num = ch
                return True
            for lit in ("true", "false", "null"):
                if ch == lit[0]:
                    self.mode = "lit"
                    self.lit_target = lit
                    self.lit_pos = 1
                    return True
            return False
        if m == "post_arr_start":
            if ch in _WS:
                return True
            if ch == "]":
                self.stack.pop()
                self._value_done()
                return True
            # fall through to a value inside the array
            self.mode = "value"
            ok = self.feed(ch)
            if not ok:
                self.mode = "post_arr_start"
            return ok
        if m == "post":
            if ch in _WS:
                return True
            if not self.stack:
                return False
            top = self.stack[-1]
            if top == self.OBJ_START:
                if ch == '"':
                    self.mode = "stringkey"
                    return True
                if ch == "}":
                    self.stack.pop()
                    self._value_done()
                    return True
                return False
            if top == self.OBJ_KEY:
                if ch == ":":
                    self.stack[-1] = self.OBJ_COLON
                    self.mode = "value"
                    return True
                return False
            if top == self.OBJ_NEXT:
                if ch == ",":
                    self.stack[-1] = self.OBJ_START
                    # next must be a key (not '}')
                    self.mode = "post_obj_key"
                    return True
                if ch == "}":
                    self.stack.pop()
                    self._value_done()
                    return True
                return False
            if top == self.ARR_VALUE:
                if ch == ",":
                    self.stack[-1] = self.ARR_NEXT
                    self.mode = "value"
                    return True
                if ch == "]":
                    self.stack.pop()
                    self._value_done()
                    return True
                return False
            return False
        if m == "post_obj_key":
            if ch in _WS:
                return True
            if ch == '"':
                self.mode = "stringkey"
                return True
            return False
        return False

    def feed_text(self, text: str) -> bool:
        for ch in text:
            if not self.feed(ch):
                return False
        return True

    def would_accept(self, text: str) -> Optional["JsonPrefixValidator"]:
        """Copy-and-feed; returns the advanced copy or None."""
        c = self.copy()
        return c if c.feed_text(text) else None

    @property
    def complete(self) -> bool:
        """True if the text so far is (or can be finished as) a complete
        JSON document by stopping now."""
        if self.done:
            return True
        # a bare top-level number is complete once it can end
        return (not self.stack and self.mode == "num"
                and self._num_could_end())


# ---------------------------------------------------------------------------
# schema-compiled guide: flat object schemas become a forced template


_TYPE_STARTS = {
    "string": '"',
    "number": "-0123456789",
    "integer": "-0123456789",
    "boolean": "tf",
    "array": "[",
    "object": "{",
    None: None,  # untyped: any JSON value
}


class SchemaGuide:
    """Compile a flat object schema (`type: object` + `properties` +
    `additionalProperties: false`) into a forced template: the literal
    structure `{"key1": <val>, "key2": <val>}` is teacher-forced
    character by character (keys in declaration order), and each value
    region runs a JsonPrefixValidator restricted to the property's
    declared type. This makes json_schema *enforced during decoding*
    for the common flat-object case, not just validated at the end.
    Nested/looser schemas fall back to the generic JSON grammar +
    end-of-request validation (GuidedJsonState)."""

    @staticmethod
    def compile(schema: dict) -> Optional["SchemaGuide"]:
        if not isinstance(schema, dict):
            return None
        props = schema.get("properties")
        if (schema.get("type") != "object" or not props
                or schema.get("additionalProperties") is not False):
            return None
        segments: List[tuple] = []
        first = True
        for key, sub in props.items():
            typ = sub.get("type") if isinstance(sub, dict) else None
            if typ not in _TYPE_STARTS:
                return None  # union/unknown type: generic fallback
            prefix = "{" if first else ", "
            segments.append(("lit", prefix + json.dumps(key) + ": "))
            segments.append(("val", typ))
            first = False
        segments.append(("lit", "}"))
        return SchemaGuide(segments)

    def __init__(self, segments: List[tuple]) -> None:
        self.segments = segments
        self.seg = 0
        self.lit_pos = 0
        self.subv: Optional[JsonPrefixValidator] = None
        self.sub_fresh = True

    def copy(self) -> "SchemaGuide":
        c = SchemaGuide.__new__(SchemaGuide)
        c.segments = self.segments
        c.seg = self.seg
        c.lit_pos = self.lit_pos
        c.subv = self.subv.copy() if self.subv is not None else None
        c.sub_fresh = self.sub_fresh
        return c

    @property
    def complete(self) -> bool:
        return self.seg >= len(self.segments)

    def feed(self, ch: str) -> bool:
        if self.complete:
            return ch in _WS
        kind, payload = self.segments[self.seg]
        if kind == "lit":
            if ch != payload[self.lit_pos]:
                return False
            self.lit_pos += 1
            if self.lit_pos == len(payload):
                self.seg += 1
                self.lit_pos = 0
                if (self.seg < len(self.segments)
                        and self.segments[self.seg][0] == "val"):
                    self.subv = JsonPrefixValidator()
                    self.sub_fresh = True
            return True
        # value region
        if self.sub_fresh:
            starts = _TYPE_STARTS[payload]
            if starts is not None and ch not in starts:
                return False
        if self.subv.feed(ch):
            self.sub_fresh = False
            return True
        if self.subv.complete:
            # the value ended; this character belongs to the next literal
            self.seg += 1
            self.subv = None
            return self.feed(ch)
        return False

    def feed_text(self, text: str) -> bool:
        for ch in text:
            if not self.feed(ch):
                return False
        return True

    def would_accept(self, text: str) -> Optional["SchemaGuide"]:
        c = self.copy()
        return c if c.feed_text(text) else None


# ---------------------------------------------------------------------------
# regex / choice guides (vLLM guided_regex / guided_choice analogues)


class RegexGuide:
    """Prefix acceptance via `regex` partial matching: a text is legal
    while it can still extend to a full match; complete once it fully
    matches (longest-generation semantics: EOS becomes legal at the
    first full match, other continuations stay legal while the pattern
    allows them)."""

    force_eos_on_complete = False  # longer matches may exist

    def __init__(self, pattern: str, text: str = "") -> None:
        import regex as _re

        self.pattern = pattern
        self._re = _re.compile(pattern)
        self.text = text

    def copy(self) -> "RegexGuide":
        c = RegexGuide.__new__(RegexGuide)
        c.pattern = self.pattern
        c._re = self._re
        c.text = self.text
        return c

    def feed_text(self, text: str) -> bool:
        cand = self.text + text
        m = self._re.fullmatch(cand, partial=True)
        if m is None:
            return False
        self.text = cand
        return True

    def would_accept(self, text: str) -> Optional["RegexGuide"]:
        c = self.copy()
        return c if c.feed_text(text) else None

    @property
    def complete(self) -> bool:
        m = self._re.fullmatch(self.text)
        return m is not None


def choice_regex(choices: List[str]) -> str:
    import regex as _re

    return "|".join(_re.escape(c) for c in choices)


# ---------------------------------------------------------------------------
# token-level guided state

def _token_texts(tokenizer, vocab_size: int) -> List[str]:
    # cached on the tokenizer instance (an id()-keyed global can alias
    # a freed tokenizer's address after GC)
    got = getattr(tokenizer, "_guided_token_texts", None)
    if got is None or len(got) < vocab_size:
        got = [None] * vocab_size
        tokenizer._guided_token_texts = got
    return got


class GuidedJsonState:
    """Per-request token-level wrapper around the prefix automaton."""

    def __init__(self, schema: Optional[dict] = None) -> None:
        # flat object schemas compile to an enforced template; anything
        # else uses the generic JSON grammar + end-of-request validation
        guide = SchemaGuide.compile(schema) if schema else None
        self.v = guide if guide is not None else JsonPrefixValidator()
        self.schema = schema
        self.consumed = 0  # output tokens already fed
        self.text = ""

    def advance(self, tokenizer, output_token_ids: List[int]) -> None:
        """Feed tokens committed since the last call."""
        for t in output_token_ids[self.consumed:]:
            if t < 0:
                break  # async placeholder not resolved yet
            txt = tokenizer.decode_token(int(t))
            self.v.feed_text(txt)
            self.text += txt
            self.consumed += 1

    def allowed_mask(self, tokenizer, logits_row, eos_token_id: int,
                     top_k: int = 64):
        """Return (allowed_token_ids, force_eos). Scans candidates from
        the top of the logits, widening until at least one legal token is
        found or the vocab is exhausted."""
        import torch

        if self.v.complete and getattr(self.v, "force_eos_on_complete",
                                       True):
            return [eos_token_id], True
        vocab = logits_row.shape[-1]
        texts = _token_texts(tokenizer, vocab)
        k = min(top_k, vocab)
        seen = 0
        while True:
            vals, idx = torch.topk(logits_row, k)
            cand = idx[seen:].tolist()
            allowed = []
            for t in cand:
                if t == eos_token_id:
                    continue
                txt = texts[t]
                if txt is None:
                    txt = tokenizer.decode_token(t)
                    texts[t] = txt
                if txt and self.v.would_accept(txt) is not None:
                    allowed.append(t)
            if self.v.complete and eos_token_id not in allowed:
                allowed.append(eos_token_id)
            if allowed:
                return allowed, False
            if k >= vocab:
                # nothing in the vocabulary extends the grammar (e.g. a
                # tokenizer with no JSON characters): allow EOS so the
                # request terminates instead of spinning
                return [eos_token_id], True
            seen = k
            k = min(k * 8, vocab)

    def finish_reason(self) -> Optional[str]:
        """Called at request end; returns an overriding finish_reason for
        schema-validation failure, else None."""
        if self.schema is None:
            return None
        try:
            obj = json.loads(self.text) if self.text.strip() else None
        except json.JSONDecodeError:
            return "error_json_schema"
        sch = self.schema or {}
        if sch.get("type") == "object" and not isinstance(obj, dict):
            return "error_json_schema"
        for req in sch.get("required", []):
            if not isinstance(obj, dict) or req not in obj:
                return "error_json_schema"
        return None


def guided_state_from_response_format(
    response_format: Optional[dict],
) -> Optional[GuidedJsonState]:
    """Map an OpenAI `response_format` body field (or the internal
    choice/regex forms `_params_from_body` builds from vLLM-style
    guided_choice / guided_regex) to a guided state."""
    if not response_format:
        return None
    kind = response_format.get("type")
    if kind == "json_object":
        return GuidedJsonState()
    if kind == "json_schema":
        js = response_format.get("json_schema") or {}
        return GuidedJsonState(schema=js.get("schema") or {})
    if kind == "regex" and response_format.get("pattern"):
        gs = GuidedJsonState()
        gs.v = RegexGuide(response_format["pattern"])
        return gs
    if kind == "choice" and response_format.get("choices"):
        gs = GuidedJsonState()
        gs.v = RegexGuide(choice_regex(
            [str(c) for c in response_format["choices"]]))
        gs.v.force_eos_on_complete = True  # a choice is a whole answer
        return gs
    return None
