"""Tokenizers for the engine.

Two implementations behind one interface:
  * SyntheticTokenizer — deterministic hash-based word tokenizer for the
    no-network environment (random-init weights make real BPE irrelevant);
    round-trips text -> ids -> distinct text, stable across processes so
    prefix-aware routing and prefix caching see consistent ids.
  * HFTokenizer — loads a real tokenizer.json from a local directory via the
    `tokenizers` library when one is provided.
"""

from __future__ import annotations

import hashlib
from typing import List, Optional


class BaseTokenizer:
    bos_token_id: int = 1
    eos_token_id: int = 2

    def encode(self, text: str) -> List[int]:
        raise NotImplementedError

    def decode(self, ids: List[int]) -> str:
        raise NotImplementedError

    def decode_token(self, token_id: int) -> str:
        return self.decode([token_id])

    def stream_decode(self, output_ids: List[int], state: dict) -> str:
        """Incremental detokenization: given the full output-id list so far
        and a per-sequence mutable state dict, return the newly decodable
        text suffix.  Single-token decode loses BPE word-boundary markers,
        so we decode a sliding window [prefix_offset:] and diff against the
        previously decoded [prefix_offset:read_offset] text (the vLLM
        scheme); a trailing U+FFFD means a multi-token UTF-8 sequence is
        still incomplete and we hold the delta until the next token.
        """
        p = state.get("p", 0)
        r = state.get("r", 0)
        new_text = self.decode(output_ids[p:])
        if new_text.endswith("�"):
            return ""
        prefix_text = self.decode(output_ids[p:r])
        state["p"] = r
        state["r"] = len(output_ids)
        return new_text[len(prefix_text):]


class SyntheticTokenizer(BaseTokenizer):
    """Stable word-level tokenizer: each whitespace word maps to an id by
    BLAKE2 hash into [16, vocab). ids 0-15 are reserved specials."""

    RESERVED = 16

    def __init__(self, vocab_size: int = 128256) -> None:
        self.vocab_size = vocab_size

    def encode(self, text: str) -> List[int]:
        ids = []
        for word in text.split():
            h = int.from_bytes(
                hashlib.blake2b(word.encode(), digest_size=4).digest(), "little"
            )
            ids.append(self.RESERVED + h % (self.vocab_size - self.RESERVED))
        return ids

    def decode(self, ids: List[int]) -> str:
        return " ".join(self.decode_token(i) for i in ids)

    def decode_token(self, token_id: int) -> str:
        if token_id == self.eos_token_id:
            return ""
        return f"w{token_id}"

    def stream_decode(self, output_ids: List[int], state: dict) -> str:
        # word-per-id: every id is independently decodable; keep the
        # historical trailing-space framing so accumulated deltas render
        # as space-separated words.
        r = state.get("r", 0)
        state["r"] = len(output_ids)
        return "".join(
            self.decode_token(t) + " " for t in output_ids[r:]
        )


class HFTokenizer(BaseTokenizer):
    def __init__(self, path: str) -> None:
        from tokenizers import Tokenizer

        import os

        f = path
        if os.path.isdir(path):
            f = os.path.join(path, "tokenizer.json")
        self.tk = Tokenizer.from_file(f)
        self.vocab_size = self.tk.get_vocab_size()

    def encode(self, text: str) -> List[int]:
        return self.tk.encode(text).ids

    def decode(self, ids: List[int]) -> str:
        return self.tk.decode(ids)


def get_tokenizer(spec: str, vocab_size: int) -> BaseTokenizer:
    if spec == "synthetic":
        return SyntheticTokenizer(vocab_size)
    return HFTokenizer(spec)


def render_chat(messages: List[dict], add_generation_prompt: bool = True,
                tools: Optional[List[dict]] = None) -> str:
    """Minimal chat template (role-tagged concatenation).

    With `tools`, function schemas are injected as a leading system block
    and the model is instructed to emit Hermes-style
    ``<tool_call>{"name": ..., "arguments": {...}}</tool_call>`` spans —
    the format server.py's parse_tool_calls() extracts into OpenAI
    `message.tool_calls` (the engine-side counterpart of vLLM's tool-call
    parsers the reference stack relies on; tutorial 13 in the reference).
    """
    import json as _json

    parts = []
    if tools:
        schemas = _json.dumps(
            [t.get("function", t) for t in tools], separators=(",", ":")
        )
        parts.append(
            "<|system|> You may call these tools: " + schemas +
            ' To call one, reply with <tool_call>{"name": <name>, '
            '"arguments": <args-object>}</tool_call>.'
        )
    for m in messages:
        content = m.get("content") or ""
        if isinstance(content, list):  # OpenAI content-part arrays
            content = " ".join(
                p.get("text", "") for p in content if isinstance(p, dict)
            )
        if m.get("tool_calls"):  # assistant turn that called tools
            calls = "".join(
                "<tool_call>" + _json.dumps({
                    "name": tc["function"]["name"],
                    "arguments": _json.loads(
                        tc["function"].get("arguments") or "{}"),
                }, separators=(",", ":")) + "</tool_call>"
                for tc in m["tool_calls"] if tc.get("function")
            )
            content = (content + " " + calls).strip()
        parts.append(f"<|{m.get('role', 'user')}|> {content}")
    if add_generation_prompt:
        parts.append("<|assistant|>")
    return "\n".join(parts)
