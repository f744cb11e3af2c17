"""PII detection on request bodies (feature-gated).

Parity: reference experimental/pii/ — regex and (optional) Presidio
analyzers behind one interface, plus the request-scanning hook used by the
request service before proxying. Actions: block (400) or redact in place.
"""

from __future__ import annotations

import logging
import re
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional, Tuple

logger = logging.getLogger("router.pii")


@dataclass
class PIIMatch:
    entity_type: str
    start: int
    end: int
    text: str


PATTERNS = {
    "EMAIL_ADDRESS": re.compile(
        r"\b[A-Za-z0-9._%+-]+@[A-Za-z0-9.-]+\.[A-Za-z]{2,}\b"
    ),
    "PHONE_NUMBER": re.compile(
        r"(?<!\d)(\+?\d{1,2}[\s.-]?)?(\(\d{3}\)|\d{3})[\s.-]?\d{3}[\s.-]?"
        r"\d{4}(?!\d)"
    ),
    "US_SSN": re.compile(r"(?<!\d)\d{3}-\d{2}-\d{4}(?!\d)"),
    "CREDIT_CARD": re.compile(r"(?<!\d)(?:\d[ -]?){13,16}(?!\d)"),
    "IP_ADDRESS": re.compile(
        r"(?<!\d)(?:\d{1,3}\.){3}\d{1,3}(?!\d)"
    ),
    "API_KEY": re.compile(r"\b(sk|pk|rk)-[A-Za-z0-9]{16,}\b"),
}


class RegexPIIAnalyzer:
    def __init__(self, entities: Optional[List[str]] = None) -> None:
        self.entities = entities or list(PATTERNS)

    def analyze(self, text: str) -> List[PIIMatch]:
        out: List[PIIMatch] = []
        for ent in self.entities:
            pat = PATTERNS.get(ent)
            if pat is None:
                continue
            for m in pat.finditer(text):
                out.append(PIIMatch(ent, m.start(), m.end(), m.group()))
        return out


class PresidioPIIAnalyzer:  # pragma: no cover - optional dependency
    def __init__(self, entities: Optional[List[str]] = None) -> None:
        from presidio_analyzer import AnalyzerEngine

        self.engine = AnalyzerEngine()
        self.entities = entities

    def analyze(self, text: str) -> List[PIIMatch]:
        results = self.engine.analyze(
            text=text, language="en", entities=self.entities
        )
        return [
            PIIMatch(r.entity_type, r.start, r.end, text[r.start : r.end])
            for r in results
        ]


def create_analyzer(kind: str = "regex", entities=None):
    if kind == "presidio":  # pragma: no cover
        return PresidioPIIAnalyzer(entities)
    return RegexPIIAnalyzer(entities)


def redact(text: str, matches: List[PIIMatch]) -> str:
    out = text
    for m in sorted(matches, key=lambda x: -x.start):
        out = out[: m.start] + f"[{m.entity_type}]" + out[m.end :]
    return out


def scan_request_body(
    body: Dict[str, Any],
    analyzer,
    action: str = "block",
) -> Tuple[bool, Dict[str, Any], List[PIIMatch]]:
    """Scan prompt/messages text. Returns (allowed, possibly-redacted body,
    matches)."""
    texts: List[Tuple[str, Any]] = []
    if isinstance(body.get("prompt"), str):
        texts.append(("prompt", None))
    for i, msg in enumerate(body.get("messages") or []):
        if isinstance(msg.get("content"), str):
            texts.append(("messages", i))
    all_matches: List[PIIMatch] = []
    new_body = dict(body)
    for kind, idx in texts:
        text = (
            new_body["prompt"]
            if kind == "prompt"
            else new_body["messages"][idx]["content"]
        )
        matches = analyzer.analyze(text)
        if not matches:
            continue
        all_matches.extend(matches)
        if action == "redact":
            red = redact(text, matches)
            if kind == "prompt":
                new_body["prompt"] = red
            else:
                msgs = [dict(m) for m in new_body["messages"]]
                msgs[idx]["content"] = red
                new_body["messages"] = msgs
    if all_matches and action == "block":
        return False, body, all_matches
    return True, new_body, all_matches
