"""Input validation and sanitisation for the chat/serving layer
(rebuild of /root/reference/Src/Main_Scripts/security/input_validator.py:
17-198: suspicious-pattern detection, HTML sanitisation, length limits)."""

from __future__ import annotations

import html
import re
from typing import Dict

MAX_INPUT_CHARS = 8192

SUSPICIOUS_PATTERNS = [
    (re.compile(r"<\s*script", re.I), "script tag"),
    (re.compile(r"javascript\s*:", re.I), "javascript: URI"),
    (re.compile(r"on(click|error|load|mouseover)\s*=", re.I), "inline handler"),
    (re.compile(r"(\bunion\b.{0,40}\bselect\b)|(;\s*drop\s+table)", re.I),
     "sql injection"),
    (re.compile(r"\.\./\.\./"), "path traversal"),
    (re.compile(r"[\x00-\x08\x0b\x0c\x0e-\x1f]"), "control characters"),
]


class InputValidator:
    def __init__(self, max_chars: int = MAX_INPUT_CHARS):
        self.max_chars = max_chars

    def validate(self, text: str) -> Dict:
        if not isinstance(text, str):
            return {"ok": False, "reason": "not a string"}
        if not text.strip():
            return {"ok": False, "reason": "empty"}
        if len(text) > self.max_chars:
            return {"ok": False,
                    "reason": f"too long ({len(text)} > {self.max_chars})"}
        for pat, label in SUSPICIOUS_PATTERNS:
            if pat.search(text):
                return {"ok": False, "reason": f"suspicious pattern: {label}"}
        return {"ok": True, "reason": None}

    def sanitize(self, text: str) -> str:
        text = html.escape(text, quote=False)
        return re.sub(r"[\x00-\x08\x0b\x0c\x0e-\x1f]", "", text).strip()
