"""Sliding-window rate limiting per action
(rebuild of /root/reference/Src/Main_Scripts/security/rate_limiter.py:8-332:
per-action limits :16-22, SecureConversationalChat wrapper :107)."""

from __future__ import annotations

import threading
import time
from collections import deque
from typing import Dict, Optional

DEFAULT_LIMITS = {
    "message": (10, 60.0),      # 10 messages / minute
    "login": (5, 60.0),
    "generate": (20, 60.0),
    "save": (4, 60.0),
}


class RateLimiter:
    def __init__(self, limits: Optional[Dict] = None):
        self.limits = dict(DEFAULT_LIMITS, **(limits or {}))
        self._events: Dict[tuple, deque] = {}
        self._lock = threading.RLock()

    def allow(self, user: str, action: str = "message") -> bool:
        limit, window = self.limits.get(action, (10, 60.0))
        now = time.time()
        with self._lock:
            q = self._events.setdefault((user, action), deque())
            while q and now - q[0] > window:
                q.popleft()
            if len(q) >= limit:
                return False
            q.append(now)
            return True

    def remaining(self, user: str, action: str = "message") -> int:
        limit, window = self.limits.get(action, (10, 60.0))
        now = time.time()
        with self._lock:
            q = self._events.get((user, action), deque())
            live = sum(1 for t in q if now - t <= window)
            return max(0, limit - live)

    def reset(self, user: Optional[str] = None):
        with self._lock:
            if user is None:
                self._events.clear()
            else:
                for k in list(self._events):
                    if k[0] == user:
                        del self._events[k]


class SecureConversationalChat:
    """Wraps a ChatInterface with rate limiting + input validation
    (reference rate_limiter.py:107-332)."""

    def __init__(self, chat, user: str = "default",
                 rate_limiter: Optional[RateLimiter] = None,
                 validator=None):
        from .input_validator import InputValidator
        self.chat = chat
        self.user = user
        self.limiter = rate_limiter or RateLimiter()
        self.validator = validator or InputValidator()

    def respond(self, text: str) -> str:
        if not self.limiter.allow(self.user, "message"):
            return "[rate limited — try again shortly]"
        verdict = self.validator.validate(text)
        if not verdict["ok"]:
            return f"[input rejected: {verdict['reason']}]"
        return self.chat.respond(self.validator.sanitize(text))
