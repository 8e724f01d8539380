"""Serving-side auth: PBKDF2 password hashing + HMAC-signed session tokens
+ lockout.

Rebuild of /root/reference/Src/Main_Scripts/security/auth.py:33-266. The
reference used PyJWT; this environment has no PyJWT wheel, so sessions are
stdlib HMAC-SHA256 signed tokens with the same claims (sub/iat/exp) and API.
"""

from __future__ import annotations

import base64
import hashlib
import hmac
import json
import os
import secrets
import time
from typing import Dict, Optional

PBKDF2_ITERATIONS = 200_000
LOCKOUT_THRESHOLD = 5
LOCKOUT_SECONDS = 300
SESSION_TTL = 3600


class SecurityManager:
    def __init__(self, secret_key: Optional[str] = None,
                 session_ttl: int = SESSION_TTL):
        self.secret = (secret_key or secrets.token_hex(32)).encode()
        self.session_ttl = session_ttl
        self.users: Dict[str, Dict] = {}
        self.failed: Dict[str, Dict] = {}

    # ---------------------------------------------------------- passwords
    def register_user(self, username: str, password: str) -> bool:
        if username in self.users or len(password) < 8:
            return False
        salt = os.urandom(16)
        dk = hashlib.pbkdf2_hmac("sha256", password.encode(), salt,
                                 PBKDF2_ITERATIONS)
        self.users[username] = {"salt": salt, "hash": dk,
                                "created": time.time()}
        return True

    def _check_password(self, username: str, password: str) -> bool:
        u = self.users.get(username)
        if u is None:
            return False
        dk = hashlib.pbkdf2_hmac("sha256", password.encode(), u["salt"],
                                 PBKDF2_ITERATIONS)
        return hmac.compare_digest(dk, u["hash"])

    def is_locked_out(self, username: str) -> bool:
        f = self.failed.get(username)
        if not f:
            return False
        if f["count"] >= LOCKOUT_THRESHOLD and \
                time.time() - f["last"] < LOCKOUT_SECONDS:
            return True
        if time.time() - f["last"] >= LOCKOUT_SECONDS:
            del self.failed[username]
        return False

    # ---------------------------------------------------------- sessions
    def authenticate(self, username: str, password: str) -> Optional[str]:
        """Returns a signed session token, or None."""
        if self.is_locked_out(username):
            return None
        if not self._check_password(username, password):
            f = self.failed.setdefault(username, {"count": 0, "last": 0.0})
            f["count"] += 1
            f["last"] = time.time()
            return None
        self.failed.pop(username, None)
        now = int(time.time())
        claims = {"sub": username, "iat": now, "exp": now + self.session_ttl}
        body = base64.urlsafe_b64encode(
            json.dumps(claims).encode()).rstrip(b"=")
        sig = base64.urlsafe_b64encode(
            hmac.new(self.secret, body, hashlib.sha256).digest()).rstrip(b"=")
        return (body + b"." + sig).decode()

    def validate_session(self, token: str) -> Optional[str]:
        """Returns the username if the token is valid and unexpired."""
        try:
            body_s, sig_s = token.split(".")
            body = body_s.encode()
            want = base64.urlsafe_b64encode(
                hmac.new(self.secret, body, hashlib.sha256).digest()).rstrip(b"=")
            if not hmac.compare_digest(want.decode(), sig_s):
                return None
            claims = json.loads(base64.urlsafe_b64decode(body + b"=="))
            if claims.get("exp", 0) < time.time():
                return None
            return claims.get("sub")
        except (ValueError, json.JSONDecodeError):
            return None
