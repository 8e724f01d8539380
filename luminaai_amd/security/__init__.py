from .auth import SecurityManager
from .input_validator import InputValidator
from .rate_limiter import RateLimiter, SecureConversationalChat

__all__ = ["InputValidator", "RateLimiter", "SecureConversationalChat",
           "SecurityManager"]
