from .limiter import RateLimiter, RateLimitDecision

__all__ = ["RateLimiter", "RateLimitDecision"]
