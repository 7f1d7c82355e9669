from .state_sync import FastLimiterBridge, StateSync

__all__ = ["FastLimiterBridge", "StateSync"]
