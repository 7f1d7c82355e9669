from .state_sync import StateSync

__all__ = ["StateSync"]
