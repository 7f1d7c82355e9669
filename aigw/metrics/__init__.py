from .genai import GenAIMetrics

__all__ = ["GenAIMetrics"]
