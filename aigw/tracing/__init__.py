from .tracing import (
    ChatSpanRecorder,
    InMemoryExporter,
    JSONLExporter,
    Span,
    Tracer,
    tracing_from_env,
)

__all__ = [
    "ChatSpanRecorder",
    "InMemoryExporter",
    "JSONLExporter",
    "Span",
    "Tracer",
    "tracing_from_env",
]
