"""Typed API schemas (pydantic discriminated unions) — the idiomatic
Python equivalent of the reference's hand-written Go union types
(internal/apischema/, ~12k LoC). See openai.py / anthropic.py."""

from aigw.apischema.openai import (
    ChatCompletionRequest,
    ChatCompletionResponse,
    CompletionRequest,
    EmbeddingsRequest,
    ErrorEnvelope,
    SchemaError,
    validate_request,
    validate_response,
)

__all__ = [
    "ChatCompletionRequest",
    "ChatCompletionResponse",
    "CompletionRequest",
    "EmbeddingsRequest",
    "ErrorEnvelope",
    "SchemaError",
    "validate_request",
    "validate_response",
]
