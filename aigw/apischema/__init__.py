"""Typed API schemas (pydantic discriminated unions) — the idiomatic
Python equivalent of the reference's hand-written Go union types
(internal/apischema/, ~12k LoC). See openai.py / anthropic.py."""

from aigw.apischema import openai as _openai
from aigw.apischema.anthropic import CountTokensRequest, MessagesRequest
from aigw.apischema.openai import (
    ChatCompletionRequest,
    ChatCompletionResponse,
    CompletionRequest,
    EmbeddingsRequest,
    ErrorEnvelope,
    SchemaError,
    validate_response,
)
from aigw.apischema.openai import SchemaError as _SchemaError


def validate_request(path: str, body: dict):
    """Typed validation across BOTH API families: OpenAI paths go through
    the openai models, /anthropic/* through the Anthropic models."""
    try:
        if path == "/anthropic/v1/messages":
            return MessagesRequest.model_validate(body)
        if path == "/anthropic/v1/messages/count_tokens":
            return CountTokensRequest.model_validate(body)
    except Exception as e:
        if hasattr(e, "errors"):
            raise _SchemaError(_openai._summarize(e)) from e
        raise _SchemaError(str(e)) from e
    return _openai.validate_request(path, body)

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
