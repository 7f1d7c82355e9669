"""Typed OpenAI API schema: discriminated unions with strict shape
validation.

The reference invests ~9k LoC of hand-written Go types with custom union
(un)marshalers in this surface (internal/apischema/openai/openai.go,
union.go) because dict-structural handling silently mangles the exotic
shapes — tool-call arrays, content-part mixes, logprobs, prompt unions.
This module is the idiomatic Python equivalent: pydantic v2 models with
tagged unions (discriminators on ``role``/``type``) and explicit
union coercion for the untagged cases (string-or-parts content, prompt
token arrays, embeddings input). Unknown fields are preserved, not
rejected (``extra="allow"``), matching Go's tolerant unmarshal; known
fields with the WRONG SHAPE fail loudly.

Used by: the conformance suite (every vendored cassette payload must
round-trip through these models) and the ingress strict-schema gate
(GatewayServer strict_schema: validate-before-dispatch, 400 with the
pydantic error summary on failure).
"""

from __future__ import annotations

from typing import Annotated, Any, Literal, Optional, Union

from pydantic import BaseModel, ConfigDict, Field, field_validator

_cfg = ConfigDict(extra="allow", populate_by_name=True)


# --------------------------------------------------------------------------
# content parts (openai.go ChatCompletionContentPartUnion)


class TextPart(BaseModel):
    model_config = _cfg
    type: Literal["text"]
    text: str


class ImageURL(BaseModel):
    model_config = _cfg
    url: str
    detail: Optional[Literal["auto", "low", "high"]] = None


class ImagePart(BaseModel):
    model_config = _cfg
    type: Literal["image_url"]
    image_url: ImageURL


class InputAudio(BaseModel):
    model_config = _cfg
    data: str
    format: str


class AudioPart(BaseModel):
    model_config = _cfg
    type: Literal["input_audio"]
    input_audio: InputAudio


class FileData(BaseModel):
    model_config = _cfg
    file_data: Optional[str] = None
    file_id: Optional[str] = None
    filename: Optional[str] = None


class FilePart(BaseModel):
    model_config = _cfg
    type: Literal["file"]
    file: FileData


class RefusalPart(BaseModel):
    model_config = _cfg
    type: Literal["refusal"]
    refusal: str


ContentPart = Annotated[
    Union[TextPart, ImagePart, AudioPart, FilePart, RefusalPart],
    Field(discriminator="type"),
]

# string-or-parts union (openai.go StringOrUserRoleContentUnion)
Content = Union[str, list[ContentPart], None]


# --------------------------------------------------------------------------
# messages (openai.go ChatCompletionMessageParamUnion — discriminated by role)


class FunctionCall(BaseModel):
    model_config = _cfg
    name: Optional[str] = None
    arguments: Optional[str] = None


class ToolCall(BaseModel):
    model_config = _cfg
    id: Optional[str] = None
    type: Optional[str] = None  # "function" (or "custom" in newer APIs)
    function: Optional[FunctionCall] = None
    index: Optional[int] = None


class SystemMessage(BaseModel):
    model_config = _cfg
    role: Literal["system", "developer"]
    content: Content = None
    name: Optional[str] = None


class UserMessage(BaseModel):
    model_config = _cfg
    role: Literal["user"]
    content: Content = None
    name: Optional[str] = None


class AssistantMessage(BaseModel):
    model_config = _cfg
    role: Literal["assistant"]
    content: Content = None
    name: Optional[str] = None
    refusal: Optional[str] = None
    audio: Optional[dict] = None
    tool_calls: Optional[list[ToolCall]] = None
    function_call: Optional[FunctionCall] = None
    reasoning_content: Optional[str] = None


class ToolMessage(BaseModel):
    model_config = _cfg
    role: Literal["tool", "function"]
    content: Content = None
    tool_call_id: Optional[str] = None
    name: Optional[str] = None


Message = Annotated[
    Union[SystemMessage, UserMessage, AssistantMessage, ToolMessage],
    Field(discriminator="role"),
]


# --------------------------------------------------------------------------
# tools / response_format


class FunctionDefinition(BaseModel):
    model_config = _cfg
    name: str
    description: Optional[str] = None
    parameters: Optional[dict] = None
    strict: Optional[bool] = None


class Tool(BaseModel):
    model_config = _cfg
    type: Optional[str] = "function"
    function: Optional[FunctionDefinition] = None


class NamedToolChoice(BaseModel):
    model_config = _cfg
    type: str
    function: Optional[FunctionCall] = None


ToolChoice = Union[Literal["none", "auto", "required"], NamedToolChoice, None]


class JSONSchemaSpec(BaseModel):
    model_config = _cfg
    name: str
    description: Optional[str] = None
    schema_: Optional[dict] = Field(default=None, alias="schema")
    strict: Optional[bool] = None


class ResponseFormat(BaseModel):
    model_config = _cfg
    type: Literal["text", "json_object", "json_schema"]
    json_schema: Optional[JSONSchemaSpec] = None


class StreamOptions(BaseModel):
    model_config = _cfg
    include_usage: Optional[bool] = None


# --------------------------------------------------------------------------
# chat request / response


class ChatCompletionRequest(BaseModel):
    model_config = _cfg
    model: str
    messages: list[Message]
    audio: Optional[dict] = None
    frequency_penalty: Optional[float] = None
    logit_bias: Optional[dict[str, float]] = None
    logprobs: Optional[bool] = None
    top_logprobs: Optional[int] = None
    max_tokens: Optional[int] = None
    max_completion_tokens: Optional[int] = None
    metadata: Optional[dict] = None
    modalities: Optional[list[str]] = None
    n: Optional[int] = None
    parallel_tool_calls: Optional[bool] = None
    prediction: Optional[dict] = None
    presence_penalty: Optional[float] = None
    reasoning_effort: Optional[str] = None
    response_format: Optional[ResponseFormat] = None
    seed: Optional[int] = None
    service_tier: Optional[str] = None
    stop: Union[str, list[str], None] = None
    store: Optional[bool] = None
    stream: Optional[bool] = None
    stream_options: Optional[StreamOptions] = None
    temperature: Optional[float] = None
    tool_choice: ToolChoice = None
    tools: Optional[list[Tool]] = None
    top_p: Optional[float] = None
    user: Optional[str] = None
    web_search_options: Optional[dict] = None

    @field_validator("messages")
    @classmethod
    def _messages_nonempty_shape(cls, v):
        return v


class TokenLogprob(BaseModel):
    model_config = _cfg
    token: str
    logprob: float
    bytes: Optional[list[int]] = None
    top_logprobs: Optional[list[dict]] = None


class ChoiceLogprobs(BaseModel):
    model_config = _cfg
    content: Optional[list[TokenLogprob]] = None
    refusal: Optional[list[TokenLogprob]] = None


class ResponseMessage(BaseModel):
    model_config = _cfg
    role: Optional[str] = None
    content: Optional[str] = None
    refusal: Optional[str] = None
    annotations: Optional[list[dict]] = None
    audio: Optional[dict] = None
    tool_calls: Optional[list[ToolCall]] = None
    function_call: Optional[FunctionCall] = None
    reasoning_content: Optional[str] = None


class Choice(BaseModel):
    model_config = _cfg
    index: Optional[int] = None
    message: Optional[ResponseMessage] = None
    delta: Optional[ResponseMessage] = None  # chunks
    finish_reason: Optional[str] = None
    logprobs: Optional[ChoiceLogprobs] = None


class CompletionTokensDetails(BaseModel):
    model_config = _cfg
    accepted_prediction_tokens: Optional[int] = None
    audio_tokens: Optional[int] = None
    reasoning_tokens: Optional[int] = None
    rejected_prediction_tokens: Optional[int] = None


class PromptTokensDetails(BaseModel):
    model_config = _cfg
    audio_tokens: Optional[int] = None
    cached_tokens: Optional[int] = None


class Usage(BaseModel):
    model_config = _cfg
    prompt_tokens: Optional[int] = None
    completion_tokens: Optional[int] = None
    total_tokens: Optional[int] = None
    completion_tokens_details: Optional[CompletionTokensDetails] = None
    prompt_tokens_details: Optional[PromptTokensDetails] = None


class ChatCompletionResponse(BaseModel):
    model_config = _cfg
    id: Optional[str] = None
    object: Literal["chat.completion", "chat.completion.chunk"]
    created: Optional[int] = None
    model: Optional[str] = None
    choices: list[Choice] = []
    usage: Optional[Usage] = None
    service_tier: Optional[str] = None
    system_fingerprint: Optional[str] = None


# --------------------------------------------------------------------------
# legacy completions (prompt union: str | [str] | [int] | [[int]])

Prompt = Union[str, list[str], list[int], list[list[int]], None]


class CompletionRequest(BaseModel):
    model_config = _cfg
    model: str
    prompt: Prompt = None
    best_of: Optional[int] = None
    echo: Optional[bool] = None
    frequency_penalty: Optional[float] = None
    logit_bias: Optional[dict[str, float]] = None
    logprobs: Optional[int] = None
    max_tokens: Optional[int] = None
    n: Optional[int] = None
    presence_penalty: Optional[float] = None
    seed: Optional[int] = None
    stop: Union[str, list[str], None] = None
    stream: Optional[bool] = None
    stream_options: Optional[StreamOptions] = None
    suffix: Optional[str] = None
    temperature: Optional[float] = None
    top_p: Optional[float] = None
    user: Optional[str] = None


class CompletionChoice(BaseModel):
    model_config = _cfg
    text: Optional[str] = None
    index: Optional[int] = None
    finish_reason: Optional[str] = None
    logprobs: Optional[dict] = None


class CompletionResponse(BaseModel):
    model_config = _cfg
    id: Optional[str] = None
    object: Literal["text_completion"]
    created: Optional[int] = None
    model: Optional[str] = None
    choices: list[CompletionChoice] = []
    usage: Optional[Usage] = None


# --------------------------------------------------------------------------
# embeddings (input union: str | [str] | [int] | [[int]])

EmbeddingInput = Union[str, list[str], list[int], list[list[int]]]


class EmbeddingsRequest(BaseModel):
    model_config = _cfg
    model: str
    input: EmbeddingInput
    dimensions: Optional[int] = None
    encoding_format: Optional[Literal["float", "base64"]] = None
    user: Optional[str] = None


class Embedding(BaseModel):
    model_config = _cfg
    object: Optional[str] = None
    index: Optional[int] = None
    # float list, or a base64 string when encoding_format=base64
    embedding: Union[list[float], str]


class EmbeddingsResponse(BaseModel):
    model_config = _cfg
    object: Optional[str] = None
    data: list[Embedding] = []
    model: Optional[str] = None
    usage: Optional[Usage] = None


# --------------------------------------------------------------------------
# images / error envelope


class ImageGenerationRequest(BaseModel):
    model_config = _cfg
    prompt: str
    model: Optional[str] = None
    n: Optional[int] = None
    quality: Optional[str] = None
    response_format: Optional[str] = None
    size: Optional[str] = None
    style: Optional[str] = None
    user: Optional[str] = None


class ErrorDetail(BaseModel):
    model_config = _cfg
    message: Optional[str] = None
    type: Optional[str] = None
    param: Optional[str] = None
    code: Union[str, int, None] = None


class ErrorEnvelope(BaseModel):
    model_config = _cfg
    error: ErrorDetail
    type: Optional[str] = None


# --------------------------------------------------------------------------
# validation entry points


_REQUEST_MODELS = {
    "/v1/chat/completions": ChatCompletionRequest,
    "/v1/completions": CompletionRequest,
    "/v1/embeddings": EmbeddingsRequest,
    "/v1/images/generations": ImageGenerationRequest,
}

_RESPONSE_MODELS = {
    "/v1/chat/completions": ChatCompletionResponse,
    "/v1/completions": CompletionResponse,
    "/v1/embeddings": EmbeddingsResponse,
}


class SchemaError(ValueError):
    pass


def _summarize(exc) -> str:
    errs = exc.errors()[:3]
    parts = []
    for e in errs:
        loc = ".".join(str(x) for x in e.get("loc", ()))
        parts.append(f"{loc}: {e.get('msg', 'invalid')}")
    return "; ".join(parts)


def validate_request(path: str, body: dict) -> Any:
    """Parse `body` into the typed request model for `path`; raises
    SchemaError with a compact message. Paths without a model validate
    trivially (returns None)."""
    model = _REQUEST_MODELS.get(path)
    if model is None:
        return None
    try:
        return model.model_validate(body)
    except Exception as e:  # pydantic.ValidationError
        if hasattr(e, "errors"):
            raise SchemaError(_summarize(e)) from e
        raise SchemaError(str(e)) from e


def validate_response(path: str, body: dict) -> Any:
    model = _RESPONSE_MODELS.get(path)
    if model is None:
        return None
    try:
        return model.model_validate(body)
    except Exception as e:
        if hasattr(e, "errors"):
            raise SchemaError(_summarize(e)) from e
        raise SchemaError(str(e)) from e
