"""Typed Anthropic Messages API schema (discriminated unions).

Parity target: internal/apischema/anthropic/ (1.8k LoC of Go types).
Same design rules as aigw.apischema.openai: tagged unions on ``type``,
string-or-blocks coercion for system/content, unknown fields preserved,
wrong shapes rejected loudly.
"""

from __future__ import annotations

from typing import Annotated, Literal, Optional, Union

from pydantic import BaseModel, ConfigDict, Field

_cfg = ConfigDict(extra="allow", populate_by_name=True)


class TextBlock(BaseModel):
    model_config = _cfg
    type: Literal["text"]
    text: str
    cache_control: Optional[dict] = None
    citations: Optional[list[dict]] = None


class ImageSource(BaseModel):
    model_config = _cfg
    type: str  # base64 | url
    media_type: Optional[str] = None
    data: Optional[str] = None
    url: Optional[str] = None


class ImageBlock(BaseModel):
    model_config = _cfg
    type: Literal["image"]
    source: ImageSource
    cache_control: Optional[dict] = None


class ToolUseBlock(BaseModel):
    model_config = _cfg
    type: Literal["tool_use"]
    id: str
    name: str
    input: dict = {}
    cache_control: Optional[dict] = None


class ToolResultBlock(BaseModel):
    model_config = _cfg
    type: Literal["tool_result"]
    tool_use_id: str
    content: Union[str, list[dict], None] = None
    is_error: Optional[bool] = None
    cache_control: Optional[dict] = None


class ThinkingBlock(BaseModel):
    model_config = _cfg
    type: Literal["thinking"]
    thinking: str
    signature: Optional[str] = None


class RedactedThinkingBlock(BaseModel):
    model_config = _cfg
    type: Literal["redacted_thinking"]
    data: str


class DocumentBlock(BaseModel):
    model_config = _cfg
    type: Literal["document"]
    source: dict
    cache_control: Optional[dict] = None
    citations: Optional[dict] = None
    context: Optional[str] = None
    title: Optional[str] = None


ContentBlock = Annotated[
    Union[TextBlock, ImageBlock, ToolUseBlock, ToolResultBlock,
          ThinkingBlock, RedactedThinkingBlock, DocumentBlock],
    Field(discriminator="type"),
]

MessageContent = Union[str, list[ContentBlock]]


class MessageParam(BaseModel):
    model_config = _cfg
    role: Literal["user", "assistant"]
    content: MessageContent


class ToolDefinition(BaseModel):
    model_config = _cfg
    name: str
    description: Optional[str] = None
    input_schema: Optional[dict] = None
    type: Optional[str] = None
    cache_control: Optional[dict] = None


class ToolChoice(BaseModel):
    model_config = _cfg
    type: str  # auto | any | tool | none
    name: Optional[str] = None
    disable_parallel_tool_use: Optional[bool] = None


class Thinking(BaseModel):
    model_config = _cfg
    type: str  # enabled | disabled
    budget_tokens: Optional[int] = None


class MessagesRequest(BaseModel):
    model_config = _cfg
    model: Optional[str] = None  # absent on vendor paths (model in URL)
    messages: list[MessageParam]
    max_tokens: Optional[int] = None
    system: Union[str, list[ContentBlock], None] = None
    metadata: Optional[dict] = None
    stop_sequences: Optional[list[str]] = None
    stream: Optional[bool] = None
    temperature: Optional[float] = None
    thinking: Optional[Thinking] = None
    tool_choice: Optional[ToolChoice] = None
    tools: Optional[list[ToolDefinition]] = None
    top_k: Optional[int] = None
    top_p: Optional[float] = None
    anthropic_version: Optional[str] = None


class AnthropicUsage(BaseModel):
    model_config = _cfg
    input_tokens: Optional[int] = None
    output_tokens: Optional[int] = None
    cache_creation_input_tokens: Optional[int] = None
    cache_read_input_tokens: Optional[int] = None


class MessagesResponse(BaseModel):
    model_config = _cfg
    id: Optional[str] = None
    type: Literal["message"]
    role: Literal["assistant"]
    content: list[ContentBlock] = []
    model: Optional[str] = None
    stop_reason: Optional[str] = None
    stop_sequence: Optional[str] = None
    usage: Optional[AnthropicUsage] = None


class CountTokensRequest(BaseModel):
    model_config = _cfg
    model: Optional[str] = None
    messages: list[MessageParam]
    system: Union[str, list[ContentBlock], None] = None
    tools: Optional[list[ToolDefinition]] = None
    tool_choice: Optional[ToolChoice] = None
    thinking: Optional[Thinking] = None
