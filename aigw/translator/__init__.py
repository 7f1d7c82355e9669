"""Translator registry — importing this package registers the full
endpoint × provider-schema matrix."""

from aigw.translator.base import (
    RequestTranslation,
    ResponseTranslation,
    TranslationError,
    Translator,
    Usage,
    get_translator,
    supported_matrix,
)

# Register translators (import side effects populate the registry).
from aigw.translator import openai_passthrough  # noqa: F401,E402
from aigw.translator import chat_anthropic  # noqa: F401,E402
from aigw.translator import chat_bedrock  # noqa: F401,E402
from aigw.translator import chat_gcp  # noqa: F401,E402
from aigw.translator import misc_endpoints  # noqa: F401,E402

__all__ = [
    "RequestTranslation",
    "ResponseTranslation",
    "TranslationError",
    "Translator",
    "Usage",
    "get_translator",
    "supported_matrix",
]
