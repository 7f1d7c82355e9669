"""Sensitive-content redaction for debug logs.

Parity with internal/redaction/redaction.go:29-60: values become
``[REDACTED LENGTH=n HASH=xxxx]`` placeholders (SHA-256-based content hash
so equal values are correlatable without being readable); the JSON-tree
redactor replaces every string/number leaf but PRESERVES ``"type"``
discriminators so redacted payload shapes stay debuggable.
"""

from __future__ import annotations

import hashlib
from typing import Union

_PRESERVED_KEYS = {"type", "role", "object", "event"}


def redact_string(value: Union[str, bytes]) -> str:
    data = value.encode("utf-8") if isinstance(value, str) else value
    digest = hashlib.sha256(data).hexdigest()[:8]
    return f"[REDACTED LENGTH={len(data)} HASH={digest}]"


def redact_json_tree(node, *, _key: str = ""):
    """Return a redacted deep copy of a parsed JSON tree."""
    if isinstance(node, dict):
        return {k: (v if k in _PRESERVED_KEYS and isinstance(v, str)
                    else redact_json_tree(v, _key=k)) for k, v in node.items()}
    if isinstance(node, list):
        return [redact_json_tree(v, _key=_key) for v in node]
    if isinstance(node, str):
        return redact_string(node)
    if isinstance(node, (int, float)) and not isinstance(node, bool):
        return 0
    return node


def redact_headers(headers: dict[str, str]) -> dict[str, str]:
    """Credential headers are ALWAYS redacted in logs (server.go:474-516)."""
    sensitive = {"authorization", "x-api-key", "api-key", "proxy-authorization",
                 "cookie", "set-cookie", "x-ai-eg-api-key", "x-aigw-aws-access-key-id", "x-aigw-aws-secret-access-key", "x-aigw-aws-session-token"}
    return {
        k: (redact_string(v) if k.lower() in sensitive else v)
        for k, v in headers.items()
    }
