"""Redaction tests (parity: internal/redaction/redaction_test.go)."""

from aigw.utils.redaction import redact_headers, redact_json_tree, redact_string


def test_redact_string_format_and_stability():
    r1 = redact_string("secret-value")
    r2 = redact_string("secret-value")
    r3 = redact_string("other")
    assert r1 == r2  # same content -> same hash (correlatable)
    assert r1 != r3
    assert r1.startswith("[REDACTED LENGTH=12 HASH=")


def test_redact_json_tree_preserves_discriminators():
    doc = {
        "model": "gpt-4o",
        "messages": [
            {"role": "user", "content": "secret prompt",
             "parts": [{"type": "text", "text": "hidden"}]}
        ],
        "max_tokens": 100,
        "stream": True,
    }
    red = redact_json_tree(doc)
    assert red["messages"][0]["role"] == "user"  # discriminator preserved
    assert red["messages"][0]["parts"][0]["type"] == "text"
    assert red["messages"][0]["content"].startswith("[REDACTED")
    assert red["messages"][0]["parts"][0]["text"].startswith("[REDACTED")
    assert red["max_tokens"] == 0
    assert red["stream"] is True
    # original untouched
    assert doc["messages"][0]["content"] == "secret prompt"


def test_redact_headers():
    h = redact_headers({"Authorization": "Bearer sk-123", "accept": "json"})
    assert h["Authorization"].startswith("[REDACTED")
    assert h["accept"] == "json"
