"""Golden-fixture translator conformance (the reference's translator test
corpus model: fixed inputs -> committed expected outputs, guarding
cross-refactor drift). Regenerate with:
    python tests/test_translator_goldens.py regenerate
"""

import json
import os
import pathlib
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


from aigw.filterapi.config import APISchemaName
from aigw.translator import get_translator

GOLDEN_PATH = pathlib.Path(__file__).parent / "goldens" / "translator_requests.json"

CHAT_REQ = {
    "model": "base-model",
    "messages": [
        {"role": "system", "content": "sys prompt"},
        {"role": "user", "content": "hello world"},
        {"role": "assistant", "content": "prev answer"},
        {"role": "user", "content": [
            {"type": "text", "text": "with image"},
            {"type": "image_url", "image_url": {"url": "data:image/png;base64,QUJD"}},
        ]},
    ],
    "max_tokens": 64,
    "temperature": 0.25,
    "top_p": 0.9,
    "stop": ["STOP"],
    "tools": [{"type": "function", "function": {
        "name": "lookup", "description": "d",
        "parameters": {"type": "object", "properties": {"q": {"type": "string"}}}}}],
    "tool_choice": "auto",
}

ANTHROPIC_REQ = {
    "model": "base-model",
    "max_tokens": 32,
    "system": "be terse",
    "messages": [{"role": "user", "content": "hi"}],
    "tools": [{"name": "t", "description": "d", "input_schema": {"type": "object"}}],
}

CASES = [
    ("/v1/chat/completions", s, CHAT_REQ)
    for s in ("OpenAI", "AWSBedrock", "AWSAnthropic", "AzureOpenAI",
              "GCPVertexAI", "GCPAnthropic", "Anthropic")
] + [
    ("/anthropic/v1/messages", s, ANTHROPIC_REQ)
    for s in ("Anthropic", "GCPAnthropic", "AWSAnthropic", "OpenAI", "AWSBedrock")
] + [
    ("/v1/embeddings", s, {"model": "base-model", "input": "embed me"})
    for s in ("OpenAI", "AzureOpenAI", "GCPVertexAI", "AWSBedrock")
] + [
    ("/v1/completions", s, {"model": "base-model", "prompt": "continue this",
                            "max_tokens": 8})
    for s in ("OpenAI", "AzureOpenAI")
]


def _translate_all():
    out = {}
    for endpoint, schema, req in CASES:
        t = get_translator(endpoint, APISchemaName(schema), api_version="2024-06-01",
                           gcp_project="proj", gcp_region="region")
        tr = t.request(json.loads(json.dumps(req)), model_override="golden-model",
                       stream=False)
        out[f"{endpoint}|{schema}"] = {
            "path": tr.path,
            "body": json.loads(tr.body),
            "headers": tr.headers,
        }
    return out


def test_request_translations_match_goldens():
    got = _translate_all()
    want = json.loads(GOLDEN_PATH.read_text())
    assert set(got) == set(want)
    for key in want:
        assert got[key] == want[key], f"translator output drifted for {key}"


if __name__ == "__main__" and "regenerate" in sys.argv:
    GOLDEN_PATH.parent.mkdir(exist_ok=True)
    GOLDEN_PATH.write_text(json.dumps(_translate_all(), indent=1, sort_keys=True))
    print(f"wrote {GOLDEN_PATH}")
