"""GenAI metric semantics (§A.2): token-type breakdown, error_type
labeling, TTFT/ITL formulas, provider derivation, Prometheus exposition."""

from aigw.metrics import GenAIMetrics
from aigw.metrics.genai import provider_from_schema
from aigw.translator import Usage


def _sample(rendered: bytes, needle: str) -> list[str]:
    return [l for l in rendered.decode().splitlines()
            if needle in l and not l.startswith("#")]


def test_token_usage_by_type_and_duration():
    m = GenAIMetrics()
    labels = m.labels(operation="chat", provider="openai", original_model="orig",
                      request_model="gpt", response_model="gpt-4o-2024")
    m.record_tokens(labels, Usage(input_tokens=7, output_tokens=3, total_tokens=10,
                                  cached_input_tokens=2, reasoning_tokens=5))
    m.record_request(labels, 1.5)
    m.record_request(labels, 0.5, error_type="upstream_502")
    out = m.render()
    sums = {l.split()[0]: float(l.split()[1])
            for l in _sample(out, "gen_ai_client_token_usage_sum")}
    def val(tt):
        return next(v for k, v in sums.items() if f'gen_ai_token_type="{tt}"' in k)
    assert val("input") == 7 and val("output") == 3 and val("total") == 10
    assert val("cached_input") == 2 and val("reasoning") == 5
    durs = _sample(out, "gen_ai_server_request_duration_seconds_count")
    assert any('error_type=""' in l and l.endswith("1.0") for l in durs)
    assert any('error_type="upstream_502"' in l for l in durs)
    # response_model falls back to request model when empty
    l2 = m.labels(operation="chat", provider="p", original_model="o",
                  request_model="rm", response_model="")
    assert l2[-1] == "rm"


def test_stream_latency_formulas():
    m = GenAIMetrics()
    labels = m.labels(operation="chat", provider="openai", original_model="m",
                      request_model="m", response_model="m")
    # ttft 0.2s; 5 output tokens over 1.0s total -> ITL = (1.0-0.2)/4 = 0.2
    m.record_stream_latency(labels, ttft_s=0.2, elapsed_s=1.0, output_tokens=5)
    out = m.render()
    ttft_sum = float(_sample(out, "time_to_first_token_seconds_sum")[0].split()[1])
    itl_sum = float(_sample(out, "time_per_output_token_seconds_sum")[0].split()[1])
    assert abs(ttft_sum - 0.2) < 1e-9
    assert abs(itl_sum - 0.2) < 1e-9
    # single-token streams record no ITL
    m2 = GenAIMetrics()
    m2.record_stream_latency(labels, ttft_s=0.1, elapsed_s=0.3, output_tokens=1)
    assert _sample(m2.render(), "time_per_output_token_seconds_count") == []


def test_provider_from_schema():
    assert provider_from_schema("OpenAI", "my-backend") == "openai"
    assert provider_from_schema("AWSBedrock", "b") == "aws.bedrock"
    assert provider_from_schema("GCPVertexAI", "b") == "gcp.vertex_ai"
    assert provider_from_schema("Anthropic", "b") == "anthropic"


def test_child_cache_reuses_instances():
    m = GenAIMetrics()
    labels = m.labels(operation="chat", provider="p", original_model="a",
                      request_model="a", response_model="a")
    c1 = m._child(m.request_duration, labels + ("",))
    c2 = m._child(m.request_duration, labels + ("",))
    assert c1 is c2
