"""OTel GenAI semantic-convention metrics with Prometheus exposition.

Instrument names and attributes follow the reference's
internal/metrics/genai.go:14-24 (OTel GenAI semconv):

- ``gen_ai.client.token.usage``      histogram, by token type (input/output/
  total/cached_input/cache_creation/reasoning)
- ``gen_ai.server.request.duration`` histogram, seconds
- ``gen_ai.server.time_to_first_token`` histogram, seconds
- ``gen_ai.server.time_per_output_token`` histogram, seconds

TTFT is recorded on the FIRST response chunk regardless of token counts;
inter-token latency is recorded once at end-of-stream as
``(elapsed - ttft) / (max_output_tokens - 1)`` (metrics_impl.go:195-222).
Prometheus requires ``_`` instead of ``.`` in names; the OTel name is kept
in the HELP string.
"""

from __future__ import annotations

from prometheus_client import CollectorRegistry, Counter, Histogram, generate_latest

_TOKEN_BUCKETS = (1, 4, 16, 64, 256, 1024, 4096, 16384, 65536, 262144, 1048576)
_SECONDS_BUCKETS = (
    0.0001, 0.00025, 0.0005, 0.001, 0.0025, 0.005, 0.01, 0.025, 0.05,
    0.1, 0.25, 0.5, 1.0, 2.5, 5.0, 10.0, 30.0, 60.0, 120.0,
)

_LABELS = [
    "gen_ai_operation_name",
    "gen_ai_provider_name",
    "gen_ai_original_model",
    "gen_ai_request_model",
    "gen_ai_response_model",
]


class GenAIMetrics:
    def __init__(self, registry: CollectorRegistry | None = None):
        self.registry = registry or CollectorRegistry()
        self._children: dict = {}
        self.token_usage = Histogram(
            "gen_ai_client_token_usage",
            "OTel gen_ai.client.token.usage: tokens used per request",
            _LABELS + ["gen_ai_token_type"],
            buckets=_TOKEN_BUCKETS,
            registry=self.registry,
        )
        self.request_duration = Histogram(
            "gen_ai_server_request_duration_seconds",
            "OTel gen_ai.server.request.duration",
            _LABELS + ["error_type"],
            buckets=_SECONDS_BUCKETS,
            registry=self.registry,
        )
        self.ttft = Histogram(
            "gen_ai_server_time_to_first_token_seconds",
            "OTel gen_ai.server.time_to_first_token",
            _LABELS,
            buckets=_SECONDS_BUCKETS,
            registry=self.registry,
        )
        self.itl = Histogram(
            "gen_ai_server_time_per_output_token_seconds",
            "OTel gen_ai.server.time_per_output_token",
            _LABELS,
            buckets=_SECONDS_BUCKETS,
            registry=self.registry,
        )
        self.requests_total = Counter(
            "aigw_requests_total",
            "Requests by endpoint/backend/status",
            ["endpoint", "backend", "status"],
            registry=self.registry,
        )
        self.ratelimit_denials = Counter(
            "aigw_ratelimit_denials_total",
            "Requests denied by token rate limiting",
            ["rule"],
            registry=self.registry,
        )
        self.retries_total = Counter(
            "aigw_upstream_retries_total",
            "Upstream retry/fallback attempts beyond the first try",
            ["route"],
            registry=self.registry,
        )
        self.cache_events = Counter(
            "aigw_semantic_cache_events_total",
            "Semantic cache hits/misses",
            ["event"],
            registry=self.registry,
        )
        # MCP gateway instruments (internal/metrics/mcp_metrics.go)
        self.mcp_requests = Counter(
            "aigw_mcp_requests_total",
            "MCP JSON-RPC requests by method/outcome",
            ["method", "outcome"],
            registry=self.registry,
        )
        self.mcp_duration = Histogram(
            "aigw_mcp_request_duration_seconds",
            "MCP request duration",
            ["method"],
            buckets=_SECONDS_BUCKETS,
            registry=self.registry,
        )

    def labels(self, *, operation: str, provider: str, original_model: str,
               request_model: str, response_model: str) -> tuple:
        # ordered to match _LABELS; consumed positionally via the child
        # cache below (prometheus_client's labels() re-validates and locks
        # on every call — ~7% of a worker's loop time uncached)
        return (operation, provider, original_model, request_model,
                response_model or request_model)

    def _child(self, metric, values: tuple):
        key = (id(metric), values)
        c = self._children.get(key)
        if c is None:
            c = self._children[key] = metric.labels(*values)
        return c

    def record_tokens(self, labels: tuple, usage) -> None:
        for token_type, value in (
            ("input", usage.input_tokens),
            ("output", usage.output_tokens),
            ("total", usage.total_tokens),
            ("cached_input", usage.cached_input_tokens),
            ("cache_creation_input", usage.cache_creation_input_tokens),
            ("reasoning", usage.reasoning_tokens),
        ):
            if value:
                self._child(self.token_usage, labels + (token_type,)).observe(value)

    def record_request(self, labels: tuple, seconds: float, error_type: str = "") -> None:
        self._child(self.request_duration, labels + (error_type or "",)).observe(seconds)

    def record_stream_latency(
        self, labels: tuple, ttft_s: float, elapsed_s: float, output_tokens: int
    ) -> None:
        self._child(self.ttft, labels).observe(ttft_s)
        if output_tokens > 1:
            self._child(self.itl, labels).observe((elapsed_s - ttft_s) / (output_tokens - 1))

    def render(self) -> bytes:
        return generate_latest(self.registry)


def provider_from_schema(schema_name: str, backend_name: str) -> str:
    """Provider attribute derived from the backend schema, falling back to
    the backend name (metrics_impl.go:86-107)."""
    mapping = {
        "OpenAI": "openai",
        "AWSBedrock": "aws.bedrock",
        "AWSAnthropic": "anthropic",
        "AzureOpenAI": "azure.ai.openai",
        "GCPVertexAI": "gcp.vertex_ai",
        "GCPAnthropic": "anthropic",
        "Anthropic": "anthropic",
        "Cohere": "cohere",
    }
    return mapping.get(schema_name, backend_name)
