"""aigw — MI355X-native AI gateway.

A from-scratch, single-node AI gateway with the capability surface of
Envoy AI Gateway (reference: /root/reference): an
OpenAI/Anthropic/Cohere-compatible HTTP front, per-provider request and
response translation with SSE / AWS-event-stream re-chunking, token-usage
accounting with cost expressions, usage-based rate limiting, backend auth
injection, provider fallback, MCP proxying, and a declarative
AIGatewayRoute/AIServiceBackend-compatible config surface.

Architecture (MI355X-first, NOT an Envoy port): one gateway shard is a
single asyncio process pinned to one GPU. There is no proxy/extproc split
(the reference needs two ext_proc filter passes only because Envoy's filter
architecture demands it — SURVEY.md §A.8); routing, translation, auth and
accounting happen in-process. The GPU owns the hot accounting/caching work:
a CDNA4 HIP BPE tokenizer, an MFMA bf16/fp8 embedding GEMM feeding a
semantic response cache resident in HBM3E, and a KV-occupancy endpoint
scorer. Cross-shard state (token-budget counters, cache index) is kept
consistent with RCCL collectives over xGMI via torch.distributed.
"""

__version__ = "0.1.0"
