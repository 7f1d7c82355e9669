"""OTLP/HTTP trace export (JSON encoding).

The reference exports spans over OTLP via the OTel SDK, configured from
the standard env vars (OTEL_EXPORTER_OTLP_ENDPOINT,
OTEL_EXPORTER_OTLP_HEADERS, OTEL_SERVICE_NAME). This is a from-scratch
exporter speaking the OTLP/HTTP 1.x JSON protocol — the same payloads an
OTel collector's ``/v1/traces`` receiver accepts — so the gateway plugs
into existing OTLP pipelines without an SDK dependency.

Spans are batched on a background thread (batch size / flush interval
mirroring the SDK's BatchSpanProcessor defaults) so export never blocks
the request path; a full queue drops spans rather than applying
backpressure.
"""

from __future__ import annotations

import json
import logging
import queue
import threading
import urllib.request

logger = logging.getLogger("aigw.tracing.otlp")

_STATUS_CODE = {"OK": 1, "ERROR": 2}


def _any_value(v):
    """Encode a python value as an OTLP AnyValue."""
    if isinstance(v, bool):
        return {"boolValue": v}
    if isinstance(v, int):
        return {"intValue": str(v)}  # int64 is a JSON string in OTLP
    if isinstance(v, float):
        return {"doubleValue": v}
    if isinstance(v, (list, tuple)):
        return {"arrayValue": {"values": [_any_value(x) for x in v]}}
    return {"stringValue": str(v)}


def _kv_list(attrs: dict) -> list:
    return [{"key": k, "value": _any_value(v)} for k, v in attrs.items()]


def span_to_otlp(span) -> dict:
    out = {
        "traceId": span.trace_id,
        "spanId": span.span_id,
        "name": span.name,
        "kind": 2,  # SPAN_KIND_SERVER
        "startTimeUnixNano": str(span.start_ns),
        "endTimeUnixNano": str(span.end_ns),
        "attributes": _kv_list(span.attributes),
        "events": [
            {
                "timeUnixNano": str(e["ts_ns"]),
                "name": e["name"],
                "attributes": _kv_list(e.get("attributes") or {}),
            }
            for e in span.events
        ],
        "status": {"code": _STATUS_CODE.get(span.status, 0)},
    }
    if span.parent_span_id:
        out["parentSpanId"] = span.parent_span_id
    return out


def encode_batch(spans: list, service_name: str) -> bytes:
    return json.dumps(
        {
            "resourceSpans": [
                {
                    "resource": {
                        "attributes": _kv_list({"service.name": service_name})
                    },
                    "scopeSpans": [
                        {
                            "scope": {"name": "aigw"},
                            "spans": [span_to_otlp(s) for s in spans],
                        }
                    ],
                }
            ]
        }
    ).encode("utf-8")


class OTLPHTTPExporter:
    def __init__(
        self,
        endpoint: str,
        *,
        headers: dict[str, str] | None = None,
        service_name: str = "ai-gateway",
        max_batch: int = 512,
        flush_interval_s: float = 5.0,
        timeout_s: float = 10.0,
        max_queue: int = 2048,
    ):
        base = endpoint.rstrip("/")
        # OTEL_EXPORTER_OTLP_ENDPOINT is the base URL; the traces signal
        # path is appended unless the caller already points at it
        self.url = base if base.endswith("/v1/traces") else base + "/v1/traces"
        self.headers = {"content-type": "application/json", **(headers or {})}
        self.service_name = service_name
        self.max_batch = max_batch
        self.flush_interval_s = flush_interval_s
        self.timeout_s = timeout_s
        self._q: queue.Queue = queue.Queue(maxsize=max_queue)
        self._closed = threading.Event()
        self._thread = threading.Thread(
            target=self._loop, name="aigw-otlp-export", daemon=True
        )
        self._thread.start()

    def export(self, span) -> None:
        try:
            self._q.put_nowait(span)
        except queue.Full:  # drop, never backpressure the request path
            pass

    def _loop(self) -> None:
        while not self._closed.is_set():
            batch = self._drain(block=True)
            if batch:
                self._send(batch)
        # final flush on close
        batch = self._drain(block=False)
        if batch:
            self._send(batch)

    def _drain(self, block: bool) -> list:
        batch = []
        try:
            timeout = self.flush_interval_s if block else 0
            batch.append(self._q.get(block=block, timeout=timeout or None))
        except queue.Empty:
            return batch
        while len(batch) < self.max_batch:
            try:
                batch.append(self._q.get_nowait())
            except queue.Empty:
                break
        return [s for s in batch if s is not None]  # drop shutdown sentinel

    def _send(self, batch: list) -> None:
        body = encode_batch(batch, self.service_name)
        req = urllib.request.Request(self.url, data=body, headers=self.headers)
        try:
            with urllib.request.urlopen(req, timeout=self.timeout_s) as resp:
                resp.read()
        except Exception as e:
            logger.warning("OTLP export of %d span(s) failed: %s", len(batch), e)

    def shutdown(self, timeout_s: float = 5.0) -> None:
        self._closed.set()
        # unblock the queue.get
        try:
            self._q.put_nowait(None)
        except queue.Full:
            pass
        self._thread.join(timeout=timeout_s)


def parse_otlp_headers(raw: str) -> dict[str, str]:
    """OTEL_EXPORTER_OTLP_HEADERS: comma-separated key=value pairs."""
    out = {}
    for pair in raw.split(","):
        if "=" in pair:
            k, v = pair.split("=", 1)
            out[k.strip()] = v.strip()
    return out
