"""OTLP/HTTP trace export against a local collector endpoint (the wire
protocol an OTel collector's /v1/traces receiver accepts)."""

import http.server
import json
import threading
import time

from aigw.tracing.otlp import OTLPHTTPExporter, encode_batch, parse_otlp_headers
from aigw.tracing.tracing import Tracer, tracing_from_env


class _Collector(http.server.BaseHTTPRequestHandler):
    received: list = []

    def do_POST(self):
        body = self.rfile.read(int(self.headers["Content-Length"]))
        type(self).received.append(
            {"path": self.path, "headers": dict(self.headers), "body": json.loads(body)}
        )
        self.send_response(200)
        self.send_header("Content-Length", "2")
        self.end_headers()
        self.wfile.write(b"{}")

    def log_message(self, *a):
        pass


def _start_collector():
    srv = http.server.HTTPServer(("127.0.0.1", 0), _Collector)
    t = threading.Thread(target=srv.serve_forever, daemon=True)
    t.start()
    return srv, srv.server_address[1]


def test_export_batches_to_collector():
    _Collector.received = []
    srv, port = _start_collector()
    try:
        tracer = Tracer(
            semconv="gen_ai",
            exporter=OTLPHTTPExporter(
                f"http://127.0.0.1:{port}",
                headers={"authorization": "Bearer tok"},
                service_name="svc-x",
                flush_interval_s=0.05,
            ),
        )
        span = tracer.start_span("gateway.request", headers={})
        span.set("gen_ai.request.model", "gpt-4o-mini")
        span.set("gen_ai.usage.input_tokens", 12)
        span.add_event("chunk", {"n": 1})
        span.status = "ERROR"
        tracer.end_span(span)
        deadline = time.time() + 5
        while not _Collector.received and time.time() < deadline:
            time.sleep(0.02)
        tracer.exporter.shutdown()
    finally:
        srv.shutdown()
    assert _Collector.received, "collector never received a batch"
    req = _Collector.received[0]
    assert req["path"] == "/v1/traces"
    assert req["headers"].get("Authorization") == "Bearer tok"
    rs = req["body"]["resourceSpans"][0]
    attrs = {a["key"]: a["value"] for a in rs["resource"]["attributes"]}
    assert attrs["service.name"] == {"stringValue": "svc-x"}
    otlp_span = rs["scopeSpans"][0]["spans"][0]
    assert otlp_span["name"] == "gateway.request"
    assert len(otlp_span["traceId"]) == 32 and len(otlp_span["spanId"]) == 16
    sattrs = {a["key"]: a["value"] for a in otlp_span["attributes"]}
    assert sattrs["gen_ai.request.model"] == {"stringValue": "gpt-4o-mini"}
    assert sattrs["gen_ai.usage.input_tokens"] == {"intValue": "12"}
    assert otlp_span["status"] == {"code": 2}
    assert otlp_span["events"][0]["name"] == "chunk"


def test_tracing_from_env_selects_otlp(monkeypatch):
    tracer = tracing_from_env(
        {
            "OTEL_EXPORTER_OTLP_ENDPOINT": "http://127.0.0.1:9/",
            "OTEL_EXPORTER_OTLP_HEADERS": "x-team=a,authorization=Bearer t",
            "OTEL_SERVICE_NAME": "gw",
        }
    )
    assert tracer is not None
    exp = tracer.exporter
    assert exp.url == "http://127.0.0.1:9/v1/traces"
    assert exp.headers["x-team"] == "a"
    assert exp.service_name == "gw"
    exp.shutdown(timeout_s=1)


def test_failed_export_drops_without_raising():
    exp = OTLPHTTPExporter("http://127.0.0.1:9", flush_interval_s=0.01, timeout_s=0.2)

    class S:
        name = "s"
        trace_id = "0" * 32
        span_id = "1" * 16
        parent_span_id = ""
        start_ns = 1
        end_ns = 2
        attributes = {}
        events = []
        status = "OK"

    exp.export(S())
    time.sleep(0.3)
    exp.shutdown()  # no exception = pass


def test_encode_value_types():
    class S:
        name = "s"
        trace_id = "a" * 32
        span_id = "b" * 16
        parent_span_id = "c" * 16
        start_ns = 5
        end_ns = 9
        attributes = {"f": 1.5, "b": True, "l": ["x", 2], "s": "y"}
        events = []
        status = "UNSET"

    body = json.loads(encode_batch([S()], "svc"))
    span = body["resourceSpans"][0]["scopeSpans"][0]["spans"][0]
    attrs = {a["key"]: a["value"] for a in span["attributes"]}
    assert attrs["f"] == {"doubleValue": 1.5}
    assert attrs["b"] == {"boolValue": True}
    assert attrs["l"]["arrayValue"]["values"][1] == {"intValue": "2"}
    assert span["parentSpanId"] == "c" * 16
    assert span["status"] == {"code": 0}
    assert parse_otlp_headers("") == {}
