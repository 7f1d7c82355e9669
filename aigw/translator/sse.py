"""Server-Sent Events incremental codec.

Stateful SSE decoding across arbitrary network-chunk boundaries — the
correctness property the reference's buffered line scanner provides
(internal/translator/util.go:40-57; SURVEY.md §7 hard part 1). A decoder
instance is per-stream; feed it raw bytes as they arrive and it yields only
COMPLETE events, buffering any partial tail.

Both LF and CRLF line endings are accepted; multi-line ``data:`` fields are
joined with newlines per the SSE spec; comment lines (``:``) are dropped.
"""

from __future__ import annotations

from dataclasses import dataclass, field


@dataclass
class SSEEvent:
    data: str = ""
    event: str = ""
    id: str = ""

    def encode(self) -> bytes:
        out = []
        if self.event:
            out.append(f"event: {self.event}")
        if self.id:
            out.append(f"id: {self.id}")
        for line in self.data.split("\n"):
            out.append(f"data: {line}")
        return ("\n".join(out) + "\n\n").encode("utf-8")


@dataclass
class PySSEDecoder:
    _buf: bytearray = field(default_factory=bytearray)
    _data_lines: list = field(default_factory=list)
    _event: str = ""
    _id: str = ""

    def feed(self, chunk: bytes) -> list[SSEEvent]:
        """Consume a network chunk, return all events completed by it."""
        self._buf.extend(chunk)
        events: list[SSEEvent] = []
        while True:
            nl = self._buf.find(b"\n")
            if nl < 0:
                break
            line = bytes(self._buf[:nl])
            del self._buf[: nl + 1]
            if line.endswith(b"\r"):
                line = line[:-1]
            ev = self._feed_line(line)
            if ev is not None:
                events.append(ev)
        return events

    def flush(self) -> list[SSEEvent]:
        """End of stream: emit any final un-terminated event."""
        events = []
        if self._buf:
            ev = self._feed_line(bytes(self._buf))
            del self._buf[:]
            if ev is not None:
                events.append(ev)
        if self._data_lines or self._event:
            events.append(self._dispatch())
        return events

    def _feed_line(self, line: bytes):
        if not line:
            if self._data_lines or self._event or self._id:
                return self._dispatch()
            return None
        if line.startswith(b":"):
            return None
        name, _, value = line.partition(b":")
        if value.startswith(b" "):
            value = value[1:]
        field_name = name.decode("utf-8", "replace")
        v = value.decode("utf-8", "replace")
        if field_name == "data":
            self._data_lines.append(v)
        elif field_name == "event":
            self._event = v
        elif field_name == "id":
            self._id = v
        return None

    def _dispatch(self) -> SSEEvent:
        ev = SSEEvent(data="\n".join(self._data_lines), event=self._event, id=self._id)
        self._data_lines = []
        self._event = ""
        self._id = ""
        return ev


class NativeSSEDecoder:
    """Same interface as PySSEDecoder, backed by the C++ incremental
    splitter (csrc/aigw_native.cpp SSEFeed)."""

    __slots__ = ("_feed",)

    def __init__(self):
        self._feed = _native.SSEFeed()

    def feed(self, chunk: bytes) -> list[SSEEvent]:
        return [
            SSEEvent(data=data.decode("utf-8", "replace"), event=event)
            for event, data in self._feed.feed(chunk)
        ]

    def flush(self) -> list[SSEEvent]:
        return [
            SSEEvent(data=data.decode("utf-8", "replace"), event=event)
            for event, data in self._feed.flush()
        ]


try:
    import aigw_native as _native

    SSEDecoder = NativeSSEDecoder
except ImportError:  # pragma: no cover - extension built in-tree everywhere
    _native = None
    SSEDecoder = PySSEDecoder


def encode_data(data: str, event: str = "") -> bytes:
    return SSEEvent(data=data, event=event).encode()


DONE_EVENT = b"data: [DONE]\n\n"
