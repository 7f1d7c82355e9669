"""AWS binary event-stream (vnd.amazon.eventstream) codec.

From-scratch implementation of the framing AWS Bedrock's ConverseStream
uses, replacing the reference's use of aws-sdk-go-v2's eventstream decoder
(internal/translator/openai_awsbedrock.go:515-545). Frame layout:

    prelude:  total_length u32 | headers_length u32 | prelude_crc u32
    headers:  (name_len u8 | name | type u8 | value)*
    payload:  bytes
    trailer:  message_crc u32  (CRC32 of everything before it)

Only the header value types Bedrock emits are implemented (bool true/false,
byte, short, int, long, bytes, string, timestamp, uuid). The decoder is
incremental: feed() accepts arbitrary chunk boundaries and yields complete
messages. The encoder exists for the mock upstream and tests.
"""

from __future__ import annotations

import struct
import zlib
from dataclasses import dataclass, field

_T_BOOL_TRUE = 0
_T_BOOL_FALSE = 1
_T_BYTE = 2
_T_SHORT = 3
_T_INT = 4
_T_LONG = 5
_T_BYTES = 6
_T_STRING = 7
_T_TIMESTAMP = 8
_T_UUID = 9


class EventStreamError(ValueError):
    pass


@dataclass
class EventStreamMessage:
    headers: dict[str, object]
    payload: bytes

    @property
    def event_type(self) -> str:
        return str(self.headers.get(":event-type", ""))

    @property
    def message_type(self) -> str:
        return str(self.headers.get(":message-type", ""))

    @property
    def exception_type(self) -> str:
        return str(self.headers.get(":exception-type", ""))


def _encode_headers(headers: dict[str, object]) -> bytes:
    out = bytearray()
    for name, value in headers.items():
        nb = name.encode("utf-8")
        out.append(len(nb))
        out.extend(nb)
        if isinstance(value, bool):
            out.append(_T_BOOL_TRUE if value else _T_BOOL_FALSE)
        elif isinstance(value, int):
            out.append(_T_LONG)
            out.extend(struct.pack(">q", value))
        elif isinstance(value, bytes):
            out.append(_T_BYTES)
            out.extend(struct.pack(">H", len(value)))
            out.extend(value)
        else:
            vb = str(value).encode("utf-8")
            out.append(_T_STRING)
            out.extend(struct.pack(">H", len(vb)))
            out.extend(vb)
    return bytes(out)


def encode_message(headers: dict[str, object], payload: bytes) -> bytes:
    hb = _encode_headers(headers)
    total = 12 + len(hb) + len(payload) + 4
    prelude = struct.pack(">II", total, len(hb))
    prelude_crc = zlib.crc32(prelude) & 0xFFFFFFFF
    body = prelude + struct.pack(">I", prelude_crc) + hb + payload
    msg_crc = zlib.crc32(body) & 0xFFFFFFFF
    return body + struct.pack(">I", msg_crc)


def encode_event(event_type: str, payload: bytes) -> bytes:
    return encode_message(
        {
            ":message-type": "event",
            ":event-type": event_type,
            ":content-type": "application/json",
        },
        payload,
    )


def _decode_headers(data: bytes) -> dict[str, object]:
    headers: dict[str, object] = {}
    i = 0
    n = len(data)
    while i < n:
        name_len = data[i]
        i += 1
        name = data[i : i + name_len].decode("utf-8")
        i += name_len
        t = data[i]
        i += 1
        if t == _T_BOOL_TRUE:
            headers[name] = True
        elif t == _T_BOOL_FALSE:
            headers[name] = False
        elif t == _T_BYTE:
            headers[name] = struct.unpack_from(">b", data, i)[0]
            i += 1
        elif t == _T_SHORT:
            headers[name] = struct.unpack_from(">h", data, i)[0]
            i += 2
        elif t == _T_INT:
            headers[name] = struct.unpack_from(">i", data, i)[0]
            i += 4
        elif t in (_T_LONG, _T_TIMESTAMP):
            headers[name] = struct.unpack_from(">q", data, i)[0]
            i += 8
        elif t in (_T_BYTES, _T_STRING):
            (vlen,) = struct.unpack_from(">H", data, i)
            i += 2
            v = data[i : i + vlen]
            i += vlen
            headers[name] = v.decode("utf-8") if t == _T_STRING else v
        elif t == _T_UUID:
            headers[name] = data[i : i + 16]
            i += 16
        else:
            raise EventStreamError(f"unknown header value type {t}")
    return headers


@dataclass
class EventStreamDecoder:
    """Incremental decoder tolerant of arbitrary chunk boundaries."""

    _buf: bytearray = field(default_factory=bytearray)

    def feed(self, chunk: bytes) -> list[EventStreamMessage]:
        self._buf.extend(chunk)
        out: list[EventStreamMessage] = []
        while len(self._buf) >= 12:
            total, hlen = struct.unpack_from(">II", self._buf, 0)
            if total < 16 or total > 16 * 1024 * 1024:
                raise EventStreamError(f"implausible frame length {total}")
            if len(self._buf) < total:
                break
            frame = bytes(self._buf[:total])
            del self._buf[:total]
            (prelude_crc,) = struct.unpack_from(">I", frame, 8)
            if zlib.crc32(frame[:8]) & 0xFFFFFFFF != prelude_crc:
                raise EventStreamError("prelude CRC mismatch")
            (msg_crc,) = struct.unpack_from(">I", frame, total - 4)
            if zlib.crc32(frame[: total - 4]) & 0xFFFFFFFF != msg_crc:
                raise EventStreamError("message CRC mismatch")
            headers = _decode_headers(frame[12 : 12 + hlen])
            payload = frame[12 + hlen : total - 4]
            out.append(EventStreamMessage(headers=headers, payload=payload))
        return out
