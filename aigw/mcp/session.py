"""Stateless encrypted MCP sessions.

Parity with internal/mcpproxy/session.go: the aggregated per-backend
session ids are serialized into an encrypted, authenticated, URL-safe blob
handed to the client as ``mcp-session-id``, so ANY gateway shard can resume
the session with zero shared state.

The reference uses PBKDF2+AES-GCM; this environment has no AES primitive
(no `cryptography`/`pycryptodome` offline), so the same properties —
key-from-seed derivation, confidentiality, integrity, nonce-per-blob —
are provided with stdlib primitives: PBKDF2-HMAC-SHA256 key derivation,
an HMAC-SHA256 counter-mode keystream for encryption, and
encrypt-then-MAC (HMAC-SHA256, constant-time verify) for authentication.
A fallback seed list supports seamless key rotation (mainlib/main.go:365-377).
"""

from __future__ import annotations

import base64
import hashlib
import hmac
import json
import secrets
from typing import Optional


class SessionError(ValueError):
    pass


class SessionCrypto:
    def __init__(self, seed: str, fallback_seeds: Optional[list[str]] = None):
        self._keys = [self._derive(seed)]
        for s in fallback_seeds or []:
            self._keys.append(self._derive(s))

    @staticmethod
    def _derive(seed: str) -> tuple[bytes, bytes]:
        km = hashlib.pbkdf2_hmac("sha256", seed.encode(), b"aigw-mcp-session", 100_000, 64)
        return km[:32], km[32:]  # (encryption key, mac key)

    @staticmethod
    def _keystream(key: bytes, nonce: bytes, n: int) -> bytes:
        out = bytearray()
        counter = 0
        while len(out) < n:
            out.extend(hmac.new(key, nonce + counter.to_bytes(8, "big"), hashlib.sha256).digest())
            counter += 1
        return bytes(out[:n])

    def seal(self, payload: dict) -> str:
        enc_key, mac_key = self._keys[0]
        pt = json.dumps(payload, separators=(",", ":")).encode()
        nonce = secrets.token_bytes(16)
        ct = bytes(a ^ b for a, b in zip(pt, self._keystream(enc_key, nonce, len(pt))))
        mac = hmac.new(mac_key, nonce + ct, hashlib.sha256).digest()
        return base64.urlsafe_b64encode(nonce + ct + mac).decode().rstrip("=")

    def open(self, token: str) -> dict:
        try:
            raw = base64.urlsafe_b64decode(token + "=" * (-len(token) % 4))
        except Exception as e:
            raise SessionError("malformed session token") from e
        if len(raw) < 48:
            raise SessionError("session token too short")
        nonce, ct, mac = raw[:16], raw[16:-32], raw[-32:]
        for enc_key, mac_key in self._keys:
            want = hmac.new(mac_key, nonce + ct, hashlib.sha256).digest()
            if hmac.compare_digest(mac, want):
                pt = bytes(a ^ b for a, b in zip(ct, self._keystream(enc_key, nonce, len(ct))))
                try:
                    return json.loads(pt)
                except ValueError as e:
                    raise SessionError("corrupt session payload") from e
        raise SessionError("session authentication failed")
