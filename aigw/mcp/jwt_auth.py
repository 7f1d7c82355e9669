"""Stdlib-only JWT (RS256) validation for MCP client authorization.

The reference's MCPRoute securityPolicy.oauth validates client access
tokens against an issuer + audiences + JWKS (api/v1beta1/mcp_route.go:
291-311, jwks at :520). This environment has no JOSE library, so the
RS256 verify is implemented directly: RSASSA-PKCS1-v1_5 verification is
one modular exponentiation (pow(sig, e, n)) followed by an
EMSA-PKCS1-v1_5 padding check over the SHA-256 digest — no key
generation, no CRT, nothing exotic on the verify path.

Supported: RS256 (the default alg of every major issuer). Tokens signed
with any other alg are rejected (never "alg: none").

JWKS sources: inline JWKS JSON (``localJWKS`` analogue) or a file path.
Remote JWKS fetch (RFC 8414 discovery) needs egress and is out of scope
here; the config accepts a pre-fetched JWKS instead.
"""

from __future__ import annotations

import base64
import hashlib
import json
import time
from dataclasses import dataclass, field

# DigestInfo prefix for SHA-256 (RFC 8017 §9.2 note 1)
_SHA256_DIGESTINFO = bytes.fromhex("3031300d060960864801650304020105000420")


class JWTError(ValueError):
    pass


def _b64url_decode(s: str) -> bytes:
    pad = -len(s) % 4
    try:
        return base64.urlsafe_b64decode(s + "=" * pad)
    except (ValueError, TypeError) as e:
        raise JWTError(f"bad base64url segment: {e}") from e


def _b64url_uint(s: str) -> int:
    return int.from_bytes(_b64url_decode(s), "big")


@dataclass
class RSAKey:
    n: int
    e: int
    kid: str = ""

    @property
    def byte_len(self) -> int:
        return (self.n.bit_length() + 7) // 8

    def verify_pkcs1_sha256(self, message: bytes, signature: bytes) -> bool:
        k = self.byte_len
        if len(signature) != k:
            return False
        em = pow(int.from_bytes(signature, "big"), self.e, self.n).to_bytes(k, "big")
        # EMSA-PKCS1-v1_5: 0x00 0x01 PS(0xff..) 0x00 DigestInfo || H
        t = _SHA256_DIGESTINFO + hashlib.sha256(message).digest()
        if k < len(t) + 11:
            return False
        expected = b"\x00\x01" + b"\xff" * (k - len(t) - 3) + b"\x00" + t
        # constant-time-ish compare (verification of a public operation,
        # but no reason to leak anything)
        import hmac as _hmac

        return _hmac.compare_digest(em, expected)


def parse_jwks(doc: dict) -> list[RSAKey]:
    keys = []
    for k in doc.get("keys") or []:
        if k.get("kty") != "RSA":
            continue
        use = k.get("use", "sig")
        if use != "sig":
            continue
        try:
            keys.append(RSAKey(n=_b64url_uint(k["n"]), e=_b64url_uint(k["e"]),
                               kid=k.get("kid", "")))
        except (KeyError, JWTError):
            continue
    return keys


@dataclass
class JWTValidator:
    issuer: str = ""
    audiences: list[str] = field(default_factory=list)
    keys: list[RSAKey] = field(default_factory=list)
    leeway_s: float = 60.0

    @classmethod
    def from_config(cls, issuer: str, audiences: list[str], jwks_json: str = "",
                    jwks_file: str = "") -> "JWTValidator":
        raw = jwks_json
        if not raw and jwks_file:
            with open(jwks_file, "r", encoding="utf-8") as f:
                raw = f.read()
        if not raw:
            raise JWTError("oauth: JWKS required (inline jwks or jwksFile)")
        doc = json.loads(raw)
        keys = parse_jwks(doc)
        if not keys:
            raise JWTError("oauth: JWKS contains no usable RSA signing keys")
        return cls(issuer=issuer, audiences=list(audiences or []), keys=keys)

    def validate(self, token: str) -> dict:
        """Returns the claims dict; raises JWTError on any failure."""
        parts = token.split(".")
        if len(parts) != 3:
            raise JWTError("token is not a JWS compact serialization")
        header_b, payload_b, sig_b = parts
        header = _json_seg(header_b, "header")
        if header.get("alg") != "RS256":
            raise JWTError(f"unsupported alg {header.get('alg')!r} (RS256 only)")
        signature = _b64url_decode(sig_b)
        signed = (header_b + "." + payload_b).encode("ascii")
        kid = header.get("kid", "")
        candidates = [k for k in self.keys if not kid or not k.kid or k.kid == kid]
        if not any(k.verify_pkcs1_sha256(signed, signature) for k in candidates):
            raise JWTError("signature verification failed")
        claims = _json_seg(payload_b, "payload")
        now = time.time()
        exp = claims.get("exp")
        if exp is not None and now > float(exp) + self.leeway_s:
            raise JWTError("token expired")
        nbf = claims.get("nbf")
        if nbf is not None and now < float(nbf) - self.leeway_s:
            raise JWTError("token not yet valid")
        if self.issuer and claims.get("iss") != self.issuer:
            raise JWTError(f"issuer mismatch: {claims.get('iss')!r}")
        if self.audiences:
            aud = claims.get("aud")
            auds = aud if isinstance(aud, list) else [aud]
            if not any(a in self.audiences for a in auds):
                raise JWTError(f"audience mismatch: {aud!r}")
        return claims


def _json_seg(seg: str, what: str) -> dict:
    try:
        doc = json.loads(_b64url_decode(seg))
    except ValueError as e:
        raise JWTError(f"bad {what} JSON: {e}") from e
    if not isinstance(doc, dict):
        raise JWTError(f"{what} is not an object")
    return doc
