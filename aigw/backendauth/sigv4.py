"""AWS Signature Version 4 request signing.

From-scratch implementation of the SigV4 algorithm (no AWS SDK), used to
sign translated Bedrock requests over the FINAL mutated body — the ordering
invariant the reference enforces at extproc/processor_impl.go:419-434 and
backendauth/aws.go:86-160. The signing host comes from the selected
upstream, matching aws.go's use of x-ai-eg-upstream-host.
"""

from __future__ import annotations

import hashlib
import hmac
import urllib.parse
from datetime import datetime, timezone

_ALGO = "AWS4-HMAC-SHA256"


def _hmac(key: bytes, msg: str) -> bytes:
    return hmac.new(key, msg.encode("utf-8"), hashlib.sha256).digest()


def _canonical_uri(path: str) -> str:
    # Each path segment URI-encoded (RFC 3986), preserving '/'.
    if not path:
        return "/"
    segs = path.split("/")
    return "/".join(urllib.parse.quote(urllib.parse.unquote(s), safe="") for s in segs)


def _canonical_query(query: str) -> str:
    if not query:
        return ""
    pairs = []
    for part in query.split("&"):
        if not part:
            continue
        k, _, v = part.partition("=")
        pairs.append(
            (
                urllib.parse.quote(urllib.parse.unquote(k), safe=""),
                urllib.parse.quote(urllib.parse.unquote(v), safe=""),
            )
        )
    pairs.sort()
    return "&".join(f"{k}={v}" for k, v in pairs)


def sign_sigv4(
    method: str,
    path: str,
    headers: dict[str, str],
    body: bytes,
    *,
    host: str,
    region: str,
    service: str,
    access_key_id: str,
    secret_access_key: str,
    session_token: str = "",
    now: datetime | None = None,
) -> dict[str, str]:
    """Return headers with SigV4 Authorization/x-amz-* applied.

    ``headers`` keys must be lowercase; the returned dict is a new mapping.
    """
    now = now or datetime.now(timezone.utc)
    amz_date = now.strftime("%Y%m%dT%H%M%SZ")
    datestamp = now.strftime("%Y%m%d")

    out = dict(headers)
    out["host"] = host
    out["x-amz-date"] = amz_date
    if session_token:
        out["x-amz-security-token"] = session_token
    payload_hash = hashlib.sha256(body).hexdigest()
    out["x-amz-content-sha256"] = payload_hash

    path_only, _, query = path.partition("?")
    signed_names = sorted(
        k for k in out if k in ("host", "content-type") or k.startswith("x-amz-")
    )
    canonical_headers = "".join(f"{k}:{out[k].strip()}\n" for k in signed_names)
    signed_headers = ";".join(signed_names)
    canonical_request = "\n".join(
        (
            method.upper(),
            _canonical_uri(path_only),
            _canonical_query(query),
            canonical_headers,
            signed_headers,
            payload_hash,
        )
    )
    scope = f"{datestamp}/{region}/{service}/aws4_request"
    string_to_sign = "\n".join(
        (
            _ALGO,
            amz_date,
            scope,
            hashlib.sha256(canonical_request.encode("utf-8")).hexdigest(),
        )
    )
    k_date = _hmac(("AWS4" + secret_access_key).encode("utf-8"), datestamp)
    k_region = _hmac(k_date, region)
    k_service = _hmac(k_region, service)
    k_signing = _hmac(k_service, "aws4_request")
    signature = hmac.new(k_signing, string_to_sign.encode("utf-8"), hashlib.sha256).hexdigest()
    out["authorization"] = (
        f"{_ALGO} Credential={access_key_id}/{scope}, "
        f"SignedHeaders={signed_headers}, Signature={signature}"
    )
    return out
