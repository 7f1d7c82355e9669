"""Backend credential injection.

One handler per credential kind, dispatched from the backend config exactly
as internal/backendauth/auth.go:18-68 does:

- static API key  -> ``Authorization: Bearer`` (api_key.go)
- Anthropic key   -> ``x-api-key`` + ``anthropic-version`` (anthropicapikey.go)
- Azure API key   -> ``api-key`` (azureapikey.go)
- Azure AD token  -> ``Authorization: Bearer`` (azure.go)
- GCP token       -> ``Authorization: Bearer`` (gcp.go; URL rewrite is done
                     by the GCP translators, which own the vendor path)
- AWS credentials -> SigV4 over the final mutated body (aws.go:86-160)

Per-request credential override from client headers is honored before the
configured credentials (credential_override.go), and the override headers
are always stripped before the request goes upstream.
"""

from __future__ import annotations

from typing import Callable, Optional

import os

from aigw import internalapi
from aigw.backendauth.sigv4 import sign_sigv4
from aigw.filterapi.config import Backend, BackendAuth


class _FileCredential:
    """mtime-cached credential file reader — picks up rotated credentials
    without a restart (the BSP rotator analogue)."""

    __slots__ = ("path", "_mtime", "_value")

    def __init__(self, path: str):
        self.path = path
        self._mtime = -1.0
        self._value = ""

    def read(self) -> str:
        try:
            mtime = os.stat(self.path).st_mtime
        except OSError:
            return self._value
        if mtime != self._mtime:
            with open(self.path, "r", encoding="utf-8") as f:
                self._value = f.read().strip()
            self._mtime = mtime
        return self._value


def _parse_aws_ini(text: str) -> dict[str, str]:
    out: dict[str, str] = {}
    for line in text.splitlines():
        line = line.strip()
        if not line or line.startswith(("#", "[", ";")):
            continue
        k, _, v = line.partition("=")
        out[k.strip().lower()] = v.strip()
    return out

# An AuthHandler mutates (headers, body) right before dispatch and returns
# the new headers. Body is already final (mutation ordering: A.8).
AuthHandler = Callable[[dict[str, str], bytes, str, str], dict[str, str]]
# signature: (headers, body, method, path) -> headers

_OVERRIDE_HEADERS = (
    internalapi.API_KEY_OVERRIDE_HEADER,
    internalapi.AWS_ACCESS_KEY_OVERRIDE_HEADER,
    internalapi.AWS_SECRET_KEY_OVERRIDE_HEADER,
    internalapi.AWS_SESSION_TOKEN_OVERRIDE_HEADER,
)


def _pop_overrides(headers: dict[str, str]) -> dict[str, str]:
    ov = {}
    for h in _OVERRIDE_HEADERS:
        v = headers.pop(h, None)
        if v is not None:
            ov[h] = v
    return ov


def build_auth_handler(backend: Backend) -> Optional[AuthHandler]:
    auth = backend.auth
    if auth is None:
        return None
    kind = auth.kind
    if kind == "api_key":
        if auth.api_key_file:
            cred = _FileCredential(auth.api_key_file)

            def handler(headers, body, method, path, _cred=cred):
                ov = _pop_overrides(headers)
                key = ov.get(internalapi.API_KEY_OVERRIDE_HEADER, _cred.read())
                headers["authorization"] = f"Bearer {key}"
                return headers

            return handler
        return _bearer_handler(auth.api_key)
    if kind == "anthropic_api_key":
        return _anthropic_handler(auth)
    if kind == "azure_api_key":
        return _azure_key_handler(auth)
    if kind == "azure_access_token":
        return _bearer_handler(auth.azure_access_token)
    if kind == "gcp_access_token":
        return _bearer_handler(auth.gcp_access_token)
    if kind == "aws_access_key_id":
        return _aws_handler(backend, auth)
    return None


def _bearer_handler(token: str) -> AuthHandler:
    def handler(headers, body, method, path):
        ov = _pop_overrides(headers)
        key = ov.get(internalapi.API_KEY_OVERRIDE_HEADER, token)
        headers["authorization"] = f"Bearer {key}"
        return headers

    return handler


def _anthropic_handler(auth: BackendAuth) -> AuthHandler:
    def handler(headers, body, method, path):
        ov = _pop_overrides(headers)
        headers["x-api-key"] = ov.get(internalapi.API_KEY_OVERRIDE_HEADER, auth.anthropic_api_key)
        headers.setdefault("anthropic-version", "2023-06-01")
        headers.pop("authorization", None)
        return headers

    return handler


def _azure_key_handler(auth: BackendAuth) -> AuthHandler:
    def handler(headers, body, method, path):
        ov = _pop_overrides(headers)
        headers["api-key"] = ov.get(internalapi.API_KEY_OVERRIDE_HEADER, auth.azure_api_key)
        headers.pop("authorization", None)
        return headers

    return handler


def _aws_handler(backend: Backend, auth: BackendAuth) -> AuthHandler:
    host = backend.upstream.hostname or backend.upstream.host
    region = auth.aws_region or "us-east-1"
    cred = _FileCredential(auth.aws_credentials_file) if auth.aws_credentials_file else None

    def handler(headers, body, method, path):
        ov = _pop_overrides(headers)
        file_creds = _parse_aws_ini(cred.read()) if cred is not None else {}
        access = ov.get(
            internalapi.AWS_ACCESS_KEY_OVERRIDE_HEADER,
            file_creds.get("aws_access_key_id", auth.aws_access_key_id),
        )
        secret = ov.get(
            internalapi.AWS_SECRET_KEY_OVERRIDE_HEADER,
            file_creds.get("aws_secret_access_key", auth.aws_secret_access_key),
        )
        token = ov.get(
            internalapi.AWS_SESSION_TOKEN_OVERRIDE_HEADER,
            file_creds.get("aws_session_token", auth.aws_session_token),
        )
        headers.pop("authorization", None)
        return sign_sigv4(
            method,
            path,
            headers,
            body,
            host=host,
            region=region,
            service="bedrock",
            access_key_id=access,
            secret_access_key=secret,
            session_token=token,
        )

    return handler
