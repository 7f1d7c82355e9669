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

Per-request credential override is honored ONLY when the backend config
sets ``credentialOverride`` (credential_override.go): the configured source
header supplies the credential; when it is absent, ``fallbackToConfigured``
selects between the static credential and a 401 (CredentialMissingError).
Override headers are consumed by the handler and the server additionally
strips every configured override header name at egress, so they never
reach an upstream.
"""

from __future__ import annotations

from typing import Callable, Optional

import os

from aigw import internalapi
from aigw.backendauth.sigv4 import sign_sigv4
from aigw.filterapi.config import Backend, BackendAuth, CredentialOverride


class CredentialMissingError(Exception):
    """Per-request credential source configured, absent, and
    fallbackToConfigured=false — the caller answers 401
    (reference: ErrCredentialMissing, credential_override.go:22-26)."""


class IncompleteAWSCredentialError(CredentialMissingError):
    """The source carried some but not all SigV4 inputs
    (reference: ErrIncompleteAWSCredential, credential_override.go:153-156)."""


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


def override_header_names(auth: BackendAuth) -> tuple[str, ...]:
    """Header names this backend's credentialOverride reads (and the server
    must strip at egress). Empty when overrides are not configured."""
    co = auth.credential_override
    if co is None:
        return ()
    if auth.kind == "aws_access_key_id":
        prefix = co.header_name or internalapi.AWS_CREDENTIAL_OVERRIDE_HEADER_PREFIX
        return internalapi.aws_credential_override_header_names(prefix)
    return (co.header_name,) if co.header_name else ()


def _resolve_override(co: CredentialOverride, headers: dict[str, str]) -> str:
    """Pop and return the per-request credential; '' when absent."""
    if not co.header_name:
        return ""
    return (headers.pop(co.header_name.lower(), "") or "").strip()


def _with_override(auth: BackendAuth, apply_static: AuthHandler,
                   apply_cred: Callable[[dict[str, str], str], dict[str, str]]) -> AuthHandler:
    """Wrap a static handler with per-request credential sourcing
    (credentialOverrideHandler.Do semantics, credential_override.go:98-118)."""
    co = auth.credential_override
    if co is None:
        return apply_static

    def handler(headers, body, method, path):
        cred = _resolve_override(co, headers)
        if not cred:
            if not co.fallback_to_configured:
                raise CredentialMissingError(
                    f"missing per-request credential header {co.header_name!r}"
                )
            return apply_static(headers, body, method, path)
        return apply_cred(headers, cred)

    return handler


def build_auth_handler(backend: Backend) -> Optional[AuthHandler]:
    auth = backend.auth
    if auth is None:
        return None
    kind = auth.kind

    def bearer_apply(headers, cred):
        headers["authorization"] = f"Bearer {cred}"
        return headers

    if kind == "api_key":
        if auth.api_key_file:
            cred = _FileCredential(auth.api_key_file)

            def static(headers, body, method, path, _cred=cred):
                headers["authorization"] = f"Bearer {_cred.read()}"
                return headers

            return _with_override(auth, static, bearer_apply)
        return _with_override(auth, _bearer_handler(auth.api_key), bearer_apply)
    if kind == "anthropic_api_key":

        def anthropic_apply(headers, cred):
            headers["x-api-key"] = cred
            headers.setdefault("anthropic-version", "2023-06-01")
            headers.pop("authorization", None)
            return headers

        return _with_override(auth, _anthropic_handler(auth), anthropic_apply)
    if kind == "azure_api_key":

        def azure_apply(headers, cred):
            headers["api-key"] = cred
            headers.pop("authorization", None)
            return headers

        return _with_override(auth, _azure_key_handler(auth), azure_apply)
    if kind == "azure_access_token":
        return _with_override(auth, _bearer_handler(auth.azure_access_token), bearer_apply)
    if kind == "gcp_access_token":
        return _with_override(auth, _bearer_handler(auth.gcp_access_token), bearer_apply)
    if kind == "aws_access_key_id":
        return _aws_handler(backend, auth)
    return None


def _bearer_handler(token: str) -> AuthHandler:
    def handler(headers, body, method, path):
        headers["authorization"] = f"Bearer {token}"
        return headers

    return handler


def _anthropic_handler(auth: BackendAuth) -> AuthHandler:
    def handler(headers, body, method, path):
        headers["x-api-key"] = auth.anthropic_api_key
        headers.setdefault("anthropic-version", "2023-06-01")
        headers.pop("authorization", None)
        return headers

    return handler


def _azure_key_handler(auth: BackendAuth) -> AuthHandler:
    def handler(headers, body, method, path):
        headers["api-key"] = auth.azure_api_key
        headers.pop("authorization", None)
        return headers

    return handler


def _aws_handler(backend: Backend, auth: BackendAuth) -> AuthHandler:
    host = backend.upstream.hostname or backend.upstream.host
    region = auth.aws_region or "us-east-1"
    cred = _FileCredential(auth.aws_credentials_file) if auth.aws_credentials_file else None
    co = auth.credential_override
    if co is not None:
        prefix = co.header_name or internalapi.AWS_CREDENTIAL_OVERRIDE_HEADER_PREFIX
        h_access, h_secret, h_session = internalapi.aws_credential_override_header_names(prefix)

    def handler(headers, body, method, path):
        access = secret = token = ""
        if co is not None:
            access = (headers.pop(h_access, "") or "").strip()
            secret = (headers.pop(h_secret, "") or "").strip()
            token = (headers.pop(h_session, "") or "").strip()
            if bool(access) != bool(secret):
                raise IncompleteAWSCredentialError(
                    "incomplete per-request AWS credential: access key ID and "
                    "secret access key must both be present"
                )
            if not access and not co.fallback_to_configured:
                raise CredentialMissingError(
                    f"missing per-request AWS credential headers {h_access!r}/{h_secret!r}"
                )
        if not access:
            file_creds = _parse_aws_ini(cred.read()) if cred is not None else {}
            access = file_creds.get("aws_access_key_id", auth.aws_access_key_id)
            secret = file_creds.get("aws_secret_access_key", auth.aws_secret_access_key)
            token = file_creds.get("aws_session_token", auth.aws_session_token)
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
