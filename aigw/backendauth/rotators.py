"""OIDC / token-exchange credential rotation.

The reference rotates backend credentials in its controller and delivers
them to the data plane through k8s Secrets mounted as files
(internal/controller/rotators/: aws_oidc_rotator.go, azure_token_rotator.go,
gcp_oidc_token_rotator.go; OIDC sourcing in controller/tokenprovider/).
This gateway keeps the same delivery contract — the data-plane auth
handlers read mtime-cached credential FILES (aigw/backendauth/auth.py
_FileCredential) — and runs the exchange flows in-process:

- OIDCTokenProvider: client-credentials grant against the issuer's
  discovered token endpoint (tokenprovider/oidc_token_provider.go);
- AwsOidcRotator: OIDC JWT -> STS AssumeRoleWithWebIdentity (Query API,
  XML response) -> AWS credentials INI file, formatted exactly like
  rotators/aws_common.go formatAWSCredentialsFile (:108-120);
- AzureTokenRotator: client secret OR federated client assertion
  (OIDC JWT) -> AAD v2 token endpoint -> bearer-token file
  (azure_token_rotator.go + tokenprovider/azure_*.go);
- GcpOidcRotator: OIDC JWT -> GCP STS token exchange
  (grant-type:token-exchange, workload-identity audience) -> optional
  service-account impersonation via iamcredentials generateAccessToken
  -> bearer-token file (gcp_oidc_token_rotator.go:192-219).

Expiry handling mirrors rotators/common.go: each credential file gets a
sidecar ``<file>.expiry`` holding the RFC3339 expiration (the Secret
annotation analogue), and a rotator refreshes once
``now >= expiry - pre_rotation_window`` (IsBufferedTimeExpired,
common.go:69-71). All endpoints are injectable so the flows run against
offline fakes in tests.
"""

from __future__ import annotations

import asyncio
import base64
import json
import logging
import os
import re
import time
import urllib.parse
import urllib.request
from dataclasses import dataclass, field
from datetime import datetime, timezone
from typing import Optional

logger = logging.getLogger("aigw.rotators")

DEFAULT_PRE_ROTATION_WINDOW_S = 300.0


class RotationError(Exception):
    pass


def _http_form_post(url: str, form: dict[str, str],
                    headers: Optional[dict[str, str]] = None) -> tuple[int, bytes]:
    data = urllib.parse.urlencode(form).encode()
    req = urllib.request.Request(url, data=data, method="POST")
    req.add_header("content-type", "application/x-www-form-urlencoded")
    for k, v in (headers or {}).items():
        req.add_header(k, v)
    try:
        with urllib.request.urlopen(req, timeout=30) as resp:
            return resp.status, resp.read()
    except urllib.error.HTTPError as e:
        return e.code, e.read()


def _http_json_post(url: str, payload: dict,
                    headers: Optional[dict[str, str]] = None) -> tuple[int, bytes]:
    data = json.dumps(payload).encode()
    req = urllib.request.Request(url, data=data, method="POST")
    req.add_header("content-type", "application/json")
    for k, v in (headers or {}).items():
        req.add_header(k, v)
    try:
        with urllib.request.urlopen(req, timeout=30) as resp:
            return resp.status, resp.read()
    except urllib.error.HTTPError as e:
        return e.code, e.read()


def _http_get(url: str) -> tuple[int, bytes]:
    req = urllib.request.Request(url, method="GET")
    try:
        with urllib.request.urlopen(req, timeout=30) as resp:
            return resp.status, resp.read()
    except urllib.error.HTTPError as e:
        return e.code, e.read()


def _jwt_exp(token: str) -> Optional[float]:
    """exp claim of an unverified JWT (the rotator only needs scheduling
    information; verification is the relying service's job)."""
    parts = token.split(".")
    if len(parts) != 3:
        return None
    try:
        payload = parts[1] + "=" * (-len(parts[1]) % 4)
        claims = json.loads(base64.urlsafe_b64decode(payload))
        exp = claims.get("exp")
        return float(exp) if exp is not None else None
    except Exception:
        return None


@dataclass
class TokenExpiry:
    token: str
    expires_at: float  # unix seconds


@dataclass
class OIDCConfig:
    """BackendSecurityPolicy OIDC source (api/v1beta1
    backendsecurity_policy.go BSPOIDC)."""

    issuer: str = ""
    client_id: str = ""
    client_secret: str = ""
    client_secret_file: str = ""
    audience: str = ""
    scopes: list[str] = field(default_factory=lambda: ["openid"])
    # test hook: skip discovery and POST straight here
    token_endpoint: str = ""


class OIDCTokenProvider:
    """Client-credentials grant against the issuer
    (tokenprovider/oidc_token_provider.go): discovery via
    /.well-known/openid-configuration, then POST to token_endpoint."""

    def __init__(self, cfg: OIDCConfig):
        self.cfg = cfg
        self._token_endpoint = cfg.token_endpoint

    def _secret(self) -> str:
        if self.cfg.client_secret_file:
            with open(self.cfg.client_secret_file, encoding="utf-8") as f:
                return f.read().strip()
        return self.cfg.client_secret

    def _discover(self) -> str:
        if self._token_endpoint:
            return self._token_endpoint
        url = self.cfg.issuer.rstrip("/") + "/.well-known/openid-configuration"
        status, body = _http_get(url)
        if status != 200:
            raise RotationError(f"OIDC discovery failed: {status}")
        doc = json.loads(body)
        ep = doc.get("token_endpoint")
        if not ep:
            raise RotationError("OIDC discovery: no token_endpoint")
        self._token_endpoint = ep
        return ep

    def get_token(self) -> TokenExpiry:
        form = {
            "grant_type": "client_credentials",
            "client_id": self.cfg.client_id,
            "client_secret": self._secret(),
        }
        if self.cfg.scopes:
            form["scope"] = " ".join(self.cfg.scopes)
        if self.cfg.audience:
            form["audience"] = self.cfg.audience
        status, body = _http_form_post(self._discover(), form)
        if status != 200:
            raise RotationError(f"OIDC token grant failed: {status} {body[:200]!r}")
        doc = json.loads(body)
        token = doc.get("access_token") or doc.get("id_token") or ""
        if not token:
            raise RotationError("OIDC token response missing access_token")
        now = time.time()
        if doc.get("expires_in") is not None:
            exp = now + float(doc["expires_in"])
        else:
            exp = _jwt_exp(token) or (now + 3600.0)
        return TokenExpiry(token, exp)


def _write_credential_file(path: str, content: str, expires_at: float) -> None:
    """Atomic write + sidecar expiry (the Secret-annotation analogue,
    rotators/common.go updateExpirationSecretAnnotation)."""
    os.makedirs(os.path.dirname(os.path.abspath(path)), exist_ok=True)
    tmp = f"{path}.tmp.{os.getpid()}"
    with open(tmp, "w", encoding="utf-8") as f:
        f.write(content)
    os.replace(tmp, path)
    iso = datetime.fromtimestamp(expires_at, tz=timezone.utc).isoformat()
    with open(path + ".expiry", "w", encoding="utf-8") as f:
        f.write(iso)


def read_expiry(path: str) -> Optional[float]:
    try:
        with open(path + ".expiry", encoding="utf-8") as f:
            return datetime.fromisoformat(f.read().strip()).timestamp()
    except (OSError, ValueError):
        return None


class Rotator:
    """Base: refresh scheduling per rotators/common.go."""

    out_file: str
    pre_rotation_window_s: float = DEFAULT_PRE_ROTATION_WINDOW_S

    def is_expired(self, now: Optional[float] = None) -> bool:
        exp = read_expiry(self.out_file)
        if exp is None:
            return True
        return (now if now is not None else time.time()) >= exp - self.pre_rotation_window_s

    def rotate(self) -> float:
        raise NotImplementedError


class AwsOidcRotator(Rotator):
    """OIDC JWT -> STS AssumeRoleWithWebIdentity -> credentials INI
    (aws_oidc_rotator.go + aws_common.go)."""

    def __init__(self, oidc: OIDCTokenProvider, *, role_arn: str, region: str,
                 out_file: str, session_prefix: str = "ai-gateway",
                 sts_endpoint: str = "", profile: str = "default",
                 pre_rotation_window_s: float = DEFAULT_PRE_ROTATION_WINDOW_S):
        self.oidc = oidc
        self.role_arn = role_arn
        self.region = region
        self.out_file = out_file
        self.session_name = f"{session_prefix}-{profile}"
        self.sts_endpoint = sts_endpoint or f"https://sts.{region}.amazonaws.com/"
        self.profile = profile
        self.pre_rotation_window_s = pre_rotation_window_s

    def rotate(self) -> float:
        web_token = self.oidc.get_token()
        status, body = _http_form_post(self.sts_endpoint, {
            "Action": "AssumeRoleWithWebIdentity",
            "Version": "2011-06-15",
            "RoleArn": self.role_arn,
            "RoleSessionName": self.session_name,
            "WebIdentityToken": web_token.token,
        })
        if status != 200:
            raise RotationError(f"STS AssumeRoleWithWebIdentity: {status} {body[:200]!r}")
        text = body.decode("utf-8", "replace")

        def tag(name: str) -> str:
            m = re.search(rf"<{name}>([^<]*)</{name}>", text)
            return m.group(1) if m else ""

        access, secret, session = (tag("AccessKeyId"), tag("SecretAccessKey"),
                                   tag("SessionToken"))
        if not access or not secret:
            raise RotationError("STS response missing credentials")
        exp_s = tag("Expiration")
        try:
            expires_at = datetime.fromisoformat(exp_s.replace("Z", "+00:00")).timestamp()
        except ValueError:
            expires_at = time.time() + 3600.0
        # formatAWSCredentialsFile layout (aws_common.go:108-120)
        lines = [f"[{self.profile}]",
                 f"aws_access_key_id = {access}",
                 f"aws_secret_access_key = {secret}"]
        if session:
            lines.append(f"aws_session_token = {session}")
        lines.append(f"region = {self.region}")
        _write_credential_file(self.out_file, "\n".join(lines) + "\n", expires_at)
        return expires_at


class AzureTokenRotator(Rotator):
    """Client-secret or federated-assertion exchange at the AAD v2 token
    endpoint (azure_token_rotator.go; azidentity ClientSecretCredential /
    ClientAssertionCredential equivalents)."""

    def __init__(self, *, tenant_id: str, client_id: str, out_file: str,
                 client_secret: str = "", oidc: Optional[OIDCTokenProvider] = None,
                 scope: str = "https://cognitiveservices.azure.com/.default",
                 authority: str = "https://login.microsoftonline.com",
                 pre_rotation_window_s: float = DEFAULT_PRE_ROTATION_WINDOW_S):
        if not client_secret and oidc is None:
            raise RotationError("azure rotation needs a client secret or an OIDC provider")
        self.tenant_id = tenant_id
        self.client_id = client_id
        self.client_secret = client_secret
        self.oidc = oidc
        self.scope = scope
        self.authority = authority.rstrip("/")
        self.out_file = out_file
        self.pre_rotation_window_s = pre_rotation_window_s

    def rotate(self) -> float:
        form = {
            "grant_type": "client_credentials",
            "client_id": self.client_id,
            "scope": self.scope,
        }
        if self.client_secret:
            form["client_secret"] = self.client_secret
        else:
            # federated workload identity: the OIDC JWT is the assertion
            form["client_assertion_type"] = (
                "urn:ietf:params:oauth:client-assertion-type:jwt-bearer")
            form["client_assertion"] = self.oidc.get_token().token
        url = f"{self.authority}/{self.tenant_id}/oauth2/v2.0/token"
        status, body = _http_form_post(url, form)
        if status != 200:
            raise RotationError(f"AAD token grant failed: {status} {body[:200]!r}")
        doc = json.loads(body)
        token = doc.get("access_token", "")
        if not token:
            raise RotationError("AAD response missing access_token")
        expires_at = time.time() + float(doc.get("expires_in", 3600))
        _write_credential_file(self.out_file, token, expires_at)
        return expires_at


class GcpOidcRotator(Rotator):
    """OIDC JWT -> GCP STS token exchange -> optional service-account
    impersonation (gcp_oidc_token_rotator.go:192-219,
    exchangeJWTForSTSToken :266-310, impersonateServiceAccount :352)."""

    GRANT_TYPE = "urn:ietf:params:oauth:grant-type:token-exchange"
    SCOPE = "https://www.googleapis.com/auth/cloud-platform"

    def __init__(self, oidc: OIDCTokenProvider, *, project_number: str,
                 pool_name: str, provider_name: str, out_file: str,
                 service_account: str = "", project_name: str = "",
                 sts_endpoint: str = "https://sts.googleapis.com/v1/token",
                 iam_endpoint: str = "https://iamcredentials.googleapis.com",
                 pre_rotation_window_s: float = DEFAULT_PRE_ROTATION_WINDOW_S):
        self.oidc = oidc
        self.project_number = project_number
        self.pool_name = pool_name
        self.provider_name = provider_name
        self.service_account = service_account
        self.project_name = project_name
        self.sts_endpoint = sts_endpoint
        self.iam_endpoint = iam_endpoint.rstrip("/")
        self.out_file = out_file
        self.pre_rotation_window_s = pre_rotation_window_s

    def rotate(self) -> float:
        jwt = self.oidc.get_token()
        audience = (f"//iam.googleapis.com/projects/{self.project_number}"
                    f"/locations/global/workloadIdentityPools/{self.pool_name}"
                    f"/providers/{self.provider_name}")
        status, body = _http_json_post(self.sts_endpoint, {
            "grantType": self.GRANT_TYPE,
            "audience": audience,
            "scope": self.SCOPE,
            "requestedTokenType": "urn:ietf:params:oauth:token-type:access_token",
            "subjectToken": jwt.token,
            "subjectTokenType": "urn:ietf:params:oauth:token-type:jwt",
        })
        if status != 200:
            raise RotationError(f"GCP STS exchange failed: {status} {body[:200]!r}")
        doc = json.loads(body)
        token = doc.get("access_token", "")
        expires_at = time.time() + float(doc.get("expires_in", 3600))
        if not token:
            raise RotationError("GCP STS response missing access_token")
        if self.service_account:
            sa = (f"{self.iam_endpoint}/v1/projects/-/serviceAccounts/"
                  f"{self.service_account}:generateAccessToken")
            status, body = _http_json_post(
                sa, {"scope": [self.SCOPE]},
                headers={"authorization": f"Bearer {token}"})
            if status != 200:
                raise RotationError(
                    f"GCP SA impersonation failed: {status} {body[:200]!r}")
            doc = json.loads(body)
            token = doc.get("accessToken", "")
            if not token:
                raise RotationError("generateAccessToken missing accessToken")
            exp = doc.get("expireTime")
            if exp:
                try:
                    expires_at = datetime.fromisoformat(
                        exp.replace("Z", "+00:00")).timestamp()
                except ValueError:
                    pass
        _write_credential_file(self.out_file, token, expires_at)
        return expires_at


def build_rotation_manager(cfg) -> Optional["RotationManager"]:
    """Scan a filterapi Config for backend auths carrying ``rotation``
    and build the manager driving them (the controller's
    BackendSecurityPolicy → rotator wiring, controller.go:120-285
    collapsed to one process). Returns None when nothing rotates."""
    rotators: list[Rotator] = []
    seen: set[str] = set()
    for route in cfg.routes:
        for b in route.backends:
            auth = b.auth
            if auth is None or auth.rotation is None:
                continue
            rot_cfg = auth.rotation
            out = rot_cfg.out_file or auth.api_key_file or auth.aws_credentials_file
            if not out:
                raise RotationError(
                    f"backend {b.name!r}: rotation needs an output file "
                    "(set auth.apiKeyFile / auth.awsCredentialsFile or "
                    "rotation.outFile)")
            if out in seen:
                continue  # one rotator per credential file
            seen.add(out)
            oidc = None
            if rot_cfg.oidc is not None:
                o = rot_cfg.oidc
                oidc = OIDCTokenProvider(OIDCConfig(
                    issuer=o.issuer, client_id=o.client_id,
                    client_secret=o.client_secret,
                    client_secret_file=o.client_secret_file,
                    audience=o.audience,
                    scopes=o.scopes or ["openid"],
                    token_endpoint=o.token_endpoint,
                ))
            if rot_cfg.kind == "aws_oidc":
                if oidc is None:
                    raise RotationError(f"backend {b.name!r}: aws_oidc needs oidc")
                rotators.append(AwsOidcRotator(
                    oidc, role_arn=rot_cfg.aws_role_arn,
                    region=rot_cfg.aws_region or auth.aws_region or "us-east-1",
                    out_file=out, sts_endpoint=rot_cfg.sts_endpoint,
                    pre_rotation_window_s=rot_cfg.pre_rotation_window_s))
            elif rot_cfg.kind == "azure":
                rotators.append(AzureTokenRotator(
                    tenant_id=rot_cfg.azure_tenant_id,
                    client_id=rot_cfg.azure_client_id,
                    client_secret=rot_cfg.azure_client_secret,
                    oidc=oidc, out_file=out,
                    scope=rot_cfg.azure_scope or
                    "https://cognitiveservices.azure.com/.default",
                    authority=rot_cfg.azure_authority or
                    "https://login.microsoftonline.com",
                    pre_rotation_window_s=rot_cfg.pre_rotation_window_s))
            elif rot_cfg.kind == "gcp_oidc":
                if oidc is None:
                    raise RotationError(f"backend {b.name!r}: gcp_oidc needs oidc")
                rotators.append(GcpOidcRotator(
                    oidc, project_number=rot_cfg.gcp_project_number,
                    pool_name=rot_cfg.gcp_pool,
                    provider_name=rot_cfg.gcp_provider,
                    service_account=rot_cfg.gcp_service_account,
                    out_file=out,
                    sts_endpoint=rot_cfg.gcp_sts_endpoint or
                    "https://sts.googleapis.com/v1/token",
                    iam_endpoint=rot_cfg.gcp_iam_endpoint or
                    "https://iamcredentials.googleapis.com",
                    pre_rotation_window_s=rot_cfg.pre_rotation_window_s))
    return RotationManager(rotators) if rotators else None


class RotationManager:
    """Drives a set of rotators from one asyncio task: rotate whatever is
    (about to be) expired, then sleep until the earliest next deadline.
    The data plane never blocks on this — auth handlers keep serving the
    previous file content until the atomic replace lands."""

    def __init__(self, rotators: list[Rotator], *, check_interval_s: float = 30.0):
        self.rotators = rotators
        self.check_interval_s = check_interval_s
        self._task: Optional[asyncio.Task] = None
        self.rotation_count = 0
        self.last_errors: dict[str, str] = {}

    def rotate_expired(self, now: Optional[float] = None) -> int:
        """Synchronous sweep (also the unit-test surface): rotate every
        rotator whose pre-rotation window has opened."""
        n = 0
        for r in self.rotators:
            if r.is_expired(now):
                try:
                    r.rotate()
                    self.rotation_count += 1
                    self.last_errors.pop(r.out_file, None)
                    n += 1
                except Exception as e:  # keep serving the old credential
                    self.last_errors[r.out_file] = str(e)
                    logger.warning("rotation failed for %s: %s", r.out_file, e)
        return n

    async def start(self) -> None:
        async def loop():
            while True:
                await asyncio.to_thread(self.rotate_expired)
                await asyncio.sleep(self.check_interval_s)

        self._task = asyncio.ensure_future(loop())

    async def stop(self) -> None:
        if self._task is not None:
            self._task.cancel()
            try:
                await self._task
            except asyncio.CancelledError:
                pass
            self._task = None
