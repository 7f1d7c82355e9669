"""OIDC / token-exchange credential rotation against offline fake
endpoints (reference: internal/controller/rotators/*_test.go +
tokenprovider/*_test.go — same flows, no cloud):

- fake OIDC issuer: /.well-known/openid-configuration + client-credentials
  token endpoint issuing short-lived JWTs;
- fake STS answering AssumeRoleWithWebIdentity with XML credentials;
- fake AAD v2 token endpoint (client secret AND federated assertion);
- fake GCP STS token exchange + iamcredentials generateAccessToken.

Covers: initial rotation, expiry-driven refresh through the
pre-rotation window, mid-flight pickup by the data-plane auth handlers
(mtime-cached files) without a restart, and failure keeping the old
credential in place.
"""

import base64
import json
import threading
import time
from datetime import datetime, timezone
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer
from urllib.parse import parse_qs

import pytest

from aigw.backendauth import build_auth_handler
from aigw.backendauth.rotators import (
    AwsOidcRotator,
    AzureTokenRotator,
    GcpOidcRotator,
    OIDCConfig,
    OIDCTokenProvider,
    RotationManager,
    read_expiry,
)
from aigw.filterapi.config import Backend, BackendAuth, Upstream


def _jwt(claims: dict) -> str:
    def b64(d):
        return base64.urlsafe_b64encode(json.dumps(d).encode()).rstrip(b"=").decode()

    return f"{b64({'alg': 'none'})}.{b64(claims)}.sig"


class FakeCloud(BaseHTTPRequestHandler):
    """One handler serving every fake endpoint; state on the server."""

    def log_message(self, *a):
        pass

    def _send(self, code, body, ctype="application/json"):
        data = body if isinstance(body, bytes) else body.encode()
        self.send_response(code)
        self.send_header("content-type", ctype)
        self.send_header("content-length", str(len(data)))
        self.end_headers()
        self.wfile.write(data)

    def do_GET(self):
        if self.path == "/oidc/.well-known/openid-configuration":
            base = f"http://127.0.0.1:{self.server.server_address[1]}"
            self._send(200, json.dumps({"token_endpoint": f"{base}/oidc/token"}))
        else:
            self._send(404, "{}")

    def do_POST(self):
        n = int(self.headers.get("content-length", 0))
        raw = self.rfile.read(n)
        st = self.server.state
        st["calls"].append(self.path)
        if self.path == "/oidc/token":
            form = parse_qs(raw.decode())
            if form.get("client_secret", [""])[0] != "s3cret":
                return self._send(401, '{"error":"invalid_client"}')
            st["oidc_issued"] += 1
            tok = _jwt({"iss": "fake", "sub": form["client_id"][0],
                        "exp": time.time() + 120, "n": st["oidc_issued"]})
            return self._send(200, json.dumps(
                {"access_token": tok, "expires_in": 120}))
        if self.path == "/sts":
            form = parse_qs(raw.decode())
            assert form["Action"] == ["AssumeRoleWithWebIdentity"]
            if not form.get("WebIdentityToken", [""])[0]:
                return self._send(400, "<Error/>", "text/xml")
            st["sts_issued"] += 1
            exp = datetime.fromtimestamp(
                time.time() + st["aws_ttl_s"], tz=timezone.utc
            ).strftime("%Y-%m-%dT%H:%M:%SZ")
            xml = f"""<AssumeRoleWithWebIdentityResponse>
  <AssumeRoleWithWebIdentityResult><Credentials>
    <AccessKeyId>ASIA{st["sts_issued"]:04d}</AccessKeyId>
    <SecretAccessKey>secret{st["sts_issued"]}</SecretAccessKey>
    <SessionToken>tok{st["sts_issued"]}</SessionToken>
    <Expiration>{exp}</Expiration>
  </Credentials></AssumeRoleWithWebIdentityResult>
</AssumeRoleWithWebIdentityResponse>"""
            return self._send(200, xml, "text/xml")
        if self.path == "/aad/tenant-1/oauth2/v2.0/token":
            form = parse_qs(raw.decode())
            ok = form.get("client_secret", [""])[0] == "az-secret" or (
                form.get("client_assertion_type", [""])[0]
                == "urn:ietf:params:oauth:client-assertion-type:jwt-bearer"
                and form.get("client_assertion", [""])[0]
            )
            if not ok:
                return self._send(401, '{"error":"invalid_client"}')
            st["aad_issued"] += 1
            return self._send(200, json.dumps(
                {"access_token": f"aad-token-{st['aad_issued']}",
                 "expires_in": 90, "token_type": "Bearer"}))
        if self.path == "/gcpsts":
            doc = json.loads(raw)
            assert doc["grantType"] == "urn:ietf:params:oauth:grant-type:token-exchange"
            assert "workloadIdentityPools/pool-1/providers/prov-1" in doc["audience"]
            st["gcp_sts_issued"] += 1
            return self._send(200, json.dumps(
                {"access_token": f"gcp-sts-{st['gcp_sts_issued']}", "expires_in": 100}))
        if self.path.endswith(":generateAccessToken"):
            auth = self.headers.get("authorization", "")
            if not auth.startswith("Bearer gcp-sts-"):
                return self._send(401, "{}")
            st["gcp_sa_issued"] += 1
            exp = datetime.fromtimestamp(time.time() + 100, tz=timezone.utc)
            return self._send(200, json.dumps(
                {"accessToken": f"gcp-sa-{st['gcp_sa_issued']}",
                 "expireTime": exp.strftime("%Y-%m-%dT%H:%M:%SZ")}))
        self._send(404, "{}")


@pytest.fixture()
def cloud():
    srv = ThreadingHTTPServer(("127.0.0.1", 0), FakeCloud)
    srv.state = {"calls": [], "oidc_issued": 0, "sts_issued": 0,
                 "aad_issued": 0, "gcp_sts_issued": 0, "gcp_sa_issued": 0,
                 "aws_ttl_s": 3600}
    t = threading.Thread(target=srv.serve_forever, daemon=True)
    t.start()
    base = f"http://127.0.0.1:{srv.server_address[1]}"
    yield srv, base
    srv.shutdown()


def _oidc(base) -> OIDCTokenProvider:
    return OIDCTokenProvider(OIDCConfig(
        issuer=f"{base}/oidc", client_id="cid", client_secret="s3cret"))


def test_oidc_provider_discovery_and_grant(cloud):
    srv, base = cloud
    p = _oidc(base)
    te = p.get_token()
    assert te.token.count(".") == 2
    assert te.expires_at > time.time() + 60
    assert "/oidc/token" in srv.state["calls"]


def test_aws_oidc_rotation_and_auth_pickup(cloud, tmp_path):
    srv, base = cloud
    out = str(tmp_path / "aws" / "credentials")
    rot = AwsOidcRotator(_oidc(base), role_arn="arn:aws:iam::123:role/gw",
                         region="us-east-1", out_file=out,
                         sts_endpoint=f"{base}/sts")
    assert rot.is_expired()  # nothing on disk yet
    exp = rot.rotate()
    assert not rot.is_expired()
    assert abs(read_expiry(out) - exp) < 1e-6
    content = open(out).read()
    assert "[default]" in content
    assert "aws_access_key_id = ASIA0001" in content
    assert "aws_session_token = tok1" in content
    assert "region = us-east-1" in content

    # the data-plane AWS handler signs with the rotated file credentials
    b = Backend(name="bedrock", upstream=Upstream(
        host="bedrock.local", port=443,
        hostname="bedrock-runtime.us-east-1.amazonaws.com"),
        auth=BackendAuth(aws_credentials_file=out, aws_region="us-east-1"))
    h = build_auth_handler(b)({}, b"{}", "POST", "/model/m/converse")
    assert "ASIA0001" in h["authorization"]

    # mid-flight refresh: a second rotation (new STS credentials) is
    # picked up by the SAME handler without restart (mtime cache)
    time.sleep(0.02)
    rot.rotate()
    h = build_auth_handler(b)({}, b"{}", "POST", "/model/m/converse")
    assert "ASIA0002" in h["authorization"]


def test_azure_rotation_secret_and_federated(cloud, tmp_path):
    srv, base = cloud
    out = str(tmp_path / "azure.token")
    rot = AzureTokenRotator(tenant_id="tenant-1", client_id="app-1",
                            client_secret="az-secret", out_file=out,
                            authority=f"{base}/aad")
    rot.rotate()
    assert open(out).read() == "aad-token-1"

    # federated credential: OIDC JWT as client assertion
    out2 = str(tmp_path / "azure-fed.token")
    rot2 = AzureTokenRotator(tenant_id="tenant-1", client_id="app-1",
                             oidc=_oidc(base), out_file=out2,
                             authority=f"{base}/aad")
    rot2.rotate()
    assert open(out2).read() == "aad-token-2"

    # the azure handler serves the rotated token
    b = Backend(name="az", auth=BackendAuth(api_key_file=out))
    h = build_auth_handler(b)({}, b"", "POST", "/x")
    assert h["authorization"] == "Bearer aad-token-1"


def test_gcp_rotation_with_impersonation(cloud, tmp_path):
    srv, base = cloud
    out = str(tmp_path / "gcp.token")
    rot = GcpOidcRotator(_oidc(base), project_number="1234",
                         pool_name="pool-1", provider_name="prov-1",
                         service_account="gw@proj.iam.gserviceaccount.com",
                         out_file=out, sts_endpoint=f"{base}/gcpsts",
                         iam_endpoint=base)
    rot.rotate()
    assert open(out).read() == "gcp-sa-1"
    assert srv.state["gcp_sts_issued"] == 1 and srv.state["gcp_sa_issued"] == 1

    # without impersonation the STS token itself is written
    out2 = str(tmp_path / "gcp-direct.token")
    rot2 = GcpOidcRotator(_oidc(base), project_number="1234",
                          pool_name="pool-1", provider_name="prov-1",
                          out_file=out2, sts_endpoint=f"{base}/gcpsts")
    rot2.rotate()
    assert open(out2).read().startswith("gcp-sts-")


def test_manager_expiry_driven_refresh_and_failure_keeps_old(cloud, tmp_path):
    srv, base = cloud
    srv.state["aws_ttl_s"] = 600
    out = str(tmp_path / "credentials")
    rot = AwsOidcRotator(_oidc(base), role_arn="arn:aws:iam::123:role/gw",
                         region="us-east-1", out_file=out,
                         sts_endpoint=f"{base}/sts",
                         pre_rotation_window_s=300)
    mgr = RotationManager([rot])
    assert mgr.rotate_expired() == 1  # initial
    assert mgr.rotate_expired() == 0  # fresh: 600s ttl, 300s window
    # step the clock past (expiry - window): refresh happens
    assert mgr.rotate_expired(now=time.time() + 400) == 1
    assert srv.state["sts_issued"] == 2

    # endpoint failure: old credential stays, error recorded
    rot.sts_endpoint = f"{base}/nope"
    before = open(out).read()
    assert mgr.rotate_expired(now=time.time() + 4000) == 0
    assert open(out).read() == before
    assert mgr.last_errors.get(out)
    # endpoint restored: next sweep succeeds again
    rot.sts_endpoint = f"{base}/sts"
    assert mgr.rotate_expired(now=time.time() + 4000) == 1


def test_manager_async_lifecycle(cloud, tmp_path):
    """The CLI-driven async loop: starts, performs the initial rotation,
    and stops cleanly."""
    import asyncio

    srv, base = cloud
    out = str(tmp_path / "credentials")
    rot = AwsOidcRotator(_oidc(base), role_arn="arn:aws:iam::1:role/g",
                        region="us-east-1", out_file=out,
                        sts_endpoint=f"{base}/sts")
    mgr = RotationManager([rot], check_interval_s=0.05)

    async def run():
        await mgr.start()
        for _ in range(100):
            if mgr.rotation_count:
                break
            await asyncio.sleep(0.02)
        await mgr.stop()

    asyncio.run(run())
    assert mgr.rotation_count >= 1
    assert "aws_access_key_id = ASIA0001" in open(out).read()


def test_build_rotation_manager_from_config(cloud, tmp_path):
    from aigw.backendauth.rotators import build_rotation_manager
    from aigw.filterapi.config import load_config

    srv, base = cloud
    out = str(tmp_path / "rotated-key")
    cfg = load_config({
        "routes": [{"name": "r", "backends": [{
            "name": "az", "schema": "AzureOpenAI",
            "upstream": {"host": "h", "port": 443},
            "auth": {"apiKeyFile": out,
                     "rotation": {"kind": "azure",
                                  "azureTenantId": "tenant-1",
                                  "azureClientId": "app-1",
                                  "azureClientSecret": "az-secret",
                                  "azureAuthority": f"{base}/aad"}},
        }]}],
    })
    mgr = build_rotation_manager(cfg)
    assert mgr is not None and len(mgr.rotators) == 1
    assert mgr.rotate_expired() == 1
    assert open(out).read().startswith("aad-token-")
    # config without rotation -> None
    cfg2 = load_config({"routes": [{"name": "r", "backends": [{
        "name": "b", "schema": "OpenAI",
        "upstream": {"host": "h", "port": 80},
        "auth": {"apiKey": "sk"}}]}]})
    assert build_rotation_manager(cfg2) is None
