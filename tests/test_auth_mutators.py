"""Backend auth + mutator tests (parity: internal/backendauth/*_test.go,
internal/headermutator, internal/bodymutator)."""

from datetime import datetime, timezone

import pytest

from aigw import internalapi
from aigw.backendauth import build_auth_handler
from aigw.backendauth.auth import CredentialMissingError, IncompleteAWSCredentialError
from aigw.backendauth.sigv4 import sign_sigv4
from aigw.filterapi.config import (
    Backend,
    BackendAuth,
    BodyMutation,
    CredentialOverride,
    HeaderMutation,
    Upstream,
)
from aigw.mutator import apply_body_mutation, apply_header_mutation


def test_api_key_bearer():
    b = Backend(name="o", auth=BackendAuth(api_key="sk-abc"))
    h = build_auth_handler(b)({}, b"", "POST", "/v1/chat/completions")
    assert h["authorization"] == "Bearer sk-abc"


def test_anthropic_key():
    b = Backend(name="a", auth=BackendAuth(anthropic_api_key="ak"))
    h = build_auth_handler(b)({"authorization": "Bearer client"}, b"", "POST", "/v1/messages")
    assert h["x-api-key"] == "ak"
    assert h["anthropic-version"] == "2023-06-01"
    assert "authorization" not in h


def test_azure_key():
    b = Backend(name="az", auth=BackendAuth(azure_api_key="zk"))
    h = build_auth_handler(b)({}, b"", "POST", "/x")
    assert h["api-key"] == "zk"


def test_credential_override_gated_on_config():
    """Override headers are ignored unless the backend opts in via
    credentialOverride (credential_override.go semantics)."""
    # not configured: client-sent override header is ignored
    b = Backend(name="o", auth=BackendAuth(api_key="sk-config"))
    h = build_auth_handler(b)({"x-client-key": "sk-override"}, b"", "POST", "/v1/x")
    assert h["authorization"] == "Bearer sk-config"

    # configured: the named header supplies the credential and is consumed
    b = Backend(name="o", auth=BackendAuth(
        api_key="sk-config",
        credential_override=CredentialOverride(header_name="x-client-key",
                                               fallback_to_configured=True)))
    h = build_auth_handler(b)({"x-client-key": "sk-override"}, b"", "POST", "/v1/x")
    assert h["authorization"] == "Bearer sk-override"
    assert "x-client-key" not in h

    # configured + absent + fallback: static credential
    h = build_auth_handler(b)({}, b"", "POST", "/v1/x")
    assert h["authorization"] == "Bearer sk-config"

    # configured + absent + NO fallback: 401 (CredentialMissingError)
    b = Backend(name="o", auth=BackendAuth(
        api_key="sk-config",
        credential_override=CredentialOverride(header_name="x-client-key")))
    with pytest.raises(CredentialMissingError):
        build_auth_handler(b)({}, b"", "POST", "/v1/x")


def test_aws_credential_override_headers():
    """AWS overrides use the x-aigw-aws- prefix trio; incomplete pairs are
    rejected (ErrIncompleteAWSCredential)."""
    b = Backend(name="aws", auth=BackendAuth(
        aws_access_key_id="AKIDSTATIC", aws_secret_access_key="s3cr3t",
        aws_region="us-east-1",
        credential_override=CredentialOverride(fallback_to_configured=True)))
    names = internalapi.aws_credential_override_header_names()
    assert names == ("x-aigw-aws-access-key-id",
                     "x-aigw-aws-secret-access-key",
                     "x-aigw-aws-session-token")
    hdrs = {names[0]: "AKIDOVER", names[1]: "oversecret"}
    h = build_auth_handler(b)(hdrs, b"{}", "POST", "/model/m/converse")
    assert "AKIDOVER" in h["authorization"]
    for n in names:
        assert n not in h

    # access key without secret -> incomplete -> 401-class error
    with pytest.raises(IncompleteAWSCredentialError):
        build_auth_handler(b)({names[0]: "AKIDONLY"}, b"{}", "POST", "/x")

    # absent + fallback: static credential signs
    h = build_auth_handler(b)({}, b"{}", "POST", "/model/m/converse")
    assert "AKIDSTATIC" in h["authorization"]


# SigV4 known-answer test: vector computed with the canonical algorithm
# (deterministic given fixed time/credentials), guards against regressions.
def test_sigv4_deterministic():
    now = datetime(2024, 1, 15, 12, 0, 0, tzinfo=timezone.utc)
    h = sign_sigv4(
        "POST",
        "/model/anthropic.claude-3/converse",
        {"content-type": "application/json"},
        b'{"messages":[]}',
        host="bedrock-runtime.us-east-1.amazonaws.com",
        region="us-east-1",
        service="bedrock",
        access_key_id="AKIDEXAMPLE",
        secret_access_key="wJalrXUtnFEMI/K7MDENG+bPxRfiCYEXAMPLEKEY",
        now=now,
    )
    assert h["x-amz-date"] == "20240115T120000Z"
    assert h["authorization"].startswith(
        "AWS4-HMAC-SHA256 Credential=AKIDEXAMPLE/20240115/us-east-1/bedrock/aws4_request, "
        "SignedHeaders=content-type;host;x-amz-content-sha256;x-amz-date, Signature="
    )
    # Signature must change when the body changes (signs the FINAL body).
    h2 = sign_sigv4(
        "POST",
        "/model/anthropic.claude-3/converse",
        {"content-type": "application/json"},
        b'{"messages":[{"role":"user"}]}',
        host="bedrock-runtime.us-east-1.amazonaws.com",
        region="us-east-1",
        service="bedrock",
        access_key_id="AKIDEXAMPLE",
        secret_access_key="wJalrXUtnFEMI/K7MDENG+bPxRfiCYEXAMPLEKEY",
        now=now,
    )
    assert h["authorization"] != h2["authorization"]


def test_sigv4_session_token_signed():
    now = datetime(2024, 1, 15, 12, 0, 0, tzinfo=timezone.utc)
    h = sign_sigv4(
        "POST", "/p", {}, b"", host="h", region="r", service="s",
        access_key_id="A", secret_access_key="S", session_token="TOK", now=now,
    )
    assert h["x-amz-security-token"] == "TOK"
    assert "x-amz-security-token" in h["authorization"]


def test_header_mutation():
    out = apply_header_mutation(
        {"a": "1", "b": "2"}, HeaderMutation(set={"C": "3"}, remove=["B"])
    )
    assert out == {"a": "1", "c": "3"}


def test_body_mutation_paths():
    doc = {"model": "m", "nested": {"keep": 1, "drop": 2}, "arr": [{"x": 1}]}
    mut = BodyMutation(
        set={"nested.new": "v", "arr.0.x": 9, "top": True},
        remove=["nested.drop", "missing.path"],
    )
    apply_body_mutation(doc, mut)
    assert doc["nested"] == {"keep": 1, "new": "v"}
    assert doc["arr"][0]["x"] == 9
    assert doc["top"] is True


def test_body_mutation_escaped_dot():
    doc = {}
    apply_body_mutation(doc, BodyMutation(set={r"metadata\.key.sub": 1}))
    assert doc == {"metadata.key": {"sub": 1}}


def test_file_based_credentials_rotate(tmp_path):
    """File-backed credentials are re-read on change (BSP rotator analogue)."""
    key_file = tmp_path / "key"
    key_file.write_text("sk-old\n")
    b = Backend(name="o", auth=BackendAuth(api_key_file=str(key_file)))
    h = build_auth_handler(b)
    assert h({}, b"", "POST", "/x")["authorization"] == "Bearer sk-old"
    import os
    import time

    key_file.write_text("sk-new\n")
    os.utime(key_file, (time.time() + 5, time.time() + 5))
    assert h({}, b"", "POST", "/x")["authorization"] == "Bearer sk-new"


def test_aws_credentials_file(tmp_path):
    cf = tmp_path / "credentials"
    cf.write_text("[default]\naws_access_key_id = AKFILE\naws_secret_access_key = SKFILE\n")
    b = Backend(
        name="aws",
        upstream=Upstream(host="bedrock.example", port=443),
        auth=BackendAuth(aws_credentials_file=str(cf), aws_region="eu-west-1"),
    )
    h = build_auth_handler(b)({}, b"{}", "POST", "/model/m/converse")
    assert "Credential=AKFILE/" in h["authorization"]
    assert "/eu-west-1/bedrock/aws4_request" in h["authorization"]


def test_sigv4_cross_validated_independent_implementation():
    """The production signer vs a from-first-principles SigV4 written
    here (RFC-style 4 steps) — catches shared-bug blindness that a
    self-consistency known-answer cannot."""
    import hashlib
    import hmac as hmac_mod
    from datetime import datetime, timezone

    method, path, body = "POST", "/model/claude/converse", b'{"messages":[]}'
    host, region, service = "bedrock-runtime.us-east-1.amazonaws.com", "us-east-1", "bedrock"
    ak, sk = "AKIDEXAMPLE", "wJalrXUtnFEMI/K7MDENG+bPxRfiCYEXAMPLEKEY"
    now = datetime(2015, 8, 30, 12, 36, 0, tzinfo=timezone.utc)

    got = sign_sigv4(method, path, {"content-type": "application/json"}, body,
                     host=host, region=region, service=service,
                     access_key_id=ak, secret_access_key=sk, now=now)

    # independent computation
    amz_date, datestamp = "20150830T123600Z", "20150830"
    payload_hash = hashlib.sha256(body).hexdigest()
    signed_names = sorted(
        got["authorization"].split("SignedHeaders=")[1].split(",")[0].split(";")
    )
    all_headers = {"host": host, "x-amz-date": amz_date,
                   "x-amz-content-sha256": payload_hash,
                   "content-type": "application/json"}
    canonical_headers = "".join(f"{n}:{all_headers[n]}\n" for n in signed_names)
    canonical = "\n".join([method, path, "", canonical_headers,
                           ";".join(signed_names), payload_hash])
    scope = f"{datestamp}/{region}/{service}/aws4_request"
    sts = "\n".join(["AWS4-HMAC-SHA256", amz_date, scope,
                     hashlib.sha256(canonical.encode()).hexdigest()])

    def hm(key, msg):
        return hmac_mod.new(key, msg.encode(), hashlib.sha256).digest()

    k = hm(hm(hm(hm(b"AWS4" + sk.encode(), datestamp), region), service),
           "aws4_request")
    expected_sig = hmac_mod.new(k, sts.encode(), hashlib.sha256).hexdigest()
    assert got["authorization"].endswith(f"Signature={expected_sig}"), (
        got["authorization"], expected_sig)
    assert got["x-amz-date"] == amz_date
