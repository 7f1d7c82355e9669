"""MCP OAuth JWT (RS256) validation — stdlib-only verify path against a
test-generated RSA keypair (key generation lives only in this test; the
gateway never generates keys)."""

import base64
import hashlib
import json
import random
import time

import pytest

from aigw.mcp.jwt_auth import JWTError, JWTValidator, parse_jwks

_SHA256_DIGESTINFO = bytes.fromhex("3031300d060960864801650304020105000420")


def _b64url(b: bytes) -> str:
    return base64.urlsafe_b64encode(b).rstrip(b"=").decode()


def _is_probable_prime(n, rng, rounds=24):
    if n < 2:
        return False
    for p in (2, 3, 5, 7, 11, 13, 17, 19, 23, 29, 31, 37):
        if n % p == 0:
            return n == p
    d, r = n - 1, 0
    while d % 2 == 0:
        d //= 2
        r += 1
    for _ in range(rounds):
        a = rng.randrange(2, n - 1)
        x = pow(a, d, n)
        if x in (1, n - 1):
            continue
        for _ in range(r - 1):
            x = pow(x, 2, n)
            if x == n - 1:
                break
        else:
            return False
    return True


def _gen_prime(bits, rng):
    while True:
        c = rng.getrandbits(bits) | (1 << (bits - 1)) | 1
        if _is_probable_prime(c, rng):
            return c


def _keypair(bits=1024, seed=1234):
    rng = random.Random(seed)
    e = 65537
    while True:
        p = _gen_prime(bits // 2, rng)
        q = _gen_prime(bits // 2, rng)
        if p == q:
            continue
        phi = (p - 1) * (q - 1)
        if phi % e:
            d = pow(e, -1, phi)
            return {"n": p * q, "e": e, "d": d}


KEY = _keypair()


def _sign(signing_input: bytes, key=KEY) -> bytes:
    k = (key["n"].bit_length() + 7) // 8
    t = _SHA256_DIGESTINFO + hashlib.sha256(signing_input).digest()
    em = b"\x00\x01" + b"\xff" * (k - len(t) - 3) + b"\x00" + t
    return pow(int.from_bytes(em, "big"), key["d"], key["n"]).to_bytes(k, "big")


def _jwks(key=KEY, kid="k1") -> str:
    n_bytes = key["n"].to_bytes((key["n"].bit_length() + 7) // 8, "big")
    e_bytes = key["e"].to_bytes(3, "big")
    return json.dumps({"keys": [{
        "kty": "RSA", "use": "sig", "alg": "RS256", "kid": kid,
        "n": _b64url(n_bytes), "e": _b64url(e_bytes),
    }]})


def _token(claims: dict, *, alg="RS256", kid="k1", key=KEY, tamper=False) -> str:
    header = _b64url(json.dumps({"alg": alg, "typ": "JWT", "kid": kid}).encode())
    payload = _b64url(json.dumps(claims).encode())
    signing_input = f"{header}.{payload}".encode()
    sig = _sign(signing_input, key)
    if tamper:
        sig = bytes([sig[0] ^ 1]) + sig[1:]
    return f"{header}.{payload}.{_b64url(sig)}"


def _validator(**kw):
    return JWTValidator.from_config(
        kw.pop("issuer", "https://auth.example"),
        kw.pop("audiences", ["mcp-api"]),
        jwks_json=kw.pop("jwks", _jwks()),
    )


def _claims(**over):
    c = {"iss": "https://auth.example", "aud": "mcp-api", "sub": "u1",
         "exp": time.time() + 600, "iat": time.time()}
    c.update(over)
    return c


def test_valid_token_accepted():
    v = _validator()
    claims = v.validate(_token(_claims()))
    assert claims["sub"] == "u1"


def test_bad_signature_rejected():
    v = _validator()
    with pytest.raises(JWTError, match="signature"):
        v.validate(_token(_claims(), tamper=True))


def test_wrong_key_rejected():
    other = _keypair(seed=99)
    v = _validator()
    with pytest.raises(JWTError, match="signature"):
        v.validate(_token(_claims(), key=other))


def test_alg_none_rejected():
    header = _b64url(json.dumps({"alg": "none", "typ": "JWT"}).encode())
    payload = _b64url(json.dumps(_claims()).encode())
    v = _validator()
    with pytest.raises(JWTError, match="alg"):
        v.validate(f"{header}.{payload}.")


def test_expired_and_future_rejected():
    v = _validator()
    with pytest.raises(JWTError, match="expired"):
        v.validate(_token(_claims(exp=time.time() - 3600)))
    with pytest.raises(JWTError, match="not yet valid"):
        v.validate(_token(_claims(nbf=time.time() + 3600)))


def test_issuer_and_audience_checks():
    v = _validator()
    with pytest.raises(JWTError, match="issuer"):
        v.validate(_token(_claims(iss="https://evil.example")))
    with pytest.raises(JWTError, match="audience"):
        v.validate(_token(_claims(aud="other-api")))
    # audience as a list with one match passes
    v.validate(_token(_claims(aud=["other", "mcp-api"])))


def test_jwks_parsing_skips_unsupported():
    doc = json.loads(_jwks())
    doc["keys"].append({"kty": "EC", "crv": "P-384", "x": "AA", "y": "AA"})
    doc["keys"].append({"kty": "oct", "k": "AA"})
    keys = parse_jwks(doc)
    assert len(keys) == 1 and keys[0].kid == "k1"


def test_es256_sign_verify_roundtrip():
    from aigw.mcp import jwt_auth as ja

    rng = random.Random(7)
    d = rng.randrange(1, ja._N)
    pub = ja._ec_mul(d, (ja._GX, ja._GY))

    def es_sign(signing_input: bytes) -> bytes:
        e = int.from_bytes(hashlib.sha256(signing_input).digest(), "big") % ja._N
        while True:
            k = rng.randrange(1, ja._N)
            pt = ja._ec_mul(k, (ja._GX, ja._GY))
            r = pt[0] % ja._N
            if r == 0:
                continue
            s = pow(k, -1, ja._N) * (e + r * d) % ja._N
            if s:
                return r.to_bytes(32, "big") + s.to_bytes(32, "big")

    jwks = json.dumps({"keys": [{
        "kty": "EC", "crv": "P-256", "use": "sig", "kid": "ec1",
        "x": _b64url(pub[0].to_bytes(32, "big")),
        "y": _b64url(pub[1].to_bytes(32, "big")),
    }]})
    v = JWTValidator.from_config("https://auth.example", ["mcp-api"], jwks_json=jwks)
    header = _b64url(json.dumps({"alg": "ES256", "typ": "JWT", "kid": "ec1"}).encode())
    payload = _b64url(json.dumps(_claims()).encode())
    sig = es_sign(f"{header}.{payload}".encode())
    assert v.validate(f"{header}.{payload}.{_b64url(sig)}")["sub"] == "u1"
    # tampered signature rejected
    bad = bytes([sig[5] ^ 1]) + sig[1:] if False else sig[:-1] + bytes([sig[-1] ^ 1])
    with pytest.raises(JWTError, match="signature"):
        v.validate(f"{header}.{payload}.{_b64url(bad)}")
    # RS256 token cannot pass against an EC-only JWKS
    rs_tok = _token(_claims())
    with pytest.raises(JWTError, match="signature"):
        v.validate(rs_tok)


def test_mcp_proxy_gate_end_to_end():
    from aigw.filterapi.config import MCPOAuth, MCPRoute
    from aigw.mcp.proxy import MCPProxy

    route = MCPRoute(
        name="r", backends=[],
        oauth=MCPOAuth(issuer="https://auth.example", audiences=["mcp-api"],
                       jwks=_jwks()),
    )
    proxy = MCPProxy(route, session_seed="s")

    class Req:
        def __init__(self, auth):
            self.headers = {"authorization": auth} if auth else {}

    assert proxy._authorize(Req(f"Bearer {_token(_claims())}")) is None
    denied = proxy._authorize(Req("Bearer not.a.jwt"))
    assert denied is not None and denied.status == 401
    assert proxy._authorize(Req(None)) is not None
    assert proxy._authorize(Req(f"Bearer {_token(_claims(), tamper=True)}")).status == 401


def test_startup_error_on_empty_jwks():
    with pytest.raises(JWTError, match="no usable"):
        JWTValidator.from_config("iss", [], jwks_json='{"keys": []}')


def test_crd_security_policy_oauth_translates():
    from aigw.controller import translate_yaml

    cfg = translate_yaml(
        """
apiVersion: aigateway.envoyproxy.io/v1beta1
kind: MCPRoute
metadata: {name: m, namespace: default}
spec:
  securityPolicy:
    oauth:
      issuer: https://auth.example
      audiences: [mcp-api]
      jwks:
        localJWKS:
          type: Inline
          inline: '{"keys":[{"kty":"RSA","use":"sig","kid":"k1","n":"AQAB","e":"AQAB"}]}'
  backendRefs:
    - name: b
      kind: Backend
      group: gateway.envoyproxy.io
---
apiVersion: gateway.envoyproxy.io/v1alpha1
kind: Backend
metadata: {name: b, namespace: default}
spec:
  endpoints: [{fqdn: {hostname: mcp.example, port: 80}}]
"""
    )
    o = cfg.mcp.routes[0].oauth
    assert o.issuer == "https://auth.example"
    assert o.audiences == ["mcp-api"]
    assert "kty" in o.jwks


def test_crd_remote_jwks_is_loud():
    from aigw.controller import translate_yaml
    from aigw.filterapi.config import ConfigError

    with pytest.raises(ConfigError, match="remoteJWKS"):
        translate_yaml(
            """
apiVersion: aigateway.envoyproxy.io/v1beta1
kind: MCPRoute
metadata: {name: m}
spec:
  securityPolicy:
    oauth:
      issuer: https://auth.example
      jwks:
        remoteJWKS: {uri: https://auth.example/jwks}
  backendRefs: []
"""
        )
