"""Snowflake tier: SQL API v2 store, key-pair JWT, OAuth PKCE exchange."""

from __future__ import annotations

import base64
import hashlib
import json

import httpx
import pytest

from agentbom_amd.api.oidc import generate_rsa_keypair, rsa_verify_pkcs1_sha256
from agentbom_amd.api.snowflake_store import (
    SnowflakeOAuthError,
    SnowflakeStore,
    build_authorize_url,
    exchange_code_for_tokens,
    keypair_jwt,
    pkce_pair,
    public_key_fingerprint,
    spki_der,
)
from agentbom_amd.utils.http_client import OfflineError, set_offline


@pytest.fixture(autouse=True)
def _online():
    set_offline(False)
    yield
    set_offline(False)


class _FakeSF:
    def __init__(self):
        self.statements: list[dict] = []
        self.answers: dict[str, dict] = {}

    def handler(self, request: httpx.Request) -> httpx.Response:
        assert request.url.path == "/api/v2/statements"
        assert request.headers["Authorization"] == "Bearer sf-tok"
        doc = json.loads(request.content.decode())
        self.statements.append(doc)
        for marker, answer in self.answers.items():
            if marker in doc["statement"]:
                return httpx.Response(200, json=answer)
        return httpx.Response(200, json={"data": []})


@pytest.fixture
def fake():
    return _FakeSF()


@pytest.fixture
def store(fake):
    client = httpx.Client(transport=httpx.MockTransport(fake.handler))
    return SnowflakeStore("https://acct.snowflakecomputing.com", "sf-tok",
                          warehouse="WH", client=client)


def test_schema_and_bound_inserts(store, fake):
    from agentbom_amd.scan.orchestrator import run_demo_scan

    store.ensure_schema()
    assert any("CREATE DATABASE IF NOT EXISTS AGENTBOM" in s["statement"]
               for s in fake.statements)
    assert any("SCAN_FINDINGS" in s["statement"] for s in fake.statements)
    assert all(s["warehouse"] == "WH" for s in fake.statements)

    report = run_demo_scan()
    n = store.insert_findings(report, tenant_id="acme")
    assert n == len(report.blast_radii) > 0
    inserts = [s for s in fake.statements if s["statement"].startswith("INSERT")]
    assert len(inserts) == n
    # every value travels as a binding, never stitched into the SQL text
    for s in inserts:
        assert "acme" not in s["statement"]
        assert s["bindings"]["3"]["value"] == "acme"
    assert any(s["bindings"]["4"]["value"].startswith("CVE-") for s in inserts)


def test_posture_query_parses_rowtype(store, fake):
    fake.answers["GROUP BY SEVERITY"] = {
        "resultSetMetaData": {"rowType": [{"name": "SEVERITY"},
                                          {"name": "FINDINGS"}]},
        "data": [["critical", "3"], ["high", "7"]]}
    rows = store.severity_posture("acme")
    assert rows == [{"SEVERITY": "critical", "FINDINGS": "3"},
                    {"SEVERITY": "high", "FINDINGS": "7"}]
    q = fake.statements[-1]
    assert q["bindings"]["1"]["value"] == "acme"


def test_http_error_raises(fake):
    client = httpx.Client(transport=httpx.MockTransport(
        lambda r: httpx.Response(422, text="bad sql")))
    store = SnowflakeStore("https://a.snowflakecomputing.com", "sf-tok",
                           client=client)
    with pytest.raises(RuntimeError, match="snowflake"):
        store.ensure_schema()


def test_offline_refused():
    set_offline(True)
    with pytest.raises(OfflineError):
        SnowflakeStore("https://a.snowflakecomputing.com", "t")


# -- key-pair JWT ----------------------------------------------------------

def _parse_der_spki(der: bytes) -> tuple[int, int]:
    """Minimal independent DER walk: return (n, e) from a SPKI blob."""
    def read_tlv(buf, off):
        tag = buf[off]
        length = buf[off + 1]
        off += 2
        if length & 0x80:
            nbytes = length & 0x7F
            length = int.from_bytes(buf[off:off + nbytes], "big")
            off += nbytes
        return tag, buf[off:off + length], off + length

    tag, body, _ = read_tlv(der, 0)
    assert tag == 0x30
    tag, alg, off = read_tlv(body, 0)
    assert tag == 0x30 and alg[:11] == bytes.fromhex("06092a864886f70d010101")
    tag, bits, _ = read_tlv(body, off)
    assert tag == 0x03 and bits[0] == 0
    tag, rsa, _ = read_tlv(bits, 1)
    assert tag == 0x30
    tag, n_raw, off = read_tlv(rsa, 0)
    assert tag == 0x02
    tag, e_raw, _ = read_tlv(rsa, off)
    assert tag == 0x02
    return int.from_bytes(n_raw, "big"), int.from_bytes(e_raw, "big")


def test_spki_der_roundtrip_and_fingerprint():
    n, e, d = generate_rsa_keypair(seed=7)
    der = spki_der(n, e)
    n2, e2 = _parse_der_spki(der)
    assert (n2, e2) == (n, e)
    fp = public_key_fingerprint(n, e)
    assert fp == "SHA256:" + base64.b64encode(hashlib.sha256(der).digest()).decode()


def test_keypair_jwt_claims_and_signature():
    n, e, d = generate_rsa_keypair(seed=7)
    tok = keypair_jwt("myorg-acct", "svc_scanner", n, e, d,
                      lifetime_s=600, now=1_700_000_000)
    h, c, s = tok.split(".")
    pad = lambda x: x + "=" * (-len(x) % 4)
    claims = json.loads(base64.urlsafe_b64decode(pad(c)))
    assert claims["sub"] == "MYORG-ACCT.SVC_SCANNER"
    assert claims["iss"].startswith("MYORG-ACCT.SVC_SCANNER.SHA256:")
    assert claims["exp"] - claims["iat"] == 600
    sig = base64.urlsafe_b64decode(pad(s))
    assert rsa_verify_pkcs1_sha256(n, e, sig, f"{h}.{c}".encode())


# -- OAuth PKCE ------------------------------------------------------------

def test_authorize_url_and_pkce():
    verifier, challenge = pkce_pair()
    assert challenge == base64.urlsafe_b64encode(
        hashlib.sha256(verifier.encode()).digest()).rstrip(b"=").decode()
    url = build_authorize_url("https://acct.snowflakecomputing.com", "cid",
                              "https://app.example/cb", "st8", challenge,
                              scope="refresh_token")
    assert url.startswith("https://acct.snowflakecomputing.com/oauth/authorize?")
    assert "code_challenge_method=S256" in url and "state=st8" in url
    assert "scope=refresh_token" in url
    with pytest.raises(SnowflakeOAuthError, match="https"):
        build_authorize_url("http://evil.example", "c", "r", "s", challenge)


def test_token_exchange_basic_auth_and_verifier():
    seen = {}

    def handler(request: httpx.Request) -> httpx.Response:
        assert request.url.path == "/oauth/token-request"
        seen["auth"] = request.headers["Authorization"]
        seen["body"] = request.content.decode()
        return httpx.Response(200, json={"access_token": "at",
                                         "refresh_token": "rt"})

    client = httpx.Client(transport=httpx.MockTransport(handler))
    body = exchange_code_for_tokens(
        "https://acct.snowflakecomputing.com", "cid", "csec",
        "https://app.example/cb", code="authcode", code_verifier="ver1",
        client=client)
    assert body["access_token"] == "at"
    assert seen["auth"] == "Basic " + base64.b64encode(b"cid:csec").decode()
    assert "code_verifier=ver1" in seen["body"]
    assert "grant_type=authorization_code" in seen["body"]

    with pytest.raises(SnowflakeOAuthError, match="secret"):
        exchange_code_for_tokens("https://a.snowflakecomputing.com", "cid",
                                 "", "r", code="c", code_verifier="v")
