"""Auth breadth: RS256/JWKS verification (pure-Python RSA) and the upstream
OAuth authorization-code flow with encrypted token storage
(reference: mcpgateway/auth.py:629-915, services/oauth_manager.py,
services/token_storage_service.py)."""

import json
import time

import httpx
import pytest

from mcp_context_forge_amd.auth import jwt as jwt_mod
from mcp_context_forge_amd.auth import rsa as rsa_mod
from mcp_context_forge_amd.auth.oauth import (AuthorizationCodeProvider, OAuthError, TokenStorage,
                                              provider_from_auth_value, sign_state, verify_state)
from mcp_context_forge_amd.config import Settings


@pytest.fixture(scope="module")
def keypair():
    return rsa_mod.generate_keypair(bits=1024)  # small + fast for tests


# ---------------------------------------------------------------- RSA / RS256


def test_rsa_sign_verify_roundtrip(keypair):
    n, e, d = keypair
    msg = b"the quick brown fox"
    sig = rsa_mod.sign_pkcs1_sha256(n, d, msg)
    assert rsa_mod.verify_pkcs1_sha256(n, e, sig, msg)
    assert not rsa_mod.verify_pkcs1_sha256(n, e, sig, msg + b"!")
    assert not rsa_mod.verify_pkcs1_sha256(n, e, sig[:-1] + b"\x00", msg)


def test_jwk_roundtrip(keypair):
    n, e, _ = keypair
    jwk = rsa_mod.public_to_jwk(n, e, kid="k1")
    n2, e2 = rsa_mod.jwk_to_public(jwk)
    assert (n2, e2) == (n, e)


def test_rs256_jwt_with_jwks(keypair):
    n, e, d = keypair
    tok = jwt_mod.create_token({"sub": "alice"}, "", algorithm="RS256",
                               private_key=(n, d), kid="k1",
                               audience="api", issuer="idp")
    jwks = rsa_mod.JWKSet(keys=[rsa_mod.public_to_jwk(n, e, kid="k1")])
    claims = jwt_mod.decode_token(tok, "unused", audience="api", issuer="idp",
                                  jwks=jwks, algorithms=("RS256",))
    assert claims["sub"] == "alice"
    # wrong key → rejected
    n2, e2, _ = rsa_mod.generate_keypair(bits=1024)
    bad = rsa_mod.JWKSet(keys=[rsa_mod.public_to_jwk(n2, e2, kid="k1")])
    with pytest.raises(jwt_mod.JWTError, match="signature mismatch"):
        jwt_mod.decode_token(tok, "unused", jwks=bad, algorithms=("RS256",))


def test_alg_confusion_rejected(keypair):
    """A token claiming HS256 must NOT pass when only RS256 is allowed —
    the classic RS->HS downgrade (attacker signs HMAC with the public key)."""
    n, e, d = keypair
    hs = jwt_mod.create_token({"sub": "mallory"}, "guessable-public-material")
    jwks = rsa_mod.JWKSet(keys=[rsa_mod.public_to_jwk(n, e, kid="k1")])
    with pytest.raises(jwt_mod.JWTError, match="not allowed"):
        jwt_mod.decode_token(hs, "guessable-public-material", jwks=jwks, algorithms=("RS256",))


def test_auth_service_accepts_rs256(keypair, run):
    from mcp_context_forge_amd.auth.service import AuthService
    from mcp_context_forge_amd.db.engine import Database

    n, e, d = keypair
    jwks_json = json.dumps({"keys": [rsa_mod.public_to_jwk(n, e, kid="idp-1")]})
    s = Settings(database_url="sqlite://", jwt_accepted_algorithms=["HS256", "RS256"],
                 jwks_inline=jwks_json)
    db = Database("sqlite://")
    db.migrate()
    auth = AuthService(db, s)
    tok = jwt_mod.create_token({"sub": "idp-user@corp"}, "", algorithm="RS256",
                               private_key=(n, d), kid="idp-1",
                               audience=s.jwt_audience, issuer=s.jwt_issuer)
    ctx = auth.authenticate(f"Bearer {tok}")
    assert ctx.user == "idp-user@corp"
    assert ctx.auth_method == "jwt"
    # HS256 still works side by side
    hs = jwt_mod.create_token({"sub": "local"}, s.jwt_secret_key,
                              audience=s.jwt_audience, issuer=s.jwt_issuer)
    assert auth.authenticate(f"Bearer {hs}").user == "local"
    db.close()


# ---------------------------------------------------------------- auth-code


class FakeIdP:
    """Token endpoint double: one valid code, refresh rotation."""

    def __init__(self):
        self.codes = {"good-code": True}
        self.refresh_gen = 0
        self.access_gen = 0

    def handler(self, request: httpx.Request) -> httpx.Response:
        form = dict(httpx.QueryParams(request.content.decode()))
        if form.get("grant_type") == "authorization_code":
            if not self.codes.pop(form.get("code"), None):
                return httpx.Response(400, json={"error": "invalid_grant"})
            self.access_gen += 1
            return httpx.Response(200, json={
                "access_token": f"AT-{self.access_gen}", "refresh_token": f"RT-{self.refresh_gen}",
                "expires_in": 3600, "token_type": "Bearer"})
        if form.get("grant_type") == "refresh_token":
            if form.get("refresh_token") != f"RT-{self.refresh_gen}":
                return httpx.Response(400, json={"error": "invalid_grant"})
            self.access_gen += 1
            self.refresh_gen += 1
            return httpx.Response(200, json={
                "access_token": f"AT-{self.access_gen}", "refresh_token": f"RT-{self.refresh_gen}",
                "expires_in": 3600})
        return httpx.Response(400, json={"error": "unsupported_grant_type"})


def _storage():
    from mcp_context_forge_amd.auth.crypto import EncryptionService
    from mcp_context_forge_amd.db.engine import Database

    db = Database("sqlite://")
    db.migrate()
    return db, TokenStorage(db, EncryptionService("seal-key"))


def test_authorization_code_flow_end_to_end(run):
    async def go():
        db, storage = _storage()
        idp = FakeIdP()
        client = httpx.AsyncClient(transport=httpx.MockTransport(idp.handler))
        prov = AuthorizationCodeProvider(
            authorize_url="https://idp.example/authorize", token_url="https://idp.example/token",
            client_id="cid", client_secret="cs", storage=storage, storage_key="gateway:g1",
            state_secret="state-secret", scopes=["mcp.read"],
            redirect_uri="https://gw.example/oauth/upstream/callback")
        # 1) browser half
        begin = prov.begin_authorization()
        assert begin["url"].startswith("https://idp.example/authorize?")
        assert "client_id=cid" in begin["url"] and "state=" in begin["url"]
        # 2) callback half: code exchange persists tokens
        await prov.complete_authorization("good-code", begin["state"], client=client)
        assert await prov.get_token(client) == "AT-1"
        # 3) 401-retry path: invalidate forces the refresh grant
        prov.invalidate()
        assert await prov.get_token(client) == "AT-2"
        # 4) expiry-based refresh: rewind the stored expiry
        ent = storage.get("gateway:g1")
        storage.put("gateway:g1", ent["access_token"], refresh_token=ent["refresh_token"],
                    expires_in=-10)
        assert await prov.get_token(client) == "AT-3"
        assert idp.access_gen == 3
        await client.aclose()
        db.close()

    run(go())


def test_stored_tokens_are_sealed(run):
    async def go():
        db, storage = _storage()
        storage.put("k1", "SECRET-ACCESS", refresh_token="SECRET-REFRESH", expires_in=60)
        # raw row never contains plaintext
        from sqlalchemy import select

        from mcp_context_forge_amd.db.models import DbOAuthToken

        with db.session() as s:
            row = s.execute(select(DbOAuthToken)).scalar_one()
            assert "SECRET-ACCESS" not in (row.access_token or "")
            assert "SECRET-REFRESH" not in (row.refresh_token or "")
        ent = storage.get("k1")
        assert ent["access_token"] == "SECRET-ACCESS"
        assert ent["refresh_token"] == "SECRET-REFRESH"
        db.close()

    run(go())


def test_state_signing():
    st = sign_state({"k": "gateway:g1", "ts": time.time()}, "s1")
    assert verify_state(st, "s1")["k"] == "gateway:g1"
    with pytest.raises(OAuthError):
        verify_state(st, "s2")
    old = sign_state({"k": "x", "ts": time.time() - 3600}, "s1")
    with pytest.raises(OAuthError, match="expired"):
        verify_state(old, "s1")


def test_provider_from_auth_value_selects_flow():
    db, storage = _storage()
    cc = provider_from_auth_value({"token_url": "https://t", "client_id": "c"})
    from mcp_context_forge_amd.auth.oauth import ClientCredentialsProvider

    assert isinstance(cc, ClientCredentialsProvider)
    ac = provider_from_auth_value(
        {"grant_type": "authorization_code", "token_url": "https://t", "client_id": "c",
         "authorize_url": "https://a"},
        storage=storage, storage_key="gateway:g9", state_secret="s")
    assert isinstance(ac, AuthorizationCodeProvider)
    with pytest.raises(OAuthError):
        provider_from_auth_value({"grant_type": "authorization_code",
                                  "token_url": "https://t", "client_id": "c"})
    db.close()


def test_gateway_oauth_routes(run):
    """HTTP half: authorize endpoint returns the provider URL; the callback
    completes against a mocked IdP and seals tokens."""
    import base64

    from mcp_context_forge_amd.engine import GatewayEngine
    from mcp_context_forge_amd.transports.http_app import build_app

    BASIC = {"Authorization": "Basic " + base64.b64encode(b"admin:changeme").decode()}

    async def go():
        s = Settings(database_url="sqlite://", federation_enabled=False, auth_required=True,
                     plugins_enabled=False, gpu_enabled=False)
        e = GatewayEngine(s)
        # a gateway row with auth-code OAuth config (registered deferred so
        # no connection is attempted now)
        gw = await e.gateway_service.register_gateway(
            name="oauth-peer", url="http://peer/mcp", auth_type="oauth",
            auth_value=json.dumps({"grant_type": "authorization_code",
                                   "authorize_url": "https://idp.example/authorize",
                                   "token_url": "https://idp.example/token",
                                   "client_id": "cid", "client_secret": "cs",
                                   "redirect_uri": "http://gw/oauth/upstream/callback"}),
            defer=True, owner_rank=0)
        app = build_app(e)
        async with app.router.lifespan_context(app):
            async with httpx.AsyncClient(transport=httpx.ASGITransport(app=app),
                                         base_url="http://gw") as c:
                r = await c.post(f"/gateways/{gw['id']}/oauth/authorize", headers=BASIC)
                assert r.status_code == 200, r.text
                assert r.json()["url"].startswith("https://idp.example/authorize?")
                state = r.json()["state"]
                # swap the provider's HTTP for the fake IdP, then hit the callback
                idp = FakeIdP()
                prov = e.gateway_service.oauth_provider_for(gw["id"])
                mock = httpx.AsyncClient(transport=httpx.MockTransport(idp.handler))
                orig = prov._post_token

                async def patched(data, client=None):
                    return await orig(data, mock)

                prov._post_token = patched
                r2 = await c.get("/oauth/upstream/callback",
                                 params={"code": "good-code", "state": state})
                assert r2.status_code == 200, r2.text
                assert r2.json()["gateway_id"] == gw["id"]
                assert await prov.get_token(mock) == "AT-1"
                await mock.aclose()

    run(go())
