"""OAuth 2.0 for upstream authentication: client-credentials AND
authorization-code flows, with encrypted token storage.

Reference analogs: services/oauth_manager.py (both flows, token refresh),
services/token_storage_service.py (encrypted stored tokens),
tool_service's token re-exchange retry (:5742 area).
"""

from __future__ import annotations

import asyncio
import base64
import hashlib
import hmac as hmac_mod
import json
import secrets
import time
from typing import Any, Dict, List, Optional

import httpx


class OAuthError(Exception):
    pass


class ClientCredentialsProvider:
    """Fetch + cache a bearer token from a token endpoint; refresh before
    expiry; `invalidate()` forces re-exchange (the 401-retry path)."""

    def __init__(self, token_url: str, client_id: str, client_secret: str,
                 scopes: Optional[List[str]] = None, audience: Optional[str] = None,
                 refresh_margin_s: float = 60.0, timeout: float = 15.0):
        self.token_url = token_url
        self.client_id = client_id
        self.client_secret = client_secret
        self.scopes = scopes or []
        self.audience = audience
        self.refresh_margin_s = refresh_margin_s
        self.timeout = timeout
        self._token: Optional[str] = None
        self._expires_at: float = 0.0
        self._lock = asyncio.Lock()
        self.exchanges = 0

    async def get_token(self, client: Optional[httpx.AsyncClient] = None) -> str:
        if self._token and time.time() < self._expires_at - self.refresh_margin_s:
            return self._token
        async with self._lock:
            if self._token and time.time() < self._expires_at - self.refresh_margin_s:
                return self._token
            data = {"grant_type": "client_credentials", "client_id": self.client_id,
                    "client_secret": self.client_secret}
            if self.scopes:
                data["scope"] = " ".join(self.scopes)
            if self.audience:
                data["audience"] = self.audience
            owns = client is None
            client = client or httpx.AsyncClient(timeout=self.timeout)
            try:
                resp = await client.post(self.token_url, data=data)
            except httpx.HTTPError as exc:
                raise OAuthError(f"token endpoint unreachable: {exc}") from exc
            finally:
                if owns:
                    await client.aclose()
            if resp.status_code >= 400:
                raise OAuthError(f"token exchange failed: HTTP {resp.status_code} {resp.text[:200]}")
            body = resp.json()
            tok = body.get("access_token")
            if not tok:
                raise OAuthError("token endpoint returned no access_token")
            self._token = tok
            self._expires_at = time.time() + float(body.get("expires_in", 3600))
            self.exchanges += 1
            return tok

    def invalidate(self) -> None:
        self._token = None
        self._expires_at = 0.0


class TokenStorage:
    """Encrypted upstream-token persistence (reference:
    services/token_storage_service.py). Token material is sealed with the
    auth_encryption_secret before hitting the oauth_tokens table."""

    def __init__(self, db, crypto):
        self.db = db
        self.crypto = crypto

    def put(self, key: str, access_token: str, refresh_token: Optional[str] = None,
            expires_in: Optional[float] = None, token_type: str = "Bearer",
            scopes: Optional[List[str]] = None) -> None:
        from sqlalchemy import select

        from ..db.models import DbOAuthToken

        with self.db.session() as s:
            row = s.execute(select(DbOAuthToken).where(DbOAuthToken.storage_key == key)).scalar_one_or_none()
            if row is None:
                row = DbOAuthToken(storage_key=key, access_token="")
                s.add(row)
            row.access_token = self.crypto.seal(access_token)
            row.refresh_token = self.crypto.seal(refresh_token) if refresh_token else row.refresh_token
            row.token_type = token_type
            row.scopes = scopes or []
            row.expires_at = time.time() + float(expires_in) if expires_in else None

    def get(self, key: str) -> Optional[Dict[str, Any]]:
        from sqlalchemy import select

        from ..db.models import DbOAuthToken

        with self.db.session() as s:
            row = s.execute(select(DbOAuthToken).where(DbOAuthToken.storage_key == key)).scalar_one_or_none()
            if row is None:
                return None
            return {
                "access_token": self.crypto.open_(row.access_token),
                "refresh_token": self.crypto.open_(row.refresh_token) if row.refresh_token else None,
                "token_type": row.token_type,
                "expires_at": row.expires_at,
                "scopes": list(row.scopes or []),
            }

    def delete(self, key: str) -> None:
        from sqlalchemy import delete as sa_delete

        from ..db.models import DbOAuthToken

        with self.db.session() as s:
            s.execute(sa_delete(DbOAuthToken).where(DbOAuthToken.storage_key == key))


def sign_state(payload: dict, secret: str) -> str:
    raw = json.dumps(payload, separators=(",", ":"), sort_keys=True).encode()
    mac = hmac_mod.new(secret.encode(), raw, hashlib.sha256).digest()[:16]
    return base64.urlsafe_b64encode(raw + mac).rstrip(b"=").decode()


def verify_state(state: str, secret: str, max_age_s: float = 600.0) -> dict:
    pad = "=" * (-len(state) % 4)
    blob = base64.urlsafe_b64decode(state + pad)
    raw, mac = blob[:-16], blob[-16:]
    want = hmac_mod.new(secret.encode(), raw, hashlib.sha256).digest()[:16]
    if not hmac_mod.compare_digest(mac, want):
        raise OAuthError("state signature mismatch")
    payload = json.loads(raw)
    if time.time() - float(payload.get("ts", 0)) > max_age_s:
        raise OAuthError("state expired")
    return payload


class AuthorizationCodeProvider:
    """Authorization-code grant for upstream auth (reference:
    oauth_manager.py auth-code flow + token exchange for upstreams).

    The admin-driven half (begin_authorization -> browser -> callback ->
    complete_authorization) persists tokens in TokenStorage; get_token()
    serves the stored access token and refreshes it with the refresh_token
    grant when expired or invalidated (the 401-retry path)."""

    def __init__(self, authorize_url: str, token_url: str, client_id: str, client_secret: str,
                 storage: TokenStorage, storage_key: str, state_secret: str,
                 scopes: Optional[List[str]] = None, redirect_uri: str = "",
                 refresh_margin_s: float = 60.0, timeout: float = 15.0):
        self.authorize_url = authorize_url
        self.token_url = token_url
        self.client_id = client_id
        self.client_secret = client_secret
        self.storage = storage
        self.storage_key = storage_key
        self.state_secret = state_secret
        self.scopes = scopes or []
        self.redirect_uri = redirect_uri
        self.refresh_margin_s = refresh_margin_s
        self.timeout = timeout
        self._lock = asyncio.Lock()
        self._forced_refresh = False
        self.exchanges = 0

    def begin_authorization(self, redirect_uri: Optional[str] = None) -> Dict[str, str]:
        """→ {url, state}: send the operator's browser to `url`."""
        state = sign_state({"k": self.storage_key, "ts": time.time(),
                            "nonce": secrets.token_urlsafe(8)}, self.state_secret)
        ru = redirect_uri or self.redirect_uri
        q = httpx.QueryParams({
            "response_type": "code", "client_id": self.client_id,
            "redirect_uri": ru, "state": state,
            **({"scope": " ".join(self.scopes)} if self.scopes else {}),
        })
        sep = "&" if "?" in self.authorize_url else "?"
        return {"url": f"{self.authorize_url}{sep}{q}", "state": state}

    async def complete_authorization(self, code: str, state: str,
                                     redirect_uri: Optional[str] = None,
                                     client: Optional[httpx.AsyncClient] = None) -> None:
        payload = verify_state(state, self.state_secret)
        if payload.get("k") != self.storage_key:
            raise OAuthError("state does not match this gateway")
        body = await self._post_token({
            "grant_type": "authorization_code", "code": code,
            "client_id": self.client_id, "client_secret": self.client_secret,
            "redirect_uri": redirect_uri or self.redirect_uri,
        }, client)
        self.storage.put(self.storage_key, body["access_token"],
                         refresh_token=body.get("refresh_token"),
                         expires_in=body.get("expires_in"),
                         token_type=body.get("token_type", "Bearer"),
                         scopes=self.scopes)

    async def get_token(self, client: Optional[httpx.AsyncClient] = None) -> str:
        ent = self.storage.get(self.storage_key)
        if ent is None:
            raise OAuthError(f"no stored authorization for {self.storage_key} "
                             "(run the authorization-code flow first)")
        fresh = ent["expires_at"] is None or time.time() < ent["expires_at"] - self.refresh_margin_s
        if fresh and not self._forced_refresh:
            return ent["access_token"]
        async with self._lock:
            ent = self.storage.get(self.storage_key)
            fresh = ent and (ent["expires_at"] is None
                             or time.time() < ent["expires_at"] - self.refresh_margin_s)
            if ent and fresh and not self._forced_refresh:
                return ent["access_token"]
            if not ent or not ent.get("refresh_token"):
                raise OAuthError("stored token expired and no refresh_token available")
            body = await self._post_token({
                "grant_type": "refresh_token", "refresh_token": ent["refresh_token"],
                "client_id": self.client_id, "client_secret": self.client_secret,
            }, client)
            self.storage.put(self.storage_key, body["access_token"],
                             refresh_token=body.get("refresh_token") or ent["refresh_token"],
                             expires_in=body.get("expires_in"),
                             token_type=body.get("token_type", "Bearer"),
                             scopes=self.scopes)
            self._forced_refresh = False
            return body["access_token"]

    def invalidate(self) -> None:
        """Upstream said 401 — force a refresh on the next get_token."""
        self._forced_refresh = True

    async def _post_token(self, data: Dict[str, str],
                          client: Optional[httpx.AsyncClient]) -> Dict[str, Any]:
        owns = client is None
        client = client or httpx.AsyncClient(timeout=self.timeout)
        try:
            resp = await client.post(self.token_url, data=data)
        except httpx.HTTPError as exc:
            raise OAuthError(f"token endpoint unreachable: {exc}") from exc
        finally:
            if owns:
                await client.aclose()
        if resp.status_code >= 400:
            raise OAuthError(f"token exchange failed: HTTP {resp.status_code} {resp.text[:200]}")
        body = resp.json()
        if not body.get("access_token"):
            raise OAuthError("token endpoint returned no access_token")
        self.exchanges += 1
        return body


def provider_from_auth_value(auth_value: dict, storage: Optional[TokenStorage] = None,
                             storage_key: str = "", state_secret: str = ""):
    """Build a provider from a gateway row's auth_value JSON
    (reference: gateway auth material, encrypted in db.py:277 EncryptedText).
    grant_type selects the flow: client_credentials (default) needs no
    storage; authorization_code needs the token store + a storage key."""
    if auth_value.get("grant_type") == "authorization_code":
        if storage is None:
            raise OAuthError("authorization_code auth requires token storage")
        return AuthorizationCodeProvider(
            authorize_url=auth_value.get("authorize_url", ""),
            token_url=auth_value["token_url"],
            client_id=auth_value["client_id"],
            client_secret=auth_value.get("client_secret", ""),
            storage=storage,
            storage_key=storage_key or auth_value.get("storage_key", auth_value["client_id"]),
            state_secret=state_secret,
            scopes=auth_value.get("scopes"),
            redirect_uri=auth_value.get("redirect_uri", ""),
        )
    return ClientCredentialsProvider(
        token_url=auth_value["token_url"],
        client_id=auth_value["client_id"],
        client_secret=auth_value.get("client_secret", ""),
        scopes=auth_value.get("scopes"),
        audience=auth_value.get("audience"),
    )
