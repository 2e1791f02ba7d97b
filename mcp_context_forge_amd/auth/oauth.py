"""OAuth 2.0 client-credentials for upstream authentication.

Reference analogs: services/oauth_manager.py (client-credentials flow,
token refresh), services/token_storage_service.py (cached tokens),
tool_service's token re-exchange retry (:5742 area). Auth-code flows and
RFC 7591 DCR are next-round items (they need a browser round-trip).
"""

from __future__ import annotations

import asyncio
import time
from typing import Dict, List, Optional

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


def provider_from_auth_value(auth_value: dict) -> ClientCredentialsProvider:
    """Build a provider from a gateway row's auth_value JSON
    (reference: gateway auth material, encrypted in db.py:277 EncryptedText)."""
    return ClientCredentialsProvider(
        token_url=auth_value["token_url"],
        client_id=auth_value["client_id"],
        client_secret=auth_value.get("client_secret", ""),
        scopes=auth_value.get("scopes"),
        audience=auth_value.get("audience"),
    )
