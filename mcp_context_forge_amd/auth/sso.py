"""SSO (OIDC authorization-code) + RFC 7591 dynamic client registration.

Reference analogs: services/sso_service.py (GitHub/Google/Entra/Okta/
Keycloak/generic-OIDC providers), services/dcr_service.py (RFC 7591).
Provider presets mirror the reference's set; the flow is the standard
authorization-code exchange: /auth/sso/{provider}/login issues the
redirect, /auth/sso/{provider}/callback exchanges the code, fetches
userinfo, upserts the user and mints a gateway JWT.
"""

from __future__ import annotations

import base64
import hashlib
import hmac
import json
import secrets
import time
from typing import Any, Dict, List, Optional

import httpx

PROVIDER_PRESETS: Dict[str, Dict[str, str]] = {
    "github": {
        "authorize_url": "https://github.com/login/oauth/authorize",
        "token_url": "https://github.com/login/oauth/access_token",
        "userinfo_url": "https://api.github.com/user",
        "email_field": "email",
        "scopes": "read:user user:email",
    },
    "google": {
        "authorize_url": "https://accounts.google.com/o/oauth2/v2/auth",
        "token_url": "https://oauth2.googleapis.com/token",
        "userinfo_url": "https://openidconnect.googleapis.com/v1/userinfo",
        "email_field": "email",
        "scopes": "openid email profile",
    },
    # templated presets: placeholders fill from register_provider kwargs
    # (reference: sso_service.py Entra/Okta/Keycloak specifics)
    "okta": {
        "authorize_url": "https://{domain}/oauth2/v1/authorize",
        "token_url": "https://{domain}/oauth2/v1/token",
        "userinfo_url": "https://{domain}/oauth2/v1/userinfo",
        "email_field": "email",
        "scopes": "openid email profile",
    },
    "keycloak": {
        "authorize_url": "{base}/realms/{realm}/protocol/openid-connect/auth",
        "token_url": "{base}/realms/{realm}/protocol/openid-connect/token",
        "userinfo_url": "{base}/realms/{realm}/protocol/openid-connect/userinfo",
        "email_field": "email",
        "scopes": "openid email profile",
    },
    "entra": {
        "authorize_url": "https://login.microsoftonline.com/{tenant}/oauth2/v2.0/authorize",
        "token_url": "https://login.microsoftonline.com/{tenant}/oauth2/v2.0/token",
        "userinfo_url": "https://graph.microsoft.com/oidc/userinfo",
        "email_field": "mail",
        "scopes": "openid email profile",
    },
    "oidc": {"email_field": "email", "scopes": "openid email profile"},
}


class SSOError(Exception):
    def __init__(self, message: str, status: int = 400):
        self.status = status
        super().__init__(message)


class SSOService:
    def __init__(self, auth_service, settings):
        self.auth = auth_service
        self.settings = settings
        self.providers: Dict[str, Dict[str, Any]] = {}
        self._client: Optional[httpx.AsyncClient] = None

    def register_provider(self, name: str, client_id: str, client_secret: str,
                          preset: str = "oidc", **overrides: str) -> Dict[str, Any]:
        cfg = dict(PROVIDER_PRESETS.get(preset, PROVIDER_PRESETS["oidc"]))
        cfg.update(overrides)
        for req in ("authorize_url", "token_url", "userinfo_url"):
            if req not in cfg:
                raise SSOError(f"provider {name}: missing {req}")
            if "{" in cfg[req]:
                # templated preset (entra {tenant}, okta {domain},
                # keycloak {base}/{realm}) fills from the kwargs
                try:
                    cfg[req] = cfg[req].format(**overrides)
                except KeyError as exc:
                    raise SSOError(f"provider {name}: preset {preset!r} needs {exc.args[0]}=") from exc
        cfg.update({"client_id": client_id, "client_secret": client_secret, "name": name})
        self.providers[name] = cfg
        return {k: v for k, v in cfg.items() if k != "client_secret"}

    # -- state signing (CSRF protection on the redirect round-trip) ----------
    def _sign_state(self, payload: dict) -> str:
        body = base64.urlsafe_b64encode(json.dumps(payload).encode()).rstrip(b"=").decode()
        sig = hmac.new(self.settings.jwt_secret_key.encode(), body.encode(), hashlib.sha256).hexdigest()[:32]
        return f"{body}.{sig}"

    def _verify_state(self, state: str) -> dict:
        try:
            body, sig = state.rsplit(".", 1)
            expect = hmac.new(self.settings.jwt_secret_key.encode(), body.encode(), hashlib.sha256).hexdigest()[:32]
            if not hmac.compare_digest(sig, expect):
                raise ValueError("bad signature")
            pad = "=" * (-len(body) % 4)
            payload = json.loads(base64.urlsafe_b64decode(body + pad))
            if payload.get("exp", 0) < time.time():
                raise ValueError("state expired")
            return payload
        except ValueError as exc:
            raise SSOError(f"invalid state: {exc}", status=403) from exc

    def login_url(self, provider: str, redirect_uri: str) -> str:
        cfg = self.providers.get(provider)
        if cfg is None:
            raise SSOError(f"unknown SSO provider {provider}", status=404)
        state = self._sign_state({"p": provider, "n": secrets.token_urlsafe(8),
                                  "exp": time.time() + 600})
        from urllib.parse import urlencode

        q = urlencode({"response_type": "code", "client_id": cfg["client_id"],
                       "redirect_uri": redirect_uri, "scope": cfg.get("scopes", "openid email"),
                       "state": state})
        return f"{cfg['authorize_url']}?{q}"

    async def handle_callback(self, provider: str, code: str, state: str,
                              redirect_uri: str) -> Dict[str, Any]:
        cfg = self.providers.get(provider)
        if cfg is None:
            raise SSOError(f"unknown SSO provider {provider}", status=404)
        st = self._verify_state(state)
        if st.get("p") != provider:
            raise SSOError("state/provider mismatch", status=403)
        if self._client is None:
            self._client = httpx.AsyncClient(timeout=15.0)
        try:
            resp = await self._client.post(cfg["token_url"], data={
                "grant_type": "authorization_code", "code": code,
                "client_id": cfg["client_id"], "client_secret": cfg["client_secret"],
                "redirect_uri": redirect_uri,
            }, headers={"accept": "application/json"})
        except httpx.HTTPError as exc:
            raise SSOError(f"token exchange failed: {exc}", status=502) from exc
        if resp.status_code >= 400:
            raise SSOError(f"token endpoint error {resp.status_code}", status=502)
        access_token = resp.json().get("access_token")
        if not access_token:
            raise SSOError("no access_token from provider", status=502)
        ui = await self._client.get(cfg["userinfo_url"],
                                    headers={"authorization": f"Bearer {access_token}"})
        if ui.status_code >= 400:
            raise SSOError(f"userinfo error {ui.status_code}", status=502)
        info = ui.json()
        email = info.get(cfg.get("email_field", "email")) or info.get("email")
        if not email:
            raise SSOError("provider returned no email", status=502)
        # upsert user (SSO users get a random local password)
        from .service import hash_password
        from ..db.models import DbUser

        with self.auth.db.session() as s:
            u = s.get(DbUser, email)
            if u is None:
                s.add(DbUser(email=email, password_hash=hash_password(secrets.token_urlsafe(24)),
                             full_name=info.get("name", ""), is_admin=False))
        from . import jwt as jwt_mod

        token = jwt_mod.create_token({"sub": email, "sso": provider}, self.settings.jwt_secret_key,
                                     expires_minutes=self.settings.token_expiry,
                                     audience=self.settings.jwt_audience, issuer=self.settings.jwt_issuer)
        return {"access_token": token, "token_type": "bearer", "email": email, "provider": provider}

    async def aclose(self) -> None:
        if self._client is not None:
            await self._client.aclose()


async def dcr_register(registration_endpoint: str, client_name: str,
                       redirect_uris: Optional[List[str]] = None,
                       grant_types: Optional[List[str]] = None,
                       initial_access_token: Optional[str] = None,
                       timeout: float = 15.0) -> Dict[str, Any]:
    """RFC 7591 dynamic client registration (reference: dcr_service.py) —
    the gateway registers itself as an OAuth client with an upstream AS."""
    headers = {"content-type": "application/json"}
    if initial_access_token:
        headers["authorization"] = f"Bearer {initial_access_token}"
    metadata = {
        "client_name": client_name,
        "grant_types": grant_types or ["client_credentials"],
        "token_endpoint_auth_method": "client_secret_post",
    }
    if redirect_uris:
        metadata["redirect_uris"] = redirect_uris
    async with httpx.AsyncClient(timeout=timeout) as client:
        resp = await client.post(registration_endpoint, json=metadata, headers=headers)
    if resp.status_code not in (200, 201):
        raise SSOError(f"DCR failed: HTTP {resp.status_code} {resp.text[:200]}", status=502)
    body = resp.json()
    if "client_id" not in body:
        raise SSOError("DCR response missing client_id", status=502)
    return body
