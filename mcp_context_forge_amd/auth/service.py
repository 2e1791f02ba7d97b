"""Authentication + RBAC.

Reference analogs: mcpgateway/auth.py (validate_token_user :1209, basic
auth, API-token lookup + revocation :629-915), services/email_auth_service
(Argon2id — here PBKDF2-SHA256, the strongest stdlib KDF),
middleware/rbac.py require_permission decorators, services/permission_service.
"""

from __future__ import annotations

import base64
import hashlib
import hmac
import os
import secrets
from dataclasses import dataclass, field
from typing import List, Optional, Set, Tuple

from sqlalchemy import select

from ..config import Settings
from ..db.engine import Database
from ..db.models import DbApiToken, DbRole, DbTeam, DbTeamMember, DbUser, DbUserRole, utcnow
from . import jwt as jwt_mod

PBKDF2_ITERS = 150_000


def hash_password(password: str, salt: Optional[bytes] = None) -> str:
    salt = salt or os.urandom(16)
    dk = hashlib.pbkdf2_hmac("sha256", password.encode(), salt, PBKDF2_ITERS)
    return f"pbkdf2${PBKDF2_ITERS}${base64.b64encode(salt).decode()}${base64.b64encode(dk).decode()}"


def verify_password(password: str, stored: str) -> bool:
    try:
        scheme, iters, salt_b64, dk_b64 = stored.split("$")
        if scheme != "pbkdf2":
            return False
        dk = hashlib.pbkdf2_hmac("sha256", password.encode(), base64.b64decode(salt_b64), int(iters))
        return hmac.compare_digest(dk, base64.b64decode(dk_b64))
    except Exception:
        return False


@dataclass
class AuthContext:
    """Per-request identity (reference: auth_context.py)."""

    user: str
    is_admin: bool = False
    auth_method: str = "anonymous"  # jwt | basic | api_token | anonymous
    teams: List[str] = field(default_factory=list)
    scopes: List[str] = field(default_factory=list)
    server_id: Optional[str] = None  # token scoped to one virtual server
    credential: Optional[str] = None  # usage-accounting key (token id / method)


class AuthError(Exception):
    def __init__(self, message: str, status: int = 401):
        self.status = status
        super().__init__(message)


class PermissionError_(Exception):
    pass


DEFAULT_PERMISSIONS = {
    # permission strings follow the reference's "<entity>.<action>" shape
    "admin": ["*"],
    "developer": ["tools.read", "tools.invoke", "resources.read", "prompts.read", "servers.read"],
    "viewer": ["tools.read", "resources.read", "prompts.read", "servers.read"],
}


class AuthService:
    def __init__(self, db: Database, settings: Settings, token_blocklist=None):
        self.db = db
        self.settings = settings
        self.token_blocklist = token_blocklist  # reference: services/token_blocklist.py
        self._perm_cache: dict = {}
        self._revocations = 0  # bumped on API-token revocation
        self._jwks = None      # lazy JWKSet when RS256 is accepted

    def jwks(self):
        """JWKSet from settings (inline JSON / file / URL), lazily built."""
        if self._jwks is None:
            from . import rsa as rsa_mod

            s = self.settings
            if s.jwks_inline:
                self._jwks = rsa_mod.JWKSet.from_json(s.jwks_inline)
            elif s.jwks_file:
                from pathlib import Path

                self._jwks = rsa_mod.JWKSet.from_json(Path(s.jwks_file).read_text())
            elif s.jwks_url:
                self._jwks = rsa_mod.JWKSet(url=s.jwks_url)
                self._jwks.refresh_from_url()
            else:
                self._jwks = rsa_mod.JWKSet()
        return self._jwks

    @property
    def revocation_epoch(self) -> int:
        """Monotone counter that changes on ANY token revocation; auth-context
        caches (RpcFastPath, edge workers) key on it so revoked tokens stop
        working immediately instead of riding out a TTL."""
        bl = self.token_blocklist.version if self.token_blocklist is not None else 0
        return self._revocations + bl

    # -- bootstrap -------------------------------------------------------------
    def bootstrap_admin(self) -> None:
        """Seed platform admin + default roles (reference: bootstrap_db.py:42-45)."""
        with self.db.session() as s:
            if s.get(DbUser, self.settings.platform_admin_email) is None:
                s.add(DbUser(email=self.settings.platform_admin_email,
                             password_hash=hash_password(self.settings.platform_admin_password),
                             full_name="Platform Admin", is_admin=True))
            for name, perms in DEFAULT_PERMISSIONS.items():
                existing = s.execute(select(DbRole).where(DbRole.name == name)).scalar_one_or_none()
                if existing is None:
                    s.add(DbRole(name=name, permissions=perms))

    # -- user management ---------------------------------------------------------
    def create_user(self, email: str, password: str, full_name: str = "", is_admin: bool = False) -> None:
        with self.db.session() as s:
            s.add(DbUser(email=email, password_hash=hash_password(password), full_name=full_name, is_admin=is_admin))

    def verify_user(self, email: str, password: str) -> Optional[AuthContext]:
        with self.db.session() as s:
            u = s.get(DbUser, email)
            if u is None or not u.is_active or not verify_password(password, u.password_hash):
                return None
            return AuthContext(user=email, is_admin=u.is_admin, auth_method="basic", teams=self._teams(s, email))

    def _teams(self, s, email: str) -> List[str]:
        rows = s.execute(select(DbTeamMember.team_id).where(DbTeamMember.user_email == email)).all()
        return [r[0] for r in rows]

    # -- API tokens (reference: token_catalog / db.py:5338) ------------------------
    def create_api_token(self, user_email: str, name: str, scopes: Optional[List[str]] = None,
                         server_id: Optional[str] = None, expires_minutes: Optional[int] = None) -> str:
        raw = "mcpg_" + secrets.token_urlsafe(32)
        token_hash = hashlib.sha256(raw.encode()).hexdigest()
        import datetime

        exp = None
        if expires_minutes:
            exp = utcnow() + datetime.timedelta(minutes=expires_minutes)
        with self.db.session() as s:
            s.add(DbApiToken(user_email=user_email, name=name, token_hash=token_hash,
                             scopes=scopes or [], server_id=server_id, expires_at=exp))
        return raw

    def verify_api_token(self, raw: str) -> Optional[AuthContext]:
        token_hash = hashlib.sha256(raw.encode()).hexdigest()
        with self.db.session() as s:
            t = s.execute(select(DbApiToken).where(DbApiToken.token_hash == token_hash)).scalar_one_or_none()
            if t is None or t.revoked:
                return None
            if t.expires_at is not None and utcnow() > t.expires_at:
                return None
            t.last_used_at = utcnow()
            u = s.get(DbUser, t.user_email)
            return AuthContext(user=t.user_email, is_admin=bool(u and u.is_admin), auth_method="api_token",
                               scopes=list(t.scopes or []), server_id=t.server_id,
                               teams=self._teams(s, t.user_email), credential=f"token:{t.id}")

    def revoke_api_token(self, token_id: str) -> bool:
        with self.db.session() as s:
            t = s.get(DbApiToken, token_id)
            if t is None:
                return False
            t.revoked = True
            self._revocations += 1
            return True

    def list_api_tokens(self, user_email: str) -> List[dict]:
        with self.db.session() as s:
            rows = s.execute(select(DbApiToken).where(DbApiToken.user_email == user_email)).scalars().all()
            return [{"id": t.id, "name": t.name, "revoked": t.revoked,
                     "server_id": t.server_id, "scopes": t.scopes} for t in rows]

    # -- request authentication ---------------------------------------------------
    def authenticate(self, authorization: Optional[str], basic_ok: bool = True) -> AuthContext:
        """Resolve an Authorization header to an AuthContext (reference:
        auth.py validate_token_user :1209 + basic + api-token chain)."""
        if not self.settings.auth_required:
            return AuthContext(user="anonymous", is_admin=True, auth_method="anonymous")
        if not authorization:
            raise AuthError("Not authenticated")
        scheme, _, value = authorization.partition(" ")
        scheme = scheme.lower()
        if scheme == "bearer" and value:
            if value.startswith("mcpg_"):
                ctx = self.verify_api_token(value)
                if ctx is None:
                    raise AuthError("Invalid or revoked API token")
                return ctx
            algs = tuple(self.settings.jwt_accepted_algorithms or ["HS256"])
            try:
                claims = jwt_mod.decode_token(value, self.settings.jwt_secret_key,
                                              audience=self.settings.jwt_audience,
                                              issuer=self.settings.jwt_issuer,
                                              jwks=self.jwks() if "RS256" in algs else None,
                                              algorithms=algs)
            except jwt_mod.JWTError as exc:
                raise AuthError(f"Invalid token: {exc}") from exc
            if self.token_blocklist is not None and self.token_blocklist.is_blocked(claims.get("jti")):
                raise AuthError("Token revoked")
            user = claims.get("sub") or claims.get("username") or claims.get("email") or "unknown"
            return AuthContext(user=user, is_admin=bool(claims.get("admin") or user == self.settings.platform_admin_email
                                                        or user == self.settings.basic_auth_user),
                               auth_method="jwt", scopes=claims.get("scopes") or [],
                               server_id=(claims.get("server_id")),
                               credential=f"jwt:{claims.get('jti') or user}")
        if scheme == "basic" and basic_ok and value:
            try:
                decoded = base64.b64decode(value).decode()
                username, _, password = decoded.partition(":")
            except Exception as exc:
                raise AuthError("Invalid basic credentials") from exc
            if username == self.settings.basic_auth_user and hmac.compare_digest(password, self.settings.basic_auth_password):
                return AuthContext(user=username, is_admin=True, auth_method="basic")
            ctx = self.verify_user(username, password)
            if ctx is None:
                raise AuthError("Invalid credentials")
            return ctx
        raise AuthError("Unsupported authorization scheme")

    # -- RBAC ----------------------------------------------------------------------
    def permissions_for(self, ctx: AuthContext) -> Set[str]:
        if ctx.is_admin:
            return {"*"}
        key = ctx.user
        if key in self._perm_cache:
            return self._perm_cache[key]
        perms: Set[str] = set()
        with self.db.session() as s:
            rows = s.execute(
                select(DbRole.permissions).join(DbUserRole, DbUserRole.role_id == DbRole.id)
                .where(DbUserRole.user_email == ctx.user)).all()
            for (p,) in rows:
                perms.update(p or [])
        if not perms:
            perms = set(DEFAULT_PERMISSIONS["developer"])  # default grant (reference default role)
        self._perm_cache[key] = perms
        return perms

    def require_permission(self, ctx: AuthContext, permission: str) -> None:
        perms = self.permissions_for(ctx)
        if "*" in perms or permission in perms:
            return
        entity = permission.split(".")[0]
        if f"{entity}.*" in perms:
            return
        raise PermissionError_(f"missing permission {permission}")

    def assign_role(self, user_email: str, role_name: str) -> None:
        with self.db.session() as s:
            role = s.execute(select(DbRole).where(DbRole.name == role_name)).scalar_one()
            s.add(DbUserRole(user_email=user_email, role_id=role.id))
        self._perm_cache.pop(user_email, None)

    # -- teams ----------------------------------------------------------------------
    def create_team(self, name: str, created_by: str, is_personal: bool = False) -> dict:
        from ..utils import slugify

        with self.db.session() as s:
            team = DbTeam(name=name, slug=slugify(name), created_by=created_by, is_personal=is_personal)
            s.add(team)
            s.flush()
            s.add(DbTeamMember(team_id=team.id, user_email=created_by, role="owner"))
            return {"id": team.id, "name": team.name, "slug": team.slug}

    def add_team_member(self, team_id: str, email: str, role: str = "member") -> None:
        with self.db.session() as s:
            s.add(DbTeamMember(team_id=team_id, user_email=email, role=role))
