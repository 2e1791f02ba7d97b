"""Native JWT: HS256 (HMAC) + RS256 (pure-Python RSA, auth/rsa.py).

Reference analog: mcpgateway/auth.py JWT verify (:629-915, HS* via PyJWT,
RS* via cryptography + JWKS) + utils/create_jwt_token. No external crypto
dependency: RS256 verification is RSASSA-PKCS1-v1_5 implemented directly.
"""

from __future__ import annotations

import base64
import hashlib
import hmac
import json
import time
from typing import Any, Dict, Optional


class JWTError(Exception):
    pass


def _b64e(data: bytes) -> str:
    return base64.urlsafe_b64encode(data).rstrip(b"=").decode()


def _b64d(data: str) -> bytes:
    pad = "=" * (-len(data) % 4)
    return base64.urlsafe_b64decode(data + pad)


def create_token(payload: Dict[str, Any], secret: str, algorithm: str = "HS256",
                 expires_minutes: Optional[int] = None, audience: Optional[str] = None,
                 issuer: Optional[str] = None, private_key: Optional[tuple] = None,
                 kid: Optional[str] = None) -> str:
    """Mint a token. HS256 signs with `secret`; RS256 signs with
    `private_key=(n, d)` (auth/rsa.py keypair)."""
    if algorithm not in ("HS256", "RS256"):
        raise JWTError(f"unsupported algorithm {algorithm}")
    if algorithm == "RS256" and private_key is None:
        raise JWTError("RS256 requires private_key=(n, d)")
    claims = dict(payload)
    now = int(time.time())
    claims.setdefault("iat", now)
    if expires_minutes is not None:
        claims.setdefault("exp", now + expires_minutes * 60)
    if audience:
        claims.setdefault("aud", audience)
    if issuer:
        claims.setdefault("iss", issuer)
    hdr: Dict[str, Any] = {"alg": algorithm, "typ": "JWT"}
    if kid:
        hdr["kid"] = kid
    header = _b64e(json.dumps(hdr, separators=(",", ":")).encode())
    body = _b64e(json.dumps(claims, separators=(",", ":")).encode())
    signing = f"{header}.{body}".encode()
    if algorithm == "HS256":
        sig = _b64e(hmac.new(secret.encode(), signing, hashlib.sha256).digest())
    else:
        from . import rsa as _rsa

        n, d = private_key
        sig = _b64e(_rsa.sign_pkcs1_sha256(n, d, signing))
    return f"{header}.{body}.{sig}"


def decode_token(token: str, secret: str, audience: Optional[str] = None,
                 issuer: Optional[str] = None, verify_exp: bool = True,
                 jwks=None, algorithms: tuple = ("HS256",)) -> Dict[str, Any]:
    """Verify + decode. `algorithms` is the ALLOWLIST (never trust the
    header alone — alg-confusion guard); RS256 needs `jwks` (auth/rsa.JWKSet)."""
    try:
        header_s, body_s, sig_s = token.split(".")
    except ValueError as exc:
        raise JWTError("malformed token") from exc
    try:
        header = json.loads(_b64d(header_s))
        claims = json.loads(_b64d(body_s))
    except Exception as exc:
        raise JWTError("undecodable token") from exc
    alg = header.get("alg")
    if alg not in algorithms:
        raise JWTError(f"alg {alg!r} not allowed (allowed: {algorithms})")
    signing = f"{header_s}.{body_s}".encode()
    if alg == "HS256":
        expected = hmac.new(secret.encode(), signing, hashlib.sha256).digest()
        if not hmac.compare_digest(expected, _b64d(sig_s)):
            raise JWTError("signature mismatch")
    elif alg == "RS256":
        from . import rsa as _rsa

        if jwks is None:
            raise JWTError("RS256 token but no JWKS configured")
        key = jwks.key_for(header.get("kid"))
        if key is None:
            raise JWTError(f"no JWKS key for kid {header.get('kid')!r}")
        if not _rsa.verify_pkcs1_sha256(key[0], key[1], _b64d(sig_s), signing):
            raise JWTError("signature mismatch")
    else:  # pragma: no cover - allowlist above
        raise JWTError(f"unsupported alg {alg}")
    if verify_exp and "exp" in claims and time.time() > claims["exp"]:
        raise JWTError("token expired")
    if audience is not None:
        # a configured audience REQUIRES the claim (reference: PyJWT
        # verify_aud rejects tokens missing a required aud)
        if "aud" not in claims:
            raise JWTError("audience claim missing")
        auds = claims["aud"] if isinstance(claims["aud"], list) else [claims["aud"]]
        if audience not in auds:
            raise JWTError("audience mismatch")
    if issuer is not None:
        if "iss" not in claims:
            raise JWTError("issuer claim missing")
        if claims["iss"] != issuer:
            raise JWTError("issuer mismatch")
    return claims
