"""Native HS256 JWT (no external dependency).

Reference analog: mcpgateway/auth.py JWT verify + utils/create_jwt_token.
HS256 only in this image (no `cryptography` wheel for RS256 — gated with a
clear error, as the reference gates optional algorithms).
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
                 issuer: Optional[str] = None) -> str:
    if algorithm != "HS256":
        raise JWTError(f"unsupported algorithm {algorithm} (HS256 only in this build)")
    claims = dict(payload)
    now = int(time.time())
    claims.setdefault("iat", now)
    if expires_minutes is not None:
        claims.setdefault("exp", now + expires_minutes * 60)
    if audience:
        claims.setdefault("aud", audience)
    if issuer:
        claims.setdefault("iss", issuer)
    header = _b64e(json.dumps({"alg": "HS256", "typ": "JWT"}, separators=(",", ":")).encode())
    body = _b64e(json.dumps(claims, separators=(",", ":")).encode())
    signing = f"{header}.{body}".encode()
    sig = _b64e(hmac.new(secret.encode(), signing, hashlib.sha256).digest())
    return f"{header}.{body}.{sig}"


def decode_token(token: str, secret: str, audience: Optional[str] = None,
                 issuer: Optional[str] = None, verify_exp: bool = True) -> Dict[str, Any]:
    try:
        header_s, body_s, sig_s = token.split(".")
    except ValueError as exc:
        raise JWTError("malformed token") from exc
    try:
        header = json.loads(_b64d(header_s))
        claims = json.loads(_b64d(body_s))
    except Exception as exc:
        raise JWTError("undecodable token") from exc
    if header.get("alg") != "HS256":
        raise JWTError(f"unsupported alg {header.get('alg')}")
    expected = hmac.new(secret.encode(), f"{header_s}.{body_s}".encode(), hashlib.sha256).digest()
    if not hmac.compare_digest(expected, _b64d(sig_s)):
        raise JWTError("signature mismatch")
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
