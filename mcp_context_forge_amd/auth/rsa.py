"""Pure-Python RSASSA-PKCS1-v1_5 (SHA-256) for RS256 JWT verification.

Reference analog: mcpgateway/auth.py:629-915 (RS*/ES* verify via PyJWT +
cryptography). This image ships no `cryptography` wheel, so verification is
implemented directly: s^e mod n and an exact EMSA-PKCS1-v1_5 encoding
compare — constant-structure, no padding-oracle surface (verification only
compares a locally computed encoding against the decrypted signature).

Signing + keygen (Miller-Rabin) are included for token minting and tests;
production deployments normally verify against an IdP's JWKS and never
hold the private key.
"""

from __future__ import annotations

import base64
import hashlib
import json
import secrets
from typing import Dict, Optional, Tuple

# DigestInfo prefix for SHA-256 (RFC 8017 §9.2 note 1)
_SHA256_PREFIX = bytes.fromhex("3031300d060960864801650304020105000420")


def _b64u_decode(data: str) -> bytes:
    pad = "=" * (-len(data) % 4)
    return base64.urlsafe_b64decode(data + pad)


def _b64u_encode(data: bytes) -> str:
    return base64.urlsafe_b64encode(data).rstrip(b"=").decode()


def _os2ip(b: bytes) -> int:
    return int.from_bytes(b, "big")


def _i2osp(x: int, n: int) -> bytes:
    return x.to_bytes(n, "big")


def _emsa_pkcs1_v15(message: bytes, em_len: int) -> bytes:
    """EMSA-PKCS1-v1_5 encoding with SHA-256 (RFC 8017 §9.2)."""
    digest = hashlib.sha256(message).digest()
    t = _SHA256_PREFIX + digest
    if em_len < len(t) + 11:
        raise ValueError("intended encoded message length too short")
    ps = b"\xff" * (em_len - len(t) - 3)
    return b"\x00\x01" + ps + b"\x00" + t


def verify_pkcs1_sha256(n: int, e: int, signature: bytes, message: bytes) -> bool:
    """RSASSA-PKCS1-v1_5 verify: full encoded-message comparison."""
    k = (n.bit_length() + 7) // 8
    if len(signature) != k:
        return False
    s = _os2ip(signature)
    if s >= n:
        return False
    em = _i2osp(pow(s, e, n), k)
    try:
        expected = _emsa_pkcs1_v15(message, k)
    except ValueError:
        return False
    return secrets.compare_digest(em, expected)


def sign_pkcs1_sha256(n: int, d: int, message: bytes) -> bytes:
    k = (n.bit_length() + 7) // 8
    em = _emsa_pkcs1_v15(message, k)
    return _i2osp(pow(_os2ip(em), d, n), k)


def jwk_to_public(jwk: Dict) -> Tuple[int, int]:
    """RFC 7517 RSA JWK → (n, e)."""
    if jwk.get("kty") != "RSA":
        raise ValueError(f"unsupported kty {jwk.get('kty')!r}")
    return _os2ip(_b64u_decode(jwk["n"])), _os2ip(_b64u_decode(jwk["e"]))


def public_to_jwk(n: int, e: int, kid: Optional[str] = None) -> Dict:
    k = (n.bit_length() + 7) // 8
    jwk = {"kty": "RSA", "alg": "RS256", "use": "sig",
           "n": _b64u_encode(_i2osp(n, k)), "e": _b64u_encode(_i2osp(e, (e.bit_length() + 7) // 8))}
    if kid:
        jwk["kid"] = kid
    return jwk


# ---------------------------------------------------------------- keygen

_SMALL_PRIMES = [2, 3, 5, 7, 11, 13, 17, 19, 23, 29, 31, 37, 41, 43, 47, 53, 59, 61, 67]


def _is_probable_prime(n: int, rounds: int = 40) -> bool:
    if n < 2:
        return False
    for p in _SMALL_PRIMES:
        if n % p == 0:
            return n == p
    d, r = n - 1, 0
    while d % 2 == 0:
        d //= 2
        r += 1
    for _ in range(rounds):
        a = secrets.randbelow(n - 3) + 2
        x = pow(a, d, n)
        if x in (1, n - 1):
            continue
        for _ in range(r - 1):
            x = (x * x) % n
            if x == n - 1:
                break
        else:
            return False
    return True


def _random_prime(bits: int) -> int:
    while True:
        cand = secrets.randbits(bits) | (1 << (bits - 1)) | 1
        if _is_probable_prime(cand):
            return cand


def generate_keypair(bits: int = 2048, e: int = 65537) -> Tuple[int, int, int]:
    """→ (n, e, d). Test/minting helper — a few seconds at 2048 bits."""
    while True:
        p = _random_prime(bits // 2)
        q = _random_prime(bits - bits // 2)
        if p == q:
            continue
        n = p * q
        phi = (p - 1) * (q - 1)
        try:
            d = pow(e, -1, phi)
        except ValueError:
            continue
        return n, e, d


class JWKSet:
    """Key set for RS256 verification; from inline JSON, a file, or a URL
    (reference: auth.py jwks_uri fetch + cache)."""

    def __init__(self, keys: Optional[list] = None, url: Optional[str] = None,
                 cache_ttl_s: float = 3600.0):
        self._by_kid: Dict[Optional[str], Tuple[int, int]] = {}
        self.url = url
        self.cache_ttl_s = cache_ttl_s
        self._fetched_at = 0.0
        for jwk in keys or []:
            self.add(jwk)

    def add(self, jwk: Dict) -> None:
        try:
            self._by_kid[jwk.get("kid")] = jwk_to_public(jwk)
        except (ValueError, KeyError):
            pass  # skip non-RSA keys (EC etc.)

    @classmethod
    def from_json(cls, text: str) -> "JWKSet":
        obj = json.loads(text)
        return cls(keys=obj.get("keys", []) if isinstance(obj, dict) else obj)

    def refresh_from_url(self, client=None) -> None:
        """Synchronous JWKS fetch with TTL (verification is sync)."""
        import time as _t

        if not self.url or _t.monotonic() - self._fetched_at < self.cache_ttl_s:
            return
        import httpx

        owns = client is None
        client = client or httpx.Client(timeout=10.0)
        try:
            resp = client.get(self.url)
            if resp.status_code == 200:
                for jwk in resp.json().get("keys", []):
                    self.add(jwk)
                self._fetched_at = _t.monotonic()
        finally:
            if owns:
                client.close()

    def key_for(self, kid: Optional[str]) -> Optional[Tuple[int, int]]:
        if kid in self._by_kid:
            return self._by_kid[kid]
        if kid is None and len(self._by_kid) == 1:
            return next(iter(self._by_kid.values()))
        # unknown kid → one refresh attempt (key rotation)
        if self.url:
            self._fetched_at = 0.0
            self.refresh_from_url()
            if kid in self._by_kid:
                return self._by_kid[kid]
        return None
