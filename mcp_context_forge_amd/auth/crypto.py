"""Credential encryption at rest.

Reference analog: db.py:277 `EncryptedText` column type + encryption_service
(gateway/tool auth material never stored plaintext).

This image has no `cryptography` package, so the cipher is built from
hashlib primitives: SHAKE-256 as an XOF keystream (key ∥ nonce → stream,
XORed with the plaintext) with HMAC-SHA256 over nonce∥ciphertext in
encrypt-then-MAC order, keys derived from the configured secret via
PBKDF2-HMAC (separate enc/mac keys by label). Legacy plaintext values pass
through `open_()` unchanged so pre-existing rows keep working.
"""

from __future__ import annotations

import base64
import hashlib
import hmac
import os
from typing import Optional

_PREFIX = "enc1:"


class EncryptionService:
    def __init__(self, secret: str, iterations: int = 100_000):
        base = secret.encode()
        self._enc_key = hashlib.pbkdf2_hmac("sha256", base, b"forge-enc-v1", iterations)
        self._mac_key = hashlib.pbkdf2_hmac("sha256", base, b"forge-mac-v1", iterations)

    def _stream(self, nonce: bytes, n: int) -> bytes:
        x = hashlib.shake_256()
        x.update(self._enc_key)
        x.update(nonce)
        return x.digest(n)

    def seal(self, plaintext: Optional[str]) -> Optional[str]:
        if plaintext is None or plaintext == "":
            return plaintext
        data = plaintext.encode()
        nonce = os.urandom(16)
        ks = self._stream(nonce, len(data))
        ct = bytes(a ^ b for a, b in zip(data, ks))
        mac = hmac.new(self._mac_key, nonce + ct, hashlib.sha256).digest()
        return _PREFIX + base64.urlsafe_b64encode(nonce + ct + mac).decode()

    def open_(self, blob: Optional[str]) -> Optional[str]:
        """Decrypt a sealed value; non-sealed (legacy plaintext) passes through."""
        if blob is None or not blob.startswith(_PREFIX):
            return blob
        raw = base64.urlsafe_b64decode(blob[len(_PREFIX):].encode())
        if len(raw) < 48:
            raise ValueError("sealed blob too short")
        nonce, ct, mac = raw[:16], raw[16:-32], raw[-32:]
        want = hmac.new(self._mac_key, nonce + ct, hashlib.sha256).digest()
        if not hmac.compare_digest(mac, want):
            raise ValueError("sealed blob failed integrity check")
        ks = self._stream(nonce, len(ct))
        return bytes(a ^ b for a, b in zip(ct, ks)).decode()

    def is_sealed(self, blob: Optional[str]) -> bool:
        return bool(blob) and blob.startswith(_PREFIX)
