"""Targeted kills for mutation-sweep survivors (profiles/mutation_report.txt
— round-3 worklist item 8). Each test pins the exact behavior whose mutant
survived the main suites; security-sensitive modules (rsa, oauth, leader)
first. Survivors NOT covered here were triaged as outcome-equivalent
(e.g. rsa.py verify `s >= n` vs `s > n`: s == n ⇒ pow() == 0 ⇒ encoded
message mismatch ⇒ False either way)."""

import asyncio
import time

import pytest

from mcp_context_forge_amd.auth import rsa as rsa_mod
from mcp_context_forge_amd.auth.oauth import ClientCredentialsProvider
from mcp_context_forge_amd.auth.rsa import (JWKSet, _emsa_pkcs1_v15, _is_probable_prime,
                                            _random_prime, generate_keypair,
                                            sign_pkcs1_sha256, verify_pkcs1_sha256)
from mcp_context_forge_amd.db.engine import Database
from mcp_context_forge_amd.services.leader import DbLeaderElector

# ------------------------------------------------------------------ rsa


def test_emsa_minimum_length_guard():
    """rsa.py:47 `em_len < len(t) + 11`: below the RFC 8017 minimum the
    encoder must RAISE, at exactly the minimum it must succeed with an
    empty-adjacent PS of 8 bytes."""
    t_len = len(rsa_mod._SHA256_PREFIX) + 32
    with pytest.raises(ValueError):
        _emsa_pkcs1_v15(b"m", t_len + 10)
    em = _emsa_pkcs1_v15(b"m", t_len + 11)
    assert em.startswith(b"\x00\x01" + b"\xff" * 8 + b"\x00")
    assert len(em) == t_len + 11


def test_miller_rabin_known_composites_and_primes():
    """rsa.py:103-113 (Miller-Rabin internals): strong pseudoprimes and
    Carmichael numbers must classify as composite, known primes as prime."""
    for composite in (561, 1105, 1729, 2465, 6601,        # Carmichael
                      3215031751,                          # strong pseudoprime base 2,3,5,7
                      2047, 1373653, 25326001,             # spsp chains
                      4, 100, 2 ** 64):
        assert not _is_probable_prime(composite), composite
    for prime in (2, 3, 61, 2 ** 31 - 1, 999999937, 67280421310721):
        assert _is_probable_prime(prime), prime


def test_random_prime_form():
    """rsa.py:131 `| (1 << bits-1) | 1`: generated primes carry the top bit
    (full width) and are odd."""
    for _ in range(3):
        p = _random_prime(24)
        assert p.bit_length() == 24
        assert p % 2 == 1
        assert _is_probable_prime(p)


def test_sign_verify_roundtrip_small_key_raises():
    """sign through a modulus below the EMSA minimum must raise, not emit
    a malformed signature."""
    # 128-bit modulus → k = 16 < 51-byte minimum
    n, e, d = generate_keypair(bits=128)
    with pytest.raises(ValueError):
        sign_pkcs1_sha256(n, d, b"msg")


def test_jwks_refresh_ignores_non_200():
    """rsa.py:179 `status_code == 200`: error responses must not populate
    keys nor advance the fetch timestamp."""

    class FakeResp:
        def __init__(self, code, body):
            self.status_code = code
            self._body = body

        def json(self):
            return self._body

    class FakeClient:
        def __init__(self, resp):
            self.resp = resp

        def get(self, url):
            return self.resp

        def close(self):
            pass

    n, e, _ = generate_keypair(bits=128)
    jwk = rsa_mod.public_to_jwk(n, e, kid="k1")
    ks = JWKSet(url="http://idp.example/jwks", cache_ttl_s=3600.0)
    ks.refresh_from_url(client=FakeClient(FakeResp(503, {"keys": []})))
    assert ks._fetched_at == 0.0 and not ks._by_kid
    ks.refresh_from_url(client=FakeClient(FakeResp(200, {"keys": [jwk]})))
    assert ks._fetched_at > 0.0
    assert ks._by_kid


# ------------------------------------------------------------------ oauth


class _FakeAsyncClient:
    """Minimal httpx.AsyncClient stand-in for the token endpoint."""

    def __init__(self):
        self.calls = 0

    async def post(self, url, data=None, headers=None):
        self.calls += 1

        class R:
            status_code = 200

            @staticmethod
            def json():
                return {"access_token": f"tok{time.time()}", "expires_in": 300}

            @staticmethod
            def raise_for_status():
                return None

        return R()


def test_oauth_refresh_margin_semantics(run):
    """oauth.py:47 `time.time() < expires_at - margin`: a token with MORE
    than the margin left is reused; one inside the margin re-exchanges."""

    async def go():
        p = ClientCredentialsProvider("http://t/token", "cid", "sec", refresh_margin_s=60.0)
        fc = _FakeAsyncClient()
        await p.get_token(client=fc)
        assert p.exchanges == 1
        # plenty of life left → cached
        p._expires_at = time.time() + 200.0
        await p.get_token(client=fc)
        assert p.exchanges == 1, "token with margin left must be reused"
        # inside the refresh margin → re-exchange
        p._expires_at = time.time() + 30.0
        await p.get_token(client=fc)
        assert p.exchanges == 2, "token inside the margin must refresh"

    run(go())


# ------------------------------------------------------------------ leader


def test_leader_flag_lifecycle(tmp_path):
    """leader.py:41/:89 `_leader` flag: False before any election, False
    again after release (and the row is actually gone)."""
    db = Database(f"sqlite:///{tmp_path}/lease.db")
    db.migrate()
    a = DbLeaderElector(db, ttl_s=5.0, holder_id="A")
    assert a.is_leader is False, "fresh elector must not claim leadership"
    assert a.try_acquire() is True
    assert a.is_leader is True
    a.release()
    assert a.is_leader is False
    # after release the lease is free: another holder wins immediately
    b = DbLeaderElector(db, ttl_s=5.0, holder_id="B")
    assert b.try_acquire() is True
    db.close()


def test_verify_rejects_malformed_signatures():
    """rsa.py:57-65 guard returns: wrong-length, >= n, and garbage
    signatures must ALL verify False (a flipped return accepts forgeries);
    and the honest round-trip must still verify True."""
    n, e, d = generate_keypair(bits=512)
    k = (n.bit_length() + 7) // 8
    msg = b"payload"
    good = sign_pkcs1_sha256(n, d, msg)
    assert verify_pkcs1_sha256(n, e, good, msg) is True
    assert verify_pkcs1_sha256(n, e, good, b"other") is False
    assert verify_pkcs1_sha256(n, e, good[:-1], msg) is False          # short
    assert verify_pkcs1_sha256(n, e, good + b"\x00", msg) is False     # long
    assert verify_pkcs1_sha256(n, e, n.to_bytes(k, "big"), msg) is False   # s >= n
    assert verify_pkcs1_sha256(n, e, b"\x7f" * k, msg) is False        # garbage
    # flipped sign bit
    bad = bytes([good[0] ^ 0x01]) + good[1:]
    assert verify_pkcs1_sha256(n, e, bad, msg) is False


def test_probable_prime_small_inputs():
    """rsa.py:98 `n < 2 → False`: 0, 1 and negatives are not prime."""
    for n in (1, 0, -7):
        assert not _is_probable_prime(n), n
    for p in (2, 3, 5):  # small-prime shortcut path (n == p)
        assert _is_probable_prime(p), p
    assert not _is_probable_prime(6)


def test_random_prime_width_repeated():
    """rsa.py:131 top-bit OR: every draw must be exactly the requested
    width (a dropped OR gives ~50% shorter candidates — 8 draws make the
    mutant's survival odds (1/2)^8)."""
    for _ in range(8):
        p = _random_prime(20)
        assert p.bit_length() == 20 and p % 2 == 1


def test_verify_short_modulus_returns_false():
    """rsa.py:65 except-path return: a modulus too small for EMSA encoding
    (k < 51) must verify False for ANY signature, never raise or accept."""
    n, e, _ = generate_keypair(bits=128)
    k = (n.bit_length() + 7) // 8
    assert verify_pkcs1_sha256(n, e, b"\x01" * k, b"msg") is False


def test_jwks_ttl_caches_and_kid_fallback():
    """rsa.py:171 TTL gate: a second refresh inside the TTL must not hit
    the URL again; rsa.py:189 single-key fallback for kid=None."""

    class CountingClient:
        def __init__(self, body):
            self.gets = 0
            self.body = body

        def get(self, url):
            self.gets += 1

            class R:
                status_code = 200
                json = staticmethod(lambda b=self.body: b)
            return R()

        def close(self):
            pass

    n, e, _ = generate_keypair(bits=128)
    jwk = rsa_mod.public_to_jwk(n, e, kid="rot1")
    cc = CountingClient({"keys": [jwk]})
    ks = JWKSet(url="http://idp/jwks", cache_ttl_s=3600.0)
    ks.refresh_from_url(client=cc)
    assert cc.gets == 1
    ks.refresh_from_url(client=cc)
    assert cc.gets == 1, "inside the TTL the cached key set must be reused"
    assert ks.key_for("rot1") == (n, e)
    assert ks.key_for(None) == (n, e)  # single-key fallback
