"""Settings / env-loading tier (reference: config.py Settings; ~1,150 env
fields there, the consulted subset here)."""

import os

from mcp_context_forge_amd.config import Settings, get_settings, reset_settings


def test_env_overrides_and_types(monkeypatch):
    monkeypatch.setenv("FORGE_PORT", "5555")
    monkeypatch.setenv("FORGE_AUTH_REQUIRED", "false")
    monkeypatch.setenv("FORGE_GPU_BATCH_WINDOW_US", "1234")
    monkeypatch.setenv("FORGE_CORS_ALLOW_ORIGINS", '["https://a.example","https://b.example"]')
    reset_settings()
    try:
        s = get_settings()
        assert s.port == 5555 and s.auth_required is False
        assert s.gpu_batch_window_us == 1234
        assert s.cors_allow_origins == ["https://a.example", "https://b.example"]
    finally:
        reset_settings()


def test_defaults_sane():
    s = Settings()
    assert s.gpu_semcache_capacity % 128 == 0
    assert s.gpu_semcache_sketch_dim % 64 == 0
    assert s.max_request_body_bytes > 0
    assert s.session_persistence is False
    assert 0 < s.gpu_batch_window_us < 100000


def test_settings_singleton_reset():
    a = get_settings()
    assert get_settings() is a
    reset_settings()
    assert get_settings() is not a
