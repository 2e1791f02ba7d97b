"""LLM provider registry + OpenAI-compatible chat-completions proxy.

Reference analogs: services/llm_provider_service.py (provider registry,
db.py:6466 LLM provider tables) and services/llm_proxy_service.py
(OpenAI-compatible proxy :103, streaming :442,529).
"""

from __future__ import annotations

import json
import time
from typing import Any, AsyncIterator, Dict, List, Optional

import httpx


class LLMProxyError(Exception):
    def __init__(self, message: str, status: int = 502):
        self.status = status
        super().__init__(message)


class LLMProviderRegistry:
    """In-memory provider registry (durable rows ride the generic registry's
    export/import; providers hold secrets so they default to memory-only)."""

    def __init__(self):
        self.providers: Dict[str, Dict[str, Any]] = {}
        self.default: Optional[str] = None

    def register(self, name: str, base_url: str, api_key: Optional[str] = None,
                 models: Optional[List[str]] = None, default_model: Optional[str] = None,
                 provider_type: str = "openai") -> Dict[str, Any]:
        p = {"name": name, "base_url": base_url.rstrip("/"), "api_key": api_key,
             "models": models or [], "default_model": default_model, "type": provider_type,
             "enabled": True}
        self.providers[name] = p
        if self.default is None:
            self.default = name
        return {k: v for k, v in p.items() if k != "api_key"}

    def get(self, name: Optional[str] = None) -> Dict[str, Any]:
        key = name or self.default
        if key is None or key not in self.providers:
            raise LLMProxyError(f"no LLM provider {name!r} configured", status=404)
        return self.providers[key]

    def for_model(self, model: Optional[str]) -> Dict[str, Any]:
        if model:
            for p in self.providers.values():
                if model in p["models"]:
                    return p
        return self.get()

    def list(self) -> List[Dict[str, Any]]:
        return [{k: v for k, v in p.items() if k != "api_key"} for p in self.providers.values()]


class LLMProxyService:
    """Proxy /v1/chat/completions to the configured provider, preserving the
    OpenAI wire format, including SSE streaming passthrough."""

    def __init__(self, registry: Optional[LLMProviderRegistry] = None, timeout: float = 120.0):
        self.registry = registry or LLMProviderRegistry()
        self._client: Optional[httpx.AsyncClient] = None

    def _client_for(self) -> httpx.AsyncClient:
        if self._client is None:
            self._client = httpx.AsyncClient(timeout=httpx.Timeout(120.0, connect=10.0))
        return self._client

    def _headers(self, provider: Dict[str, Any]) -> Dict[str, str]:
        h = {"content-type": "application/json"}
        if provider.get("api_key"):
            h["authorization"] = f"Bearer {provider['api_key']}"
        return h

    async def chat_completions(self, body: Dict[str, Any], provider_name: Optional[str] = None) -> Dict[str, Any]:
        provider = self.registry.for_model(body.get("model")) if provider_name is None \
            else self.registry.get(provider_name)
        if body.get("model") is None and provider.get("default_model"):
            body = {**body, "model": provider["default_model"]}
        url = provider["base_url"] + "/chat/completions"
        try:
            resp = await self._client_for().post(url, json=body, headers=self._headers(provider))
        except httpx.HTTPError as exc:
            raise LLMProxyError(f"provider unreachable: {exc}") from exc
        if resp.status_code >= 400:
            raise LLMProxyError(f"provider error {resp.status_code}: {resp.text[:300]}", status=resp.status_code)
        return resp.json()

    async def chat_completions_stream(self, body: Dict[str, Any],
                                      provider_name: Optional[str] = None) -> AsyncIterator[bytes]:
        provider = self.registry.for_model(body.get("model")) if provider_name is None \
            else self.registry.get(provider_name)
        body = {**body, "stream": True}
        url = provider["base_url"] + "/chat/completions"
        client = self._client_for()
        async with client.stream("POST", url, json=body, headers=self._headers(provider)) as resp:
            if resp.status_code >= 400:
                detail = (await resp.aread())[:300]
                raise LLMProxyError(f"provider error {resp.status_code}: {detail!r}", status=resp.status_code)
            async for chunk in resp.aiter_bytes():
                yield chunk

    async def aclose(self) -> None:
        if self._client is not None:
            await self._client.aclose()
