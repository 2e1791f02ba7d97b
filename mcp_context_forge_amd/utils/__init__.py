"""Shared helpers (reference analog: mcpgateway/utils/, 15k LoC, trimmed)."""

from __future__ import annotations

import asyncio
import random
import re
import time
import unicodedata
from typing import Any, Awaitable, Callable, Optional, TypeVar

T = TypeVar("T")

_slug_re = re.compile(r"[^a-z0-9]+")


def slugify(name: str) -> str:
    """Gateway/tool slug (reference: mcpgateway/utils/create_slug.py behavior)."""
    s = unicodedata.normalize("NFKD", name).encode("ascii", "ignore").decode()
    s = _slug_re.sub("-", s.lower()).strip("-")
    return s or "x"


def qualified_tool_name(gateway_name: str, original_name: str) -> str:
    """Federated tool naming: '<gateway-slug>-<original>' (reference: gateway_service._update_or_create_tools :5648)."""
    return f"{slugify(gateway_name)}-{original_name}"


class RetryManager:
    """Exponential backoff with jitter (reference: utils/retry_manager.py:183 ResilientHttpClient)."""

    def __init__(self, max_retries: int = 3, base_delay_ms: int = 100, max_delay_ms: int = 5000, jitter: float = 0.25):
        self.max_retries = max_retries
        self.base_delay_ms = base_delay_ms
        self.max_delay_ms = max_delay_ms
        self.jitter = jitter

    def delay_s(self, attempt: int) -> float:
        d = min(self.base_delay_ms * (2 ** attempt), self.max_delay_ms) / 1000.0
        return d * (1.0 + random.uniform(-self.jitter, self.jitter))

    async def run(self, fn: Callable[[], Awaitable[T]], retry_on: tuple = (Exception,)) -> T:
        last: Optional[BaseException] = None
        for attempt in range(self.max_retries + 1):
            try:
                return await fn()
            except retry_on as exc:  # noqa: PERF203
                last = exc
                if attempt == self.max_retries:
                    break
                await asyncio.sleep(self.delay_s(attempt))
        assert last is not None
        raise last


from .jsonpath import jsonpath_filter  # noqa: F401  (full evaluator in utils/jsonpath.py)


class TokenBucket:
    """Rate-limit token bucket (reference: middleware/rate_limit_middleware.py)."""

    def __init__(self, rate_per_minute: int, burst: int):
        self.rate = rate_per_minute / 60.0
        self.capacity = float(burst)
        self.tokens = float(burst)
        self.last = time.monotonic()

    def allow(self, n: float = 1.0) -> bool:
        now = time.monotonic()
        self.tokens = min(self.capacity, self.tokens + (now - self.last) * self.rate)
        self.last = now
        if self.tokens >= n:
            self.tokens -= n
            return True
        return False


def monotonic_ms() -> float:
    return time.monotonic() * 1000.0
