"""Builtin plugin set.

Per-plugin reference analogs (behavioral port, no code copied):
  deny_filter        → plugins/deny_filter/deny.py (deny-list word filter)
  regex_filter       → plugins/regex_filter/search_replace.py
  pii_filter         → cpex-pii-filter (detect/mask emails, SSNs, cards, phones, IPs)
  schema_guard       → plugins/schema_guard (JSON-Schema arg/result validation)
  toon_encoder       → plugins/toon_encoder (JSON→TOON result compression)
  content_moderation → plugins/content_moderation (category classifier)
  harmful_content    → plugins/harmful_content_detector (keyword+classifier)
  response_cache_by_prompt → semantic cosine cache (plugins/response_cache_by_prompt)
  cached_tool_result → plugins/cached_tool_result (TTL exact-match cache)
  circuit_breaker    → plugins/circuit_breaker (per-tool error-rate breaker)
  argument_normalizer→ plugins/argument_normalizer (unicode/whitespace normalize)
  output_length_guard→ plugins/output_length_guard
  header_injector / header_filter → plugins/header_injector, plugins/header_filter

Each plugin implements the per-request async hooks (the CPU/reference path
and the parity oracle). GPU-capable plugins additionally expose the data
the batched pipeline needs (`scan_tables()`, `classifier()`, `featurize
dim`), and the pipeline applies their semantics batch-wide with HIP kernels
— see gpu/pipeline.py. Detection parity between the two paths is asserted
in tests/test_gpu_parity.py.
"""

from __future__ import annotations

import hashlib
import json
import re
import time
import unicodedata
from typing import Any, Dict, List, Optional, Tuple

import numpy as np

from ..ops import dfa
from ..ops.featurize import featurize
from ..utils.jsonschema import validate as schema_validate
from . import toon as toon_codec
from .framework import HookType, Plugin, PluginContext, PluginResult


def _text_of(payload: Any) -> str:
    """Canonical text of an args/result payload for scanning/featurizing."""
    if payload is None:
        return ""
    if isinstance(payload, str):
        return payload
    try:
        return json.dumps(payload, separators=(",", ":"), sort_keys=True, default=str)
    except Exception:
        return str(payload)


def _walk_strings(payload: Any, fn) -> Any:
    if isinstance(payload, str):
        return fn(payload)
    if isinstance(payload, dict):
        return {k: _walk_strings(v, fn) for k, v in payload.items()}
    if isinstance(payload, list):
        return [_walk_strings(v, fn) for v in payload]
    return payload


# ---------------------------------------------------------------------------


class DenyFilterPlugin(Plugin):
    """Block requests containing deny-listed words (reference: deny.py:118)."""

    name = "deny_filter"
    hooks = (HookType.TOOL_PRE_INVOKE, HookType.PROMPT_PRE_FETCH, HookType.AGENT_PRE_INVOKE)
    priority = 10
    gpu_capable = True

    def __init__(self, config: Optional[Dict[str, Any]] = None):
        super().__init__(config)
        self.words: List[str] = list(self.config.get("words") or ["forbidden", "blocked_word"])
        self.case_insensitive: bool = bool(self.config.get("case_insensitive", True))
        self._tables = dfa.compile_literals(self.words, self.case_insensitive)

    def scan_tables(self) -> dfa.ScanTables:
        return self._tables

    async def tool_pre_invoke(self, ctx: PluginContext) -> PluginResult:
        text = _text_of(ctx.args)
        hay = text.lower() if self.case_insensitive else text
        for w in self.words:
            needle = w.lower() if self.case_insensitive else w
            if needle in hay:
                return PluginResult.block(f"deny word {w!r} present", code="deny")
        return PluginResult.ok()

    prompt_pre_fetch = tool_pre_invoke
    agent_pre_invoke = tool_pre_invoke


class RegexFilterPlugin(Plugin):
    """Regex search/replace on args and results (reference: search_replace.py:190)."""

    name = "regex_filter"
    hooks = (HookType.TOOL_PRE_INVOKE, HookType.TOOL_POST_INVOKE, HookType.PROMPT_PRE_FETCH)
    priority = 20
    gpu_capable = True

    def __init__(self, config: Optional[Dict[str, Any]] = None):
        super().__init__(config)
        rules = self.config.get("rules") or [{"search": r"crap", "replace": "crud"}]
        self.rules: List[Tuple[re.Pattern, str]] = [
            (re.compile(r["search"], re.IGNORECASE if r.get("ignorecase", True) else 0), r["replace"]) for r in rules
        ]
        # GPU prefilter tables: the DFA flags requests containing any rule match;
        # only flagged requests pay the host rewrite.
        try:
            self._tables = dfa.compile_patterns(
                [r["search"] for r in rules],
                case_insensitive=any(r.get("ignorecase", True) for r in rules),
            )
        except ValueError:
            self._tables = None

    def scan_tables(self) -> Optional[dfa.ScanTables]:
        return self._tables

    def apply_rules(self, text: str) -> str:
        for pat, repl in self.rules:
            text = pat.sub(repl, text)
        return text

    async def tool_pre_invoke(self, ctx: PluginContext) -> PluginResult:
        return PluginResult.ok(_walk_strings(ctx.args, self.apply_rules))

    async def tool_post_invoke(self, ctx: PluginContext) -> PluginResult:
        return PluginResult.ok(_walk_strings(ctx.args, self.apply_rules))

    prompt_pre_fetch = tool_pre_invoke


# PII pattern set (shared between CPU re-path and the GPU DFA bank).
PII_PATTERNS: List[Tuple[str, str, str]] = [
    # (name, dfa_pattern, python_regex)
    ("ssn", r"\d{3}-\d{2}-\d{4}", r"\b\d{3}-\d{2}-\d{4}\b"),
    ("email", r"[A-Za-z0-9._%+\-]+@[A-Za-z0-9.\-]+\.[A-Za-z][A-Za-z]+", r"[A-Za-z0-9._%+\-]+@[A-Za-z0-9.\-]+\.[A-Za-z]{2,}"),
    ("credit_card", r"\d{4}[ \-]\d{4}[ \-]\d{4}[ \-]\d{4}", r"\b\d{4}[ \-]\d{4}[ \-]\d{4}[ \-]\d{4}\b"),
    ("phone", r"\(?\d{3}\)?[ .\-]\d{3}[ .\-]\d{4}", r"\(?\b\d{3}\)?[ .\-]\d{3}[ .\-]\d{4}\b"),
    ("ipv4", r"\d{1,3}\.\d{1,3}\.\d{1,3}\.\d{1,3}", r"\b\d{1,3}\.\d{1,3}\.\d{1,3}\.\d{1,3}\b"),
    ("aws_key", r"AKIA[0-9A-Z]{16}", r"\bAKIA[0-9A-Z]{16}\b"),
]


class PIIFilterPlugin(Plugin):
    """Detect and mask PII in args and results (reference: cpex-pii-filter)."""

    name = "pii_filter"
    hooks = (HookType.TOOL_PRE_INVOKE, HookType.TOOL_POST_INVOKE, HookType.PROMPT_PRE_FETCH,
             HookType.AGENT_PRE_INVOKE, HookType.AGENT_POST_INVOKE)
    priority = 30
    gpu_capable = True

    def __init__(self, config: Optional[Dict[str, Any]] = None):
        super().__init__(config)
        self.action: str = self.config.get("action", "mask")  # mask | block | audit
        names = set(self.config.get("categories") or [n for n, _, _ in PII_PATTERNS])
        self.active = [(n, d, re.compile(p)) for (n, d, p) in PII_PATTERNS if n in names]
        self._tables = dfa.compile_patterns([d for (_, d, _) in self.active])

    def scan_tables(self) -> dfa.ScanTables:
        return self._tables

    def mask_text(self, text: str) -> Tuple[str, List[str]]:
        found: List[str] = []
        for name, _d, rx in self.active:
            if rx.search(text):
                found.append(name)
                text = rx.sub(f"[{name.upper()}_REDACTED]", text)
        return text, found

    def mask_text_subset(self, text: str, pattern_bits: int) -> Tuple[str, List[str]]:
        """Masked-pattern fast path: bit i of `pattern_bits` is the GPU scan
        bank's accept bit for self.active[i] (same compile order) — only the
        patterns the DFA flagged need their Python regex run."""
        found: List[str] = []
        for i, (name, _d, rx) in enumerate(self.active):
            if not (pattern_bits >> i) & 1:
                continue
            if rx.search(text):
                found.append(name)
                text = rx.sub(f"[{name.upper()}_REDACTED]", text)
        return text, found

    async def _apply(self, ctx: PluginContext) -> PluginResult:
        found_all: List[str] = []

        def fn(s: str) -> str:
            masked, found = self.mask_text(s)
            found_all.extend(found)
            return masked

        new_payload = _walk_strings(ctx.args, fn)
        if found_all and self.action == "block":
            return PluginResult.block(f"PII detected: {sorted(set(found_all))}", code="pii")
        if found_all and self.action == "mask":
            return PluginResult.ok(new_payload, pii=sorted(set(found_all)))
        if found_all:  # audit mode: annotate only
            return PluginResult.ok(pii=sorted(set(found_all)))
        return PluginResult.ok()

    tool_pre_invoke = _apply
    tool_post_invoke = _apply
    prompt_pre_fetch = _apply
    agent_pre_invoke = _apply
    agent_post_invoke = _apply


class SchemaGuardPlugin(Plugin):
    """Validate args/results against the tool's JSON schema (reference: schema_guard.py:168)."""

    name = "schema_guard"
    hooks = (HookType.TOOL_PRE_INVOKE, HookType.TOOL_POST_INVOKE)
    priority = 40
    gpu_capable = True  # GPU does the byte-level structural guard over the batch

    def __init__(self, config: Optional[Dict[str, Any]] = None):
        super().__init__(config)
        self.max_depth = int(self.config.get("max_depth", 64))
        self.max_string = int(self.config.get("max_string", 1 << 20))

    async def tool_pre_invoke(self, ctx: PluginContext) -> PluginResult:
        schema = ctx.state.get("input_schema")
        if schema:
            errs = schema_validate(ctx.args or {}, schema)
            if errs:
                return PluginResult.block("schema violation: " + "; ".join(errs[:5]), code="schema")
        return PluginResult.ok()

    async def tool_post_invoke(self, ctx: PluginContext) -> PluginResult:
        schema = ctx.state.get("output_schema")
        if schema:
            payload = ctx.args
            if isinstance(payload, dict) and "structuredContent" in payload:
                payload = payload["structuredContent"]
            errs = schema_validate(payload, schema)
            if errs:
                return PluginResult.block("output schema violation: " + "; ".join(errs[:5]), code="schema")
        return PluginResult.ok()


class ToonEncoderPlugin(Plugin):
    """Compress JSON tool results to TOON (reference: toon_encoder)."""

    name = "toon_encoder"
    hooks = (HookType.TOOL_POST_INVOKE,)
    priority = 900
    gpu_capable = True

    def __init__(self, config: Optional[Dict[str, Any]] = None):
        super().__init__(config)
        self.min_savings = float(self.config.get("min_savings", 0.1))
        self.min_size = int(self.config.get("min_size", 256))

    def encode_result(self, result: Any) -> Optional[Dict[str, Any]]:
        """Returns modified tool-result dict or None if not worth encoding."""
        if not isinstance(result, dict):
            return None
        sc = result.get("structuredContent")
        if sc is None:
            return None
        j = len(json.dumps(sc, separators=(",", ":")))
        if j < self.min_size:  # cheap reject before paying for the TOON encode
            return None
        encoded = toon_codec.encode(sc)
        t = len(encoded)
        frac = 1.0 - t / j if j else 0.0
        if frac < self.min_savings:
            return None
        new = dict(result)
        new["content"] = [{"type": "text", "text": encoded}]
        meta = new.get("_meta")
        if "_meta" not in new:
            meta = new["_meta"] = {}
        if isinstance(meta, dict):  # a non-dict _meta from the upstream is left alone
            meta["toon"] = {"json_bytes": j, "toon_bytes": t, "savings": round(frac, 4)}
        return new

    async def tool_post_invoke(self, ctx: PluginContext) -> PluginResult:
        new = self.encode_result(ctx.args)
        return PluginResult.ok(new) if new is not None else PluginResult.ok()


class ContentModerationPlugin(Plugin):
    """Category-score moderation classifier (reference: content_moderation.py:846).

    The reference calls external APIs (Watson/OpenAI/patterns); per
    BASELINE.json this build runs a local bf16-MFMA classifier over hashed
    features (random-init weights). CPU path = fp32 torch forward.
    """

    name = "content_moderation"
    hooks = (HookType.TOOL_PRE_INVOKE, HookType.PROMPT_PRE_FETCH, HookType.AGENT_PRE_INVOKE)
    priority = 50
    gpu_capable = True

    def __init__(self, config: Optional[Dict[str, Any]] = None):
        super().__init__(config)
        from ..models.classifier import HashedTextClassifier

        self.dim = int(self.config.get("dim", 4096))
        self.threshold = float(self.config.get("threshold", 0.95))
        self.model = HashedTextClassifier(
            dim=self.dim,
            hidden=int(self.config.get("hidden", 1024)),
            classes=int(self.config.get("classes", 8)),
            seed=int(self.config.get("seed", 1234)),
        ).eval()

    def classifier(self):
        return self.model

    def score_text(self, text: str) -> np.ndarray:
        import torch

        feats = torch.from_numpy(featurize(text, self.dim)).unsqueeze(0)
        with torch.no_grad():
            return self.model(feats)[0].numpy()

    async def tool_pre_invoke(self, ctx: PluginContext) -> PluginResult:
        scores = self.score_text(_text_of(ctx.args))
        worst = float(scores.max())
        if worst >= self.threshold:
            from ..models.classifier import category_names

            cat = category_names(len(scores))[int(scores.argmax())]
            return PluginResult.block(f"moderation: category {cat} score {worst:.3f}", code="moderation")
        return PluginResult.ok(metadata={"moderation_max": worst})

    prompt_pre_fetch = tool_pre_invoke
    agent_pre_invoke = tool_pre_invoke


class HarmfulContentPlugin(Plugin):
    """Keyword-bank harm detector (reference: harmful_content_detector.py:207)."""

    name = "harmful_content_detector"
    hooks = (HookType.TOOL_PRE_INVOKE, HookType.TOOL_POST_INVOKE, HookType.AGENT_PRE_INVOKE)
    priority = 60
    gpu_capable = True

    DEFAULT_BANK = {
        "violence": ["kill them all", "how to make a bomb"],
        "self_harm": ["ways to hurt myself"],
        "illegal": ["buy stolen credit cards"],
    }

    def __init__(self, config: Optional[Dict[str, Any]] = None):
        super().__init__(config)
        self.bank: Dict[str, List[str]] = self.config.get("bank") or dict(self.DEFAULT_BANK)
        phrases, cats = [], []
        for cat, plist in self.bank.items():
            for p in plist:
                phrases.append(p)
                cats.append(cat)
        self.phrases, self.cats = phrases, cats
        self._tables = dfa.compile_literals(phrases, case_insensitive=True)

    def scan_tables(self) -> dfa.ScanTables:
        return self._tables

    async def tool_pre_invoke(self, ctx: PluginContext) -> PluginResult:
        hay = _text_of(ctx.args).lower()
        for phrase, cat in zip(self.phrases, self.cats):
            if phrase.lower() in hay:
                return PluginResult.block(f"harmful content ({cat})", code="harmful_content")
        return PluginResult.ok()

    tool_post_invoke = tool_pre_invoke
    agent_pre_invoke = tool_pre_invoke


class ResponseCacheByPromptPlugin(Plugin):
    """Semantic result cache via cosine over hashed count vectors
    (reference: response_cache_by_prompt.py `_vectorize`:55 `_cos_sim`:74 `_find_best`:163).

    CPU path keeps the matrix in numpy; the GPU pipeline keeps it resident
    in HBM and runs the similarity matmul on MFMA.
    """

    name = "response_cache_by_prompt"
    hooks = (HookType.TOOL_PRE_INVOKE, HookType.TOOL_POST_INVOKE)
    priority = 5  # before everything: a cache hit skips the chain
    gpu_capable = True

    def __init__(self, config: Optional[Dict[str, Any]] = None):
        super().__init__(config)
        self.dim = int(self.config.get("dim", 4096))
        self.threshold = float(self.config.get("threshold", 0.92))
        self.capacity = int(self.config.get("capacity", 4096))
        self.ttl = float(self.config.get("ttl", 600.0))
        # result substitution is gated on an EXPLICIT per-tool allowlist
        # (reference default: empty cacheable_tools — a near-miss cosine hit
        # on an unlisted tool must never replace a live invocation), and
        # entries are tenant-scoped: the identity compared on hit is
        # (tool, user), so one user's result is never served to another.
        self.cacheable_tools = frozenset(self.config.get("cacheable_tools") or [])
        # preallocated ring (matrix never reallocates — mirrors the GPU cache)
        self.vectors = np.zeros((self.capacity, self.dim), dtype=np.float32)
        self.entries: List[Optional[Tuple[Tuple[str, str], float, Any]]] = [None] * self.capacity
        self.size = 0
        self.write_ptr = 0
        self.hits = 0
        self.misses = 0

    def cacheable(self, tool: str) -> bool:
        return tool in self.cacheable_tools

    def lookup(self, tool: str, text: str, user: Optional[str] = None) -> Optional[Any]:
        if self.size == 0:
            self.misses += 1
            return None
        v = featurize(text, self.dim)
        sims = self.vectors[: self.size] @ v
        best = int(np.argmax(sims))
        now = time.monotonic()
        ent = self.entries[best]
        if ent is not None:
            ent_key, ts, result = ent
            if sims[best] >= self.threshold and ent_key == (tool, user or "") and now - ts <= self.ttl:
                self.hits += 1
                return result
        self.misses += 1
        return None

    def insert(self, tool: str, text: str, result: Any, user: Optional[str] = None) -> None:
        slot = self.write_ptr
        self.vectors[slot] = featurize(text, self.dim)
        self.entries[slot] = ((tool, user or ""), time.monotonic(), result)
        self.write_ptr = (self.write_ptr + 1) % self.capacity
        self.size = min(self.size + 1, self.capacity)

    async def tool_pre_invoke(self, ctx: PluginContext) -> PluginResult:
        if not self.cacheable(ctx.name):
            return PluginResult.ok()
        hit = self.lookup(ctx.name, _text_of(ctx.args), user=ctx.user)
        if hit is not None:
            ctx.state["cache_hit"] = hit
        return PluginResult.ok()

    async def tool_post_invoke(self, ctx: PluginContext) -> PluginResult:
        if self.cacheable(ctx.name) and "cache_hit" not in ctx.state and ctx.args is not None:
            req_text = ctx.state.get("request_text", "")
            if req_text:
                self.insert(ctx.name, req_text, ctx.args, user=ctx.user)
        return PluginResult.ok()


class CachedToolResultPlugin(Plugin):
    """Exact-match TTL result cache (reference: cached_tool_result)."""

    name = "cached_tool_result"
    hooks = (HookType.TOOL_PRE_INVOKE, HookType.TOOL_POST_INVOKE)
    priority = 6

    def __init__(self, config: Optional[Dict[str, Any]] = None):
        super().__init__(config)
        self.ttl = float(self.config.get("ttl", 300.0))
        self.store: Dict[str, Tuple[float, Any]] = {}

    def _key(self, name: str, args: Any, user: Optional[str] = None) -> str:
        # user is part of the key: exact-match results never cross tenants
        return hashlib.sha256((name + "\x00" + (user or "") + "\x00" + _text_of(args)).encode()).hexdigest()

    async def tool_pre_invoke(self, ctx: PluginContext) -> PluginResult:
        k = self._key(ctx.name, ctx.args, user=ctx.user)
        ent = self.store.get(k)
        if ent and time.monotonic() - ent[0] <= self.ttl:
            ctx.state["cache_hit"] = ent[1]
        ctx.state["exact_cache_key"] = k
        return PluginResult.ok()

    async def tool_post_invoke(self, ctx: PluginContext) -> PluginResult:
        k = ctx.state.get("exact_cache_key")
        if k and "cache_hit" not in ctx.state:
            self.store[k] = (time.monotonic(), ctx.args)
        return PluginResult.ok()


class CircuitBreakerPlugin(Plugin):
    """Per-tool error-rate breaker (reference: plugins/circuit_breaker)."""

    name = "circuit_breaker"
    hooks = (HookType.TOOL_PRE_INVOKE, HookType.TOOL_POST_INVOKE)
    priority = 8

    def __init__(self, config: Optional[Dict[str, Any]] = None):
        super().__init__(config)
        self.error_threshold = float(self.config.get("error_threshold", 0.5))
        self.window = int(self.config.get("window", 20))
        self.cooldown = float(self.config.get("cooldown", 30.0))
        self.state: Dict[str, Dict[str, Any]] = {}

    def _st(self, tool: str) -> Dict[str, Any]:
        return self.state.setdefault(tool, {"results": [], "open_until": 0.0})

    async def tool_pre_invoke(self, ctx: PluginContext) -> PluginResult:
        st = self._st(ctx.name)
        if time.monotonic() < st["open_until"]:
            return PluginResult.block(f"circuit open for tool {ctx.name}", code="circuit_open")
        return PluginResult.ok()

    async def tool_post_invoke(self, ctx: PluginContext) -> PluginResult:
        st = self._st(ctx.name)
        is_error = bool(isinstance(ctx.args, dict) and ctx.args.get("isError"))
        st["results"].append(is_error)
        if len(st["results"]) > self.window:
            st["results"] = st["results"][-self.window:]
        if len(st["results"]) >= self.window and (sum(st["results"]) / len(st["results"])) >= self.error_threshold:
            st["open_until"] = time.monotonic() + self.cooldown
            st["results"] = []
        return PluginResult.ok()


class ArgumentNormalizerPlugin(Plugin):
    """Unicode/whitespace normalization of string args (reference: argument_normalizer)."""

    name = "argument_normalizer"
    hooks = (HookType.TOOL_PRE_INVOKE, HookType.PROMPT_PRE_FETCH)
    priority = 15

    def __init__(self, config: Optional[Dict[str, Any]] = None):
        super().__init__(config)
        self.form = self.config.get("unicode_form", "NFC")
        self.collapse_ws = bool(self.config.get("collapse_whitespace", True))
        self.strip = bool(self.config.get("strip", True))

    def norm(self, s: str) -> str:
        s = unicodedata.normalize(self.form, s)
        if self.collapse_ws:
            s = re.sub(r"[ \t\f\v]+", " ", s)
        if self.strip:
            s = s.strip()
        return s

    async def tool_pre_invoke(self, ctx: PluginContext) -> PluginResult:
        return PluginResult.ok(_walk_strings(ctx.args, self.norm))

    prompt_pre_fetch = tool_pre_invoke


class OutputLengthGuardPlugin(Plugin):
    """Cap result size (reference: output_length_guard)."""

    name = "output_length_guard"
    hooks = (HookType.TOOL_POST_INVOKE,)
    priority = 950

    def __init__(self, config: Optional[Dict[str, Any]] = None):
        super().__init__(config)
        self.max_chars = int(self.config.get("max_chars", 1 << 20))
        self.action = self.config.get("action", "truncate")  # truncate | block

    async def tool_post_invoke(self, ctx: PluginContext) -> PluginResult:
        result = ctx.args
        if not isinstance(result, dict):
            return PluginResult.ok()
        total = 0
        for c in result.get("content", []):
            if isinstance(c, dict) and isinstance(c.get("text"), str):
                total += len(c["text"])
        if total <= self.max_chars:
            return PluginResult.ok()
        if self.action == "block":
            return PluginResult.block(f"output too large ({total} chars)", code="output_length")
        new = dict(result)
        new["content"] = [
            {**c, "text": c["text"][: self.max_chars]} if isinstance(c, dict) and isinstance(c.get("text"), str) else c
            for c in result.get("content", [])
        ]
        return PluginResult.ok(new)


class HeaderInjectorPlugin(Plugin):
    """Inject headers on outbound calls (reference: header_injector)."""

    name = "header_injector"
    hooks = (HookType.HTTP_PRE_REQUEST, HookType.TOOL_PRE_INVOKE)
    priority = 70

    def __init__(self, config: Optional[Dict[str, Any]] = None):
        super().__init__(config)
        self.headers: Dict[str, str] = dict(self.config.get("headers") or {})

    async def http_pre_request(self, ctx: PluginContext) -> PluginResult:
        ctx.headers.update(self.headers)
        return PluginResult.ok()

    async def tool_pre_invoke(self, ctx: PluginContext) -> PluginResult:
        ctx.headers.update(self.headers)
        return PluginResult.ok()


BUILTIN_PLUGINS = {
    p.name: p
    for p in (
        DenyFilterPlugin,
        RegexFilterPlugin,
        PIIFilterPlugin,
        SchemaGuardPlugin,
        ToonEncoderPlugin,
        ContentModerationPlugin,
        HarmfulContentPlugin,
        ResponseCacheByPromptPlugin,
        CachedToolResultPlugin,
        CircuitBreakerPlugin,
        ArgumentNormalizerPlugin,
        OutputLengthGuardPlugin,
        HeaderInjectorPlugin,
    )
}
