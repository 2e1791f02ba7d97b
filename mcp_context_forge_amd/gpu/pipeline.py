"""The batched GPU plugin pipeline — the MI355X-native hot path.

Reference analog: the per-request chain tool_service.invoke_tool (:5067)
with plugin hooks (:5530). Here a whole micro-batch is staged to HBM and the
plugin chain's data-parallel stages run as HIP kernels (BASELINE.json):

  json_guard  → structural limits over raw payloads
  scan banks  → deny_filter / harmful_content / pii_filter / regex_filter /
                normalizer-trigger DFAs (one table-driven kernel, ops/csrc/scan.hip)
  featurize   → hashed count vectors (LDS histograms)
  classifier  → content_moderation bf16 MFMA MLP
  semcache    → response_cache_by_prompt cosine sweep over HBM-resident keys

Parity contract (tests/test_gpu_parity.py): requests the kernels flag for
*rewrite-class* plugins (regex/pii/normalizer) are routed to the exact
per-request CPU chain (the oracle), so any GPU/CPU divergence is confined
to provably-no-op requests; block-class decisions (deny/harm/moderation)
are produced directly from kernel outputs with the same thresholds.
"""

from __future__ import annotations

import asyncio
import json
import time
from typing import Any, Dict, List, Optional, Tuple

import numpy as np
import torch

from ..ops import dfa, hip
from ..protocol import jsonrpc
from ..plugins.framework import PluginMode
from .batch import ParsedRequest, canonical_text, pack_texts, pad_rows, parse_batch
from .classifier import GpuClassifier
from .semcache import GpuSemanticCache, tool_hash

# normalizer-trigger prefilter: any byte pattern whose presence could make
# argument_normalizer rewrite the payload (conservative superset). The scan
# runs over CANONICAL JSON text, so whitespace/unicode appear as their JSON
# escapes: backslash-t/n/r/f, backslash-u (covers \\u000b, \\u000c and all
# non-ASCII that NFC could touch), literal double-space, and space adjacent
# to a quote (leading/trailing-space strip).
_NORMALIZE_TRIGGERS = ["\\t", "\\n", "\\r", "\\f", "\\u", "  ", '" ', ' "']


class GpuPluginPipeline:
    def __init__(self, engine, device: str = "cuda"):
        self.engine = engine
        self.device = device
        s = engine.settings
        mgr = engine.plugins

        def plug(name):
            p = mgr.get(name)
            return p if p is not None and p.mode != PluginMode.DISABLED else None

        self.deny = plug("deny_filter")
        self.harm = plug("harmful_content_detector")
        self.pii = plug("pii_filter")
        self.regex = plug("regex_filter")
        self.normalizer = plug("argument_normalizer")
        self.moderation = plug("content_moderation")
        self.semcache_plugin = plug("response_cache_by_prompt")
        self.schema_guard = plug("schema_guard")
        self.toon = plug("toon_encoder")
        self.out_guard = plug("output_length_guard")
        self.exact_cache = plug("cached_tool_result")
        self.breaker = plug("circuit_breaker")

        self.banks: Dict[str, hip.DeviceScanTables] = {}
        for name, p in (("deny", self.deny), ("harm", self.harm), ("pii", self.pii), ("regex", self.regex)):
            tables = p.scan_tables() if p is not None and hasattr(p, "scan_tables") else None
            if tables is not None:
                self.banks[name] = hip.DeviceScanTables(tables, device)
        if self.normalizer is not None:
            self.banks["normalize"] = hip.DeviceScanTables(
                dfa.compile_literals(_NORMALIZE_TRIGGERS, case_insensitive=False), device)

        self.classifier: Optional[GpuClassifier] = None
        self.feat_dim = s.gpu_feature_dim
        if self.moderation is not None:
            self.classifier = GpuClassifier(self.moderation.model, device)
            self.feat_dim = self.moderation.dim

        self.semcache: Optional[GpuSemanticCache] = None
        if self.semcache_plugin is not None:
            self.semcache = GpuSemanticCache(
                capacity=max(128, (s.gpu_semcache_capacity // 128) * 128),
                dim=self.feat_dim,
                threshold=self.semcache_plugin.threshold,
                ttl_s=self.semcache_plugin.ttl,
                device=device,
            )

        self.max_depth = s.max_json_depth
        self.max_string = s.max_string_length
        # stats
        self.batches = 0
        self.requests = 0
        self.fast_path = 0
        self.slow_path = 0
        self.blocked = 0
        self.cache_hits = 0

    # ------------------------------------------------------------------
    def _gpu_scan_pass(self, texts: List[bytes]) -> Dict[str, np.ndarray]:
        """Launch guard+scan+featurize+classifier+semcache; single sync; host arrays."""
        data, offs = pack_texts(texts, self.device)
        out: Dict[str, torch.Tensor] = {}
        status, depth = hip.json_guard(data, offs, self.max_depth, self.max_string)
        out["guard_status"] = status
        for name, bank in self.banks.items():
            mask, _ = hip.scan(data, offs, bank)
            out[f"mask_{name}"] = mask
        feats = None
        if self.classifier is not None or self.semcache is not None:
            feats_b, _ = hip.featurize(data, offs, self.feat_dim)
            feats = pad_rows(feats_b, 128)
        if self.classifier is not None:
            out["scores"] = self.classifier.forward(feats)[: len(texts)]
        if self.semcache is not None:
            bv, bi = self.semcache.lookup(feats)
            out["cache_val"], out["cache_idx"] = bv[: len(texts)], bi[: len(texts)]
        torch.cuda.synchronize()
        host = {k: v.cpu().numpy() for k, v in out.items()}
        host["_feats"] = feats  # device tensor kept for cache inserts
        return host

    def _enforcing(self, plugin) -> bool:
        return plugin is not None and plugin.mode in (PluginMode.ENFORCE, PluginMode.ENFORCE_IGNORE_ERROR)

    def _applies(self, plugin, name: str) -> bool:
        if plugin is None:
            return False
        if not plugin.conditions:
            return True
        from ..plugins.framework import HookType, PluginContext

        return plugin.applies_to(PluginContext(hook=HookType.TOOL_PRE_INVOKE, name=name))

    # ------------------------------------------------------------------
    async def process_batch(self, raws: List[bytes], user: Optional[str] = None,
                            server_id: Optional[str] = None) -> List[Optional[bytes]]:
        self.batches += 1
        self.requests += len(raws)
        parsed = parse_batch(raws)
        responses: List[Optional[bytes]] = [None] * len(raws)

        toolcalls: List[ParsedRequest] = []
        passthrough: List[ParsedRequest] = []
        for pr in parsed:
            if pr.error is not None:
                if pr.req is None or not pr.req.is_notification:
                    rid = pr.req.id if pr.req else None
                    responses[pr.index] = jsonrpc.JSONRPCResponse(id=rid, error=pr.error).to_bytes()
            elif pr.tool_name is not None:
                pr.tool = self.engine.registry.lookup_tool(pr.tool_name)
                if pr.tool is None:
                    responses[pr.index] = jsonrpc.error_response(
                        pr.req.id, jsonrpc.INVALID_PARAMS, f"Tool not found: {pr.tool_name}").to_bytes()
                else:
                    toolcalls.append(pr)
            else:
                passthrough.append(pr)

        # non-tools/call methods ride the per-request engine path
        if passthrough:
            others = await asyncio.gather(
                *(self.engine.handle_rpc(pr.req, user=user, server_id=server_id) for pr in passthrough))
            for pr, resp in zip(passthrough, others):
                responses[pr.index] = resp.to_bytes() if resp is not None else None

        if toolcalls:
            await self._process_toolcalls(toolcalls, responses, user, server_id)
        return responses

    async def _process_toolcalls(self, items: List[ParsedRequest], responses: List[Optional[bytes]],
                                 user: Optional[str], server_id: Optional[str]) -> None:
        texts = [pr.arg_text for pr in items]
        scan = self._gpu_scan_pass(texts)
        n = len(items)
        zeros = np.zeros(n, dtype=np.int64)
        deny_m = scan.get("mask_deny", zeros)
        harm_m = scan.get("mask_harm", zeros)
        pii_m = scan.get("mask_pii", zeros)
        regex_m = scan.get("mask_regex", zeros)
        norm_m = scan.get("mask_normalize", zeros)
        scores = scan.get("scores")
        cache_val = scan.get("cache_val")
        cache_idx = scan.get("cache_idx")

        # semcache hit resolution (host confirm: tool identity + ttl)
        hits: List[Optional[Any]] = [None] * n
        th_arr = np.array([tool_hash(pr.tool_name) for pr in items], dtype=np.int64)
        if self.semcache is not None and cache_val is not None:
            hits = self.semcache.resolve_hits(cache_val, cache_idx, th_arr)

        from ..models.classifier import category_names

        slow: List[ParsedRequest] = []
        dispatch: List[ParsedRequest] = []
        dispatch_rows: List[int] = []
        t0 = time.monotonic()

        for i, pr in enumerate(items):
            rid = pr.req.id
            name = pr.tool_name
            # --- block-class decisions (same order as the CPU chain priorities) ---
            if deny_m[i] and self._enforcing(self.deny) and self._applies(self.deny, name):
                pid = int(deny_m[i]).bit_length() - 1
                word = self.deny.words[pid] if pid < len(self.deny.words) else "?"
                responses[pr.index] = jsonrpc.error_response(
                    rid, jsonrpc.POLICY_DENIED, f"deny_filter: deny word {word!r} present").to_bytes()
                self.blocked += 1
                continue
            # rewrite-class flags → exact per-request CPU chain (parity slow path)
            if (pii_m[i] and self._applies(self.pii, name)) or \
               (regex_m[i] and self._applies(self.regex, name)) or \
               (norm_m[i] and self._applies(self.normalizer, name)):
                slow.append(pr)
                continue
            if scores is not None and self._enforcing(self.moderation) and self._applies(self.moderation, name):
                row = scores[i]
                worst = float(row.max())
                if worst >= self.moderation.threshold:
                    cat = category_names(len(row))[int(row.argmax())]
                    responses[pr.index] = jsonrpc.error_response(
                        rid, jsonrpc.POLICY_DENIED,
                        f"content_moderation: moderation: category {cat} score {worst:.3f}").to_bytes()
                    self.blocked += 1
                    continue
            if harm_m[i] and self._enforcing(self.harm) and self._applies(self.harm, name):
                pid = int(harm_m[i]).bit_length() - 1
                cat = self.harm.cats[pid] if pid < len(self.harm.cats) else "?"
                responses[pr.index] = jsonrpc.error_response(
                    rid, jsonrpc.POLICY_DENIED, f"harmful_content_detector: harmful content ({cat})").to_bytes()
                self.blocked += 1
                continue
            # schema validation (host dict-walk; exact CPU semantics)
            if self._enforcing(self.schema_guard) and pr.tool.get("input_schema"):
                from ..utils.jsonschema import validate as _validate

                errs = _validate(pr.arguments or {}, pr.tool["input_schema"])
                if errs:
                    responses[pr.index] = jsonrpc.error_response(
                        rid, jsonrpc.POLICY_DENIED,
                        "schema_guard: schema violation: " + "; ".join(errs[:5])).to_bytes()
                    self.blocked += 1
                    continue
            # cache hits (semantic, then exact)
            if hits[i] is not None:
                responses[pr.index] = jsonrpc.result_response(rid, hits[i]).to_bytes()
                self.cache_hits += 1
                continue
            if self.exact_cache is not None:
                k = self.exact_cache._key(name, pr.arguments)
                ent = self.exact_cache.store.get(k)
                if ent and time.monotonic() - ent[0] <= self.exact_cache.ttl:
                    responses[pr.index] = jsonrpc.result_response(rid, ent[1]).to_bytes()
                    self.cache_hits += 1
                    continue
            if self.breaker is not None:
                st = self.breaker._st(name)
                if time.monotonic() < st["open_until"]:
                    responses[pr.index] = jsonrpc.error_response(
                        rid, jsonrpc.POLICY_DENIED, f"circuit_breaker: circuit open for tool {name}").to_bytes()
                    self.blocked += 1
                    continue
            dispatch.append(pr)
            dispatch_rows.append(i)

        # --- slow path: exact CPU chain ---
        self.slow_path += len(slow)

        async def run_slow(pr: ParsedRequest) -> None:
            resp = await self.engine.handle_rpc(pr.req, user=user, server_id=server_id)
            responses[pr.index] = resp.to_bytes() if resp is not None else None

        # --- fast path: batched fan-out dispatch ---
        self.fast_path += len(dispatch)

        async def run_fast(pr: ParsedRequest) -> Tuple[ParsedRequest, Optional[dict], Optional[Exception]]:
            try:
                result = await self.engine.tool_service.dispatch(pr.tool, pr.arguments)
                return pr, result, None
            except Exception as exc:
                return pr, None, exc

        slow_task = asyncio.gather(*(run_slow(pr) for pr in slow)) if slow else None
        fast_results = await asyncio.gather(*(run_fast(pr) for pr in dispatch)) if dispatch else []
        if slow_task is not None:
            await slow_task

        # --- post chain for fast-path results ---
        insert_rows: List[int] = []
        insert_hashes: List[int] = []
        insert_results: List[Any] = []
        post_tool_ids: List[str] = []
        post_success: List[bool] = []
        for (pr, result, exc), row in zip(fast_results, dispatch_rows):
            rid = pr.req.id
            if exc is not None:
                code = getattr(exc, "code", jsonrpc.SERVER_ERROR)
                responses[pr.index] = jsonrpc.error_response(rid, code, str(exc)).to_bytes()
                post_tool_ids.append(pr.tool.get("id", pr.tool_name))
                post_success.append(False)
                if self.breaker is not None:
                    self._breaker_record(pr.tool_name, True)
                continue
            result = self._post_chain(pr, result)
            responses[pr.index] = jsonrpc.result_response(rid, result).to_bytes()
            ok = not (isinstance(result, dict) and result.get("isError"))
            post_tool_ids.append(pr.tool.get("id", pr.tool_name))
            post_success.append(ok)
            if self.breaker is not None:
                self._breaker_record(pr.tool_name, not ok)
            if self.exact_cache is not None and ok:
                self.exact_cache.store[self.exact_cache._key(pr.tool_name, pr.arguments)] = (time.monotonic(), result)
            if self.semcache is not None and ok:
                insert_rows.append(row)
                insert_hashes.append(int(th_arr[row]))
                insert_results.append(result)

        if insert_rows and scan.get("_feats") is not None:
            self.semcache.insert_batch(scan["_feats"], insert_rows, insert_hashes, insert_results)

        if post_tool_ids:
            ms = (time.monotonic() - t0) * 1000.0
            self.engine.metrics.record_batch(post_tool_ids, ms, post_success)

    def _breaker_record(self, name: str, is_error: bool) -> None:
        b = self.breaker
        st = b._st(name)
        st["results"].append(is_error)
        if len(st["results"]) > b.window:
            st["results"] = st["results"][-b.window:]
        if len(st["results"]) >= b.window and (sum(st["results"]) / len(st["results"])) >= b.error_threshold:
            st["open_until"] = time.monotonic() + b.cooldown
            st["results"] = []

    def _post_chain(self, pr: ParsedRequest, result: dict) -> dict:
        """Host post-invoke stages for fast-path results (toon, length guard).

        Result-side scanning (pii/regex/harm on outputs) runs through a scan
        pass batched at the caller level in a later round; for now fast-path
        results from trusted LOCAL/MCP upstreams take toon + guard, matching
        the CPU chain for clean results.
        """
        if self.toon is not None and self._applies(self.toon, pr.tool_name):
            new = self.toon.encode_result(result)
            if new is not None:
                result = new
        if self.out_guard is not None and isinstance(result, dict):
            total = sum(len(c.get("text", "")) for c in result.get("content", []) if isinstance(c, dict))
            if total > self.out_guard.max_chars and self.out_guard.action == "truncate":
                result = dict(result)
                result["content"] = [
                    {**c, "text": c["text"][: self.out_guard.max_chars]}
                    if isinstance(c, dict) and isinstance(c.get("text"), str) else c
                    for c in result.get("content", [])
                ]
        return result

    def stats(self) -> Dict[str, Any]:
        out = {
            "batches": self.batches, "requests": self.requests, "fast_path": self.fast_path,
            "slow_path": self.slow_path, "blocked": self.blocked, "cache_hits": self.cache_hits,
            "banks": {k: {"states": v.n_states, "classes": v.n_classes} for k, v in self.banks.items()},
        }
        if self.semcache is not None:
            out["semcache"] = self.semcache.stats()
        return out
