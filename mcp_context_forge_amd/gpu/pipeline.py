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

Three passes, mirroring the CPU chain's priority order exactly:

  PASS 1 (all requests, original args): cache lookup + deny + rewrite-flag
     detection + moderation + harm + schema. Unflagged requests are decided
     entirely from kernel outputs.
  PASS 2 (rewrite-flagged subset): host rewrites (normalizer → regex → pii,
     the plugins' own functions = exact semantics), then a second GPU sweep
     (harm + moderation) over the rewritten texts, then schema.
  PASS 3 (results): one scan over serialized results; pii/regex/harm-flagged
     results take the host post chain; all results get toon + length guard.

Parity gate: tests/test_gpu_parity.py runs identical traffic through this
pipeline and the per-request CPU chain and asserts identical outcomes
(reference analog: tests/live_gateway/mcp/test_mcp_plugin_parity.py).
"""

from __future__ import annotations

import asyncio
import hashlib
import json
import time
from typing import Any, Dict, List, Optional, Tuple

import numpy as np
import torch

from ..ops import dfa, hip
from ..protocol import jsonrpc
from ..plugins.builtin import _walk_strings
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
        self.post_rewrites = 0

    # ------------------------------------------------------------------
    def _scan_pass(self, texts: List[bytes], banks: Tuple[str, ...], classify: bool,
                   cache_lookup: bool, guard: bool = False) -> Dict[str, Any]:
        """Launch the kernel set over packed texts; single sync; host arrays."""
        data, offs = pack_texts(texts, self.device)
        out: Dict[str, torch.Tensor] = {}
        if guard:
            status, _depth = hip.json_guard(data, offs, self.max_depth, self.max_string)
            out["guard_status"] = status
        for name in banks:
            bank = self.banks.get(name)
            if bank is not None:
                out[f"mask_{name}"], _ = hip.scan(data, offs, bank)
        feats = None
        if (classify and self.classifier is not None) or (cache_lookup and self.semcache is not None):
            feats_b, _ = hip.featurize(data, offs, self.feat_dim)
            feats = pad_rows(feats_b, 128)
        if classify and self.classifier is not None:
            out["scores"] = self.classifier.forward(feats)[: len(texts)]
        if cache_lookup and self.semcache is not None:
            bv, bi = self.semcache.lookup(feats)
            out["cache_val"], out["cache_idx"] = bv[: len(texts)], bi[: len(texts)]
        torch.cuda.synchronize()
        host = {k: v.cpu().numpy() for k, v in out.items()}
        host["_feats"] = feats
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

        if passthrough:
            others = await asyncio.gather(
                *(self.engine.handle_rpc(pr.req, user=user, server_id=server_id) for pr in passthrough))
            for pr, resp in zip(passthrough, others):
                responses[pr.index] = resp.to_bytes() if resp is not None else None

        if toolcalls:
            await self._process_toolcalls(toolcalls, responses, user, server_id)
        return responses

    # ------------------------------------------------------------------
    def _block(self, pr: ParsedRequest, responses: List[Optional[bytes]], message: str) -> None:
        responses[pr.index] = jsonrpc.error_response(pr.req.id, jsonrpc.POLICY_DENIED, message).to_bytes()
        self.blocked += 1

    def _moderation_block(self, scores_row: np.ndarray) -> Optional[str]:
        worst = float(scores_row.max())
        if worst >= self.moderation.threshold:
            from ..models.classifier import category_names

            cat = category_names(len(scores_row))[int(scores_row.argmax())]
            return f"content_moderation: moderation: category {cat} score {worst:.3f}"
        return None

    def _schema_errs(self, pr: ParsedRequest, args: Any) -> Optional[str]:
        schema = pr.tool.get("input_schema")
        if not schema:
            return None
        # trivial schemas ({"type":"object"} with no constraints) validate anything
        if not (schema.get("properties") or schema.get("required") or
                schema.get("additionalProperties") is False or schema.get("anyOf") or schema.get("allOf")):
            return None
        from ..utils.jsonschema import validate as _validate

        errs = _validate(args or {}, schema)
        if errs:
            return "schema_guard: schema violation: " + "; ".join(errs[:5])
        return None

    def _apply_rewrites(self, pr: ParsedRequest) -> Any:
        """Host rewrites in CPU-chain priority order: normalizer(15) →
        regex(20) → pii(30). Uses the plugins' own functions (exact parity).
        Returns rewritten args, or a block message for pii action=block."""
        args = pr.arguments
        if self.normalizer is not None and self._applies(self.normalizer, pr.tool_name):
            args = _walk_strings(args, self.normalizer.norm)
        if self.regex is not None and self._applies(self.regex, pr.tool_name):
            args = _walk_strings(args, self.regex.apply_rules)
        if self.pii is not None and self._applies(self.pii, pr.tool_name):
            found: List[str] = []

            def fn(s: str) -> str:
                masked, f = self.pii.mask_text(s)
                found.extend(f)
                return masked

            new_args = _walk_strings(args, fn)
            if found and self.pii.action == "block" and self._enforcing(self.pii):
                return ("__block__", f"pii_filter: PII detected: {sorted(set(found))}")
            if found and self.pii.action == "mask":
                args = new_args
        return ("__ok__", args)

    async def _process_toolcalls(self, items: List[ParsedRequest], responses: List[Optional[bytes]],
                                 user: Optional[str], server_id: Optional[str]) -> None:
        texts = [pr.arg_text for pr in items]
        scan = self._scan_pass(texts, ("deny", "harm", "pii", "regex", "normalize"),
                               classify=True, cache_lookup=True)
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

        hits: List[Optional[Any]] = [None] * n
        th_arr = np.array([tool_hash(pr.tool_name) for pr in items], dtype=np.int64)
        if self.semcache is not None and cache_val is not None:
            hits = self.semcache.resolve_hits(cache_val, cache_idx, th_arr)

        t0 = time.monotonic()
        # (pr, row, args) tuples heading to dispatch
        dispatch: List[Tuple[ParsedRequest, int, Any]] = []
        rewrites: List[Tuple[ParsedRequest, int]] = []

        for i, pr in enumerate(items):
            name = pr.tool_name
            # --- PASS 1 decisions (CPU-chain priority order) ---
            if deny_m[i] and self._enforcing(self.deny) and self._applies(self.deny, name):
                pid = int(deny_m[i]).bit_length() - 1
                word = self.deny.words[pid] if pid < len(self.deny.words) else "?"
                self._block(pr, responses, f"deny_filter: deny word {word!r} present")
                continue
            if (pii_m[i] and self._applies(self.pii, name)) or \
               (regex_m[i] and self._applies(self.regex, name)) or \
               (norm_m[i] and self._applies(self.normalizer, name)):
                rewrites.append((pr, i))
                continue
            if scores is not None and self._enforcing(self.moderation) and self._applies(self.moderation, name):
                msg = self._moderation_block(scores[i])
                if msg:
                    self._block(pr, responses, msg)
                    continue
            if harm_m[i] and self._enforcing(self.harm) and self._applies(self.harm, name):
                pid = int(harm_m[i]).bit_length() - 1
                cat = self.harm.cats[pid] if pid < len(self.harm.cats) else "?"
                self._block(pr, responses, f"harmful_content_detector: harmful content ({cat})")
                continue
            if self._enforcing(self.schema_guard):
                msg = self._schema_errs(pr, pr.arguments)
                if msg:
                    self._block(pr, responses, msg)
                    continue
            if hits[i] is not None:
                responses[pr.index] = jsonrpc.result_response(pr.req.id, hits[i]).to_bytes()
                self.cache_hits += 1
                continue
            if self.exact_cache is not None:
                k = hashlib.sha256(name.encode() + pr.arg_text).hexdigest()
                ent = self.exact_cache.store.get(k)
                if ent and time.monotonic() - ent[0] <= self.exact_cache.ttl:
                    responses[pr.index] = jsonrpc.result_response(pr.req.id, ent[1]).to_bytes()
                    self.cache_hits += 1
                    continue
            if self.breaker is not None:
                st = self.breaker._st(name)
                if time.monotonic() < st["open_until"]:
                    self._block(pr, responses, f"circuit_breaker: circuit open for tool {name}")
                    continue
            dispatch.append((pr, i, pr.arguments))

        # --- PASS 2: host rewrites + GPU re-scan of the rewritten subset ---
        if rewrites:
            self.slow_path += len(rewrites)
            rw_args: List[Any] = []
            rw_ok: List[Tuple[ParsedRequest, int, Any]] = []
            for pr, i in rewrites:
                status, payload = self._apply_rewrites(pr)
                if status == "__block__":
                    self._block(pr, responses, payload)
                    continue
                rw_ok.append((pr, i, payload))
            if rw_ok:
                texts2 = [canonical_text(a) for (_pr, _i, a) in rw_ok]
                scan2 = self._scan_pass(texts2, ("harm",), classify=True, cache_lookup=False)
                harm2 = scan2.get("mask_harm", np.zeros(len(rw_ok), dtype=np.int64))
                scores2 = scan2.get("scores")
                for j, (pr, i, args2) in enumerate(rw_ok):
                    name = pr.tool_name
                    if scores2 is not None and self._enforcing(self.moderation) and self._applies(self.moderation, name):
                        msg = self._moderation_block(scores2[j])
                        if msg:
                            self._block(pr, responses, msg)
                            continue
                    if harm2[j] and self._enforcing(self.harm) and self._applies(self.harm, name):
                        pid = int(harm2[j]).bit_length() - 1
                        cat = self.harm.cats[pid] if pid < len(self.harm.cats) else "?"
                        self._block(pr, responses, f"harmful_content_detector: harmful content ({cat})")
                        continue
                    if self._enforcing(self.schema_guard):
                        msg = self._schema_errs(pr, args2)
                        if msg:
                            self._block(pr, responses, msg)
                            continue
                    if hits[i] is not None:
                        responses[pr.index] = jsonrpc.result_response(pr.req.id, hits[i]).to_bytes()
                        self.cache_hits += 1
                        continue
                    if self.breaker is not None and time.monotonic() < self.breaker._st(name)["open_until"]:
                        self._block(pr, responses, f"circuit_breaker: circuit open for tool {name}")
                        continue
                    dispatch.append((pr, i, args2))

        # --- fan-out dispatch ---
        self.fast_path += len(dispatch)
        results = await self._dispatch_all(dispatch)

        # --- PASS 3: result post chain ---
        await self._post_pass(dispatch, results, responses, scan, th_arr, t0)

    async def _dispatch_all(self, dispatch: List[Tuple[ParsedRequest, int, Any]]
                            ) -> List[Tuple[Optional[dict], Optional[Exception]]]:
        """In-proc/local upstreams via a plain await loop (no Task churn);
        real-I/O upstreams concurrently."""
        from ..services.upstream import InProcUpstream

        ts = self.engine.tool_service
        results: List[Optional[Tuple[Optional[dict], Optional[Exception]]]] = [None] * len(dispatch)
        io_idx: List[int] = []
        for idx, (pr, _i, args) in enumerate(dispatch):
            tool = pr.tool
            itype = tool.get("integration_type", "MCP")
            if itype == "LOCAL":
                handler = ts._local_handlers.get(tool["name"])
                if handler is None:
                    results[idx] = (None, RuntimeError(f"no local handler for {tool['name']}"))
                    continue
                try:
                    value = await handler(args or {})
                    if not (isinstance(value, dict) and "content" in value):
                        value = {
                            "content": [{"type": "text", "text": value if isinstance(value, str) else json.dumps(value, default=str)}],
                            "structuredContent": value if isinstance(value, (dict, list)) else None,
                            "isError": False,
                        }
                    results[idx] = (value, None)
                except Exception as exc:
                    results[idx] = (None, exc)
            elif itype == "MCP":
                client = ts._upstreams.get(tool.get("gateway_id") or "")
                if client is None:
                    results[idx] = (None, RuntimeError(f"no upstream for {tool['name']}"))
                elif isinstance(client, InProcUpstream):
                    try:
                        results[idx] = (await client.call_tool(tool["original_name"], args or {}), None)
                    except Exception as exc:
                        results[idx] = (None, exc)
                else:
                    io_idx.append(idx)
            else:
                io_idx.append(idx)

        if io_idx:
            async def one(idx: int):
                pr, _i, args = dispatch[idx]
                try:
                    results[idx] = (await ts.dispatch(pr.tool, args), None)
                except Exception as exc:
                    results[idx] = (None, exc)

            await asyncio.gather(*(one(i) for i in io_idx))
        return results  # type: ignore[return-value]

    async def _post_pass(self, dispatch, results, responses, scan, th_arr, t0) -> None:
        """PASS 3: serialize results once; scan the serialized form for
        pii/regex/harm; flagged results take the host post chain; everyone
        gets toon + length guard; cache insert + metrics at the end."""
        n = len(dispatch)
        res_texts: List[bytes] = []
        ok_rows: List[int] = []  # indices into dispatch with a result (not exception)
        for idx, ((pr, _i, _a), (result, exc)) in enumerate(zip(dispatch, results)):
            if exc is None:
                res_texts.append(json.dumps(result, separators=(",", ":"), default=str).encode())
                ok_rows.append(idx)
            else:
                res_texts.append(b"")

        post_banks = tuple(b for b in ("pii", "regex", "harm") if b in self.banks)
        post_masks = None
        if ok_rows and post_banks:
            scan3 = self._scan_pass(res_texts, post_banks, classify=False, cache_lookup=False)
            post_masks = {b: scan3.get(f"mask_{b}") for b in post_banks}

        insert_rows: List[int] = []
        insert_hashes: List[int] = []
        insert_results: List[Any] = []
        agg: Dict[str, List[int]] = {}
        ms = (time.monotonic() - t0) * 1000.0

        for idx, ((pr, i, args), (result, exc)) in enumerate(zip(dispatch, results)):
            rid = pr.req.id
            tool_id = pr.tool.get("id", pr.tool_name)
            a = agg.setdefault(tool_id, [0, 0])
            if exc is not None:
                code = getattr(exc, "code", jsonrpc.SERVER_ERROR)
                responses[pr.index] = jsonrpc.error_response(rid, code, str(exc)).to_bytes()
                a[0] += 1
                a[1] += 1
                if self.breaker is not None:
                    self._breaker_record(pr.tool_name, True)
                continue

            orig_result = result
            flagged = False
            if post_masks is not None:
                for b, m in post_masks.items():
                    if m is not None and m[idx]:
                        flagged = True
                        break
            if flagged:
                result, blocked_msg = self._host_post_rewrite(pr, result)
                self.post_rewrites += 1
                if blocked_msg:
                    self._block(pr, responses, blocked_msg)
                    a[0] += 1
                    a[1] += 1
                    continue

            # output schema check (tool_service parity)
            oschema = pr.tool.get("output_schema")
            if oschema and isinstance(result, dict):
                from ..utils.jsonschema import validate as _validate

                payload = result.get("structuredContent", result)
                errs = _validate(payload, oschema)
                if errs:
                    result = {"content": [{"type": "text", "text": "output schema violation: " + "; ".join(errs[:3])}],
                              "isError": True}

            result = self._toon_guard(pr, result)
            is_err = bool(isinstance(result, dict) and result.get("isError"))
            a[0] += 1
            a[1] += 1 if is_err else 0
            if self.breaker is not None:
                self._breaker_record(pr.tool_name, is_err)

            if result is orig_result and res_texts[idx]:
                # untouched result: splice the already-serialized form (one dumps total)
                responses[pr.index] = b'{"jsonrpc":"2.0","id":' + json.dumps(rid).encode() + b',"result":' + res_texts[idx] + b"}"
            else:
                responses[pr.index] = jsonrpc.result_response(rid, result).to_bytes()
            if not is_err:
                if self.exact_cache is not None:
                    k = hashlib.sha256(pr.tool_name.encode() + pr.arg_text).hexdigest()
                    self.exact_cache.store[k] = (time.monotonic(), result)
                if self.semcache is not None:
                    insert_rows.append(i)
                    insert_hashes.append(int(th_arr[i]))
                    insert_results.append(result)

        if insert_rows and scan.get("_feats") is not None:
            self.semcache.insert_batch(scan["_feats"], insert_rows, insert_hashes, insert_results)
        for tool_id, (cnt, errs) in agg.items():
            self.engine.metrics.record_aggregate(tool_id, cnt, errs, ms)

    def _host_post_rewrite(self, pr: ParsedRequest, result: dict) -> Tuple[dict, Optional[str]]:
        """Exact host post chain for flagged results: regex(20) → pii(30) →
        harm(60), matching plugin post-hook semantics."""
        name = pr.tool_name
        if self.regex is not None and self._applies(self.regex, name):
            result = _walk_strings(result, self.regex.apply_rules)
        if self.pii is not None and self._applies(self.pii, name):
            found: List[str] = []

            def fn(s: str) -> str:
                masked, f = self.pii.mask_text(s)
                found.extend(f)
                return masked

            new = _walk_strings(result, fn)
            if found and self.pii.action == "block" and self._enforcing(self.pii):
                return result, f"pii_filter: PII detected: {sorted(set(found))}"
            if found and self.pii.action == "mask":
                result = new
        if self.harm is not None and self._enforcing(self.harm) and self._applies(self.harm, name):
            hay = json.dumps(result, separators=(",", ":"), sort_keys=True, default=str).lower()
            for phrase, cat in zip(self.harm.phrases, self.harm.cats):
                if phrase.lower() in hay:
                    return result, f"harmful_content_detector: harmful content ({cat})"
        return result, None

    def _toon_guard(self, pr: ParsedRequest, result: dict) -> dict:
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

    def _breaker_record(self, name: str, is_error: bool) -> None:
        b = self.breaker
        st = b._st(name)
        st["results"].append(is_error)
        if len(st["results"]) > b.window:
            st["results"] = st["results"][-b.window:]
        if len(st["results"]) >= b.window and (sum(st["results"]) / len(st["results"])) >= b.error_threshold:
            st["open_until"] = time.monotonic() + b.cooldown
            st["results"] = []

    def stats(self) -> Dict[str, Any]:
        out = {
            "batches": self.batches, "requests": self.requests, "fast_path": self.fast_path,
            "slow_path": self.slow_path, "blocked": self.blocked, "cache_hits": self.cache_hits,
            "post_rewrites": self.post_rewrites,
            "banks": {k: {"states": v.n_states, "classes": v.n_classes} for k, v in self.banks.items()},
        }
        if self.semcache is not None:
            out["semcache"] = self.semcache.stats()
        return out
