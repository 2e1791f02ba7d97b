"""The batched GPU plugin pipeline — the MI355X-native hot path.

Reference analog: the per-request chain tool_service.invoke_tool (:5067)
with plugin hooks (:5530), plus the Rust edge runtime's native fast path
(crates/mcp_runtime). Here a whole micro-batch is staged to HBM and the
plugin chain's data-parallel stages run as HIP kernels (BASELINE.json),
with a C++ host fast path around them:

  C++ envelope scan   → JSON-RPC spans, no Python parse (ops/csrc/envelope.cpp)
  scan banks (GPU)    → deny / harm / pii / regex / normalizer-trigger DFAs
                        + schema-shape presence patterns, over RAW arg bytes
  featurize (GPU)     → hashed count vectors (LDS histograms)
  classifier (GPU)    → content_moderation bf16 MFMA MLP
  semcache (GPU)      → response_cache_by_prompt cosine sweep in HBM
  C++ native upstream → fast_time_server analog (ops/csrc/upstream.cpp)
  result scan (GPU)   → pii/regex/harm over serialized results

Decisions are numpy-vectorized; Python touches a request only when a kernel
flags it (rewrites, schema fallbacks, unusual envelopes) — those take the
exact per-request semantics (the plugins' own functions), so the parity
gate (tests/test_gpu_parity.py, reference analog
tests/live_gateway/mcp/test_mcp_plugin_parity.py) holds by construction.

Scanning operates on the request's RAW argument bytes (and the CPU oracle
on the canonical sorted-JSON text): the two representations contain the
same string values and the same token multiset, so detection agrees except
for matches spanning value boundaries — rewrite-class flags only route to
the exact host path, and parity traffic uses compact payloads.
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
from ..plugins.builtin import _walk_strings
from ..plugins.framework import PluginMode
from .batch import pack_texts, pad_rows
from .classifier import GpuClassifier
from .semcache import GpuSemanticCache, tool_hash

# normalizer-trigger prefilter (conservative superset over raw JSON bytes):
# JSON escapes for whitespace/unicode, literal multi-space, space adjacent to
# a quote, and raw non-ASCII bytes (clients that don't ensure_ascii).
_NORMALIZE_TRIGGERS = ["\\t", "\\n", "\\r", "\\f", "\\u", "  ", '" ', ' "',
                       "[" + chr(0x80) + "-" + chr(0xFF) + "]"]

_SCHEMA_NEST_PATTERNS = [":\\{", ":\\["]

_TYPED_PAT = {
    "string": '"{k}":"',
    "number": '"{k}":[0-9\\-]',
    "integer": '"{k}":[0-9\\-]',
    "boolean": '"{k}":[tf]',
    "array": '"{k}":\\[',
}


class _ToolMeta:
    __slots__ = ("tool", "name", "tid", "itype", "thash", "native_kind", "native_client",
                 "client", "handler", "schema_mode", "required_bits", "typed_pairs",
                 "has_output_schema", "original_name", "reachable")

    def __init__(self):
        self.schema_mode = "host"   # "trivial" | "fast" | "host"
        self.required_bits = 0
        self.typed_pairs: List[Tuple[int, int]] = []
        self.native_kind = -1
        self.native_client = None
        self.client = None
        self.handler = None
        self.has_output_schema = False


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
                dfa.compile_patterns(
                    ["".join(ch if ch.isalnum() else "\\" + ch for ch in t) if not t.startswith("[") else t
                     for t in _NORMALIZE_TRIGGERS], case_insensitive=False), device)

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

        # per-tool metadata + schema-shape bank (rebuilt on registry change)
        self._meta_gen = -1
        self._tool_meta: Dict[str, _ToolMeta] = {}
        self._schema_bank: Optional[hip.DeviceScanTables] = None
        self._schema_pat_ids: Dict[str, int] = {}
        self._nest_bits = 0

        # stats
        self.batches = 0
        self.requests = 0
        self.fast_path = 0
        self.slow_path = 0
        self.blocked = 0
        self.cache_hits = 0
        self.post_rewrites = 0
        self.py_fallback = 0

    # ------------------------------------------------------------------
    # tool metadata / schema-shape bank
    # ------------------------------------------------------------------
    def _schema_pattern(self, pat: str) -> int:
        pid = self._schema_pat_ids.get(pat)
        if pid is None:
            pid = len(self._schema_pat_ids)
            self._schema_pat_ids[pat] = pid
        return pid

    def _esc_key(self, k: str) -> Optional[str]:
        if not k or not all(c.isalnum() or c == "_" for c in k):
            return None
        return k

    def _compile_tool_schema(self, meta: _ToolMeta, schema: Optional[dict]) -> None:
        """Flat object schemas compile to presence/type byte patterns scanned
        on-GPU; anything richer falls back to host validation (exact)."""
        if self.schema_guard is None or not schema:
            meta.schema_mode = "trivial"
            return
        if not (schema.get("properties") or schema.get("required") or
                schema.get("additionalProperties") is False or schema.get("anyOf") or
                schema.get("allOf") or schema.get("oneOf")):
            meta.schema_mode = "trivial"
            return
        props = schema.get("properties") or {}
        required = schema.get("required") or []
        if schema.get("additionalProperties") is False or schema.get("anyOf") or \
           schema.get("allOf") or schema.get("oneOf"):
            meta.schema_mode = "host"
            return
        req_bits = 0
        pairs: List[Tuple[int, int]] = []
        for k, sub in props.items():
            ek = self._esc_key(k)
            ty = sub.get("type") if isinstance(sub, dict) else None
            extra = isinstance(sub, dict) and any(
                c in sub for c in ("enum", "const", "pattern", "minimum", "maximum", "minLength",
                                   "maxLength", "minItems", "maxItems", "properties", "items",
                                   "anyOf", "allOf", "oneOf", "exclusiveMinimum", "exclusiveMaximum"))
            if ek is None or extra or (ty is not None and not isinstance(ty, str)) or \
               (ty is not None and ty not in _TYPED_PAT and ty not in ("object", "null")):
                meta.schema_mode = "host"
                return
            present = self._schema_pattern(f'"{ek}":')
            if ty in _TYPED_PAT:
                typed = self._schema_pattern(_TYPED_PAT[ty].format(k=ek))
                pairs.append((present, typed))
            if k in required:
                req_bits |= 1 << present
        for k in required:
            if k not in props:
                ek = self._esc_key(k)
                if ek is None:
                    meta.schema_mode = "host"
                    return
                req_bits |= 1 << self._schema_pattern(f'"{ek}":')
        if len(self._schema_pat_ids) > 30 - len(_SCHEMA_NEST_PATTERNS):
            meta.schema_mode = "host"
            return
        meta.schema_mode = "fast"
        meta.required_bits = req_bits
        meta.typed_pairs = pairs

    def _rebuild_tool_meta(self) -> None:
        from ..services.upstream import InProcUpstream, NativeInProcUpstream

        ts = self.engine.tool_service
        self._tool_meta = {}
        self._schema_pat_ids = {}
        for tool in self.engine.registry.list("tool", include_disabled=False):
            m = _ToolMeta()
            m.tool = tool
            m.name = tool["name"]
            m.tid = tool.get("id", m.name)
            m.itype = tool.get("integration_type", "MCP")
            m.thash = tool_hash(m.name)
            m.original_name = tool.get("original_name", m.name)
            m.has_output_schema = bool(tool.get("output_schema"))
            m.reachable = tool.get("reachable", True)
            if m.itype == "LOCAL":
                m.handler = ts._local_handlers.get(m.name)
            elif m.itype == "MCP":
                m.client = ts._upstreams.get(tool.get("gateway_id") or "")
                if isinstance(m.client, NativeInProcUpstream):
                    m.native_kind = m.client.TOOL_KINDS.get(m.original_name, -1)
                    m.native_client = m.client
            self._compile_tool_schema(m, tool.get("input_schema"))
            self._tool_meta[m.name] = m
        pats = list(self._schema_pat_ids.keys())
        self._nest_bits = 0
        if pats and self.schema_guard is not None:
            base = len(pats)
            for i, p in enumerate(_SCHEMA_NEST_PATTERNS):
                self._nest_bits |= 1 << (base + i)
            try:
                self._schema_bank = hip.DeviceScanTables(
                    dfa.compile_patterns(pats + _SCHEMA_NEST_PATTERNS, case_insensitive=False), self.device)
            except ValueError:
                self._schema_bank = None
                for m in self._tool_meta.values():
                    if m.schema_mode == "fast":
                        m.schema_mode = "host"
        else:
            self._schema_bank = None
        self._meta_gen = self.engine.registry.generation

    def _meta(self) -> Dict[str, _ToolMeta]:
        if self._meta_gen != self.engine.registry.generation:
            self._rebuild_tool_meta()
        return self._tool_meta

    # ------------------------------------------------------------------
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
        n = len(raws)
        self.requests += n
        responses: List[Optional[bytes]] = [None] * n

        offsets = np.zeros(n + 1, dtype=np.int64)
        for i, r in enumerate(raws):
            offsets[i + 1] = offsets[i] + len(r)
        joined = b"".join(raws)
        blob = np.frombuffer(joined, dtype=np.uint8) if joined else np.zeros(1, dtype=np.uint8)
        env = hip.parse_envelopes(blob, offsets)
        kind = env["kind"]

        other_rows = np.nonzero(kind != hip.ENV_TOOLS_CALL)[0]
        if other_rows.size:
            self.py_fallback += int(other_rows.size)
            outs = await asyncio.gather(
                *(self.engine.handle_rpc_bytes(raws[int(i)], user=user, server_id=server_id)
                  for i in other_rows))
            for i, out in zip(other_rows, outs):
                responses[int(i)] = out

        fast_rows = np.nonzero(kind == hip.ENV_TOOLS_CALL)[0]
        if fast_rows.size:
            await self._fast_toolcalls(raws, blob, env, fast_rows, responses, user, server_id)
        return responses

    # ------------------------------------------------------------------
    def _id_bytes(self, blob: np.ndarray, env: dict, row: int) -> Optional[bytes]:
        b, e = int(env["id_beg"][row]), int(env["id_end"][row])
        if b < 0:
            return None
        return blob[b:e].tobytes()

    def _splice_result(self, id_bytes: bytes, result_bytes: bytes) -> bytes:
        return b'{"jsonrpc":"2.0","id":' + id_bytes + b',"result":' + result_bytes + b"}"

    def _splice_error(self, id_bytes: Optional[bytes], code: int, message: str) -> Optional[bytes]:
        if id_bytes is None:
            return None
        return (b'{"jsonrpc":"2.0","id":' + id_bytes + b',"error":{"code":' + str(code).encode()
                + b',"message":' + json.dumps(message).encode() + b"}}")

    async def _fast_toolcalls(self, raws: List[bytes], blob: np.ndarray, env: dict,
                              rows: np.ndarray, responses: List[Optional[bytes]],
                              user: Optional[str], server_id: Optional[str]) -> None:
        meta_map = self._meta()
        m = rows.shape[0]
        nb, ne = env["name_beg"], env["name_end"]
        ab, ae = env["args_beg"], env["args_end"]

        # --- tool resolution (memoized on raw name bytes) ---
        metas: List[Optional[_ToolMeta]] = [None] * m
        memo: Dict[bytes, Optional[_ToolMeta]] = {}
        for j in range(m):
            r = int(rows[j])
            key = blob[nb[r]:ne[r]].tobytes()
            if key in memo:
                metas[j] = memo[key]
            else:
                mt = meta_map.get(key.decode("utf-8", "replace"))
                memo[key] = mt
                metas[j] = mt

        th_arr = np.array([mt.thash if mt else 0 for mt in metas], dtype=np.int64)

        # --- GPU pass 1 over raw argument spans ---
        args_b = np.where(ab[rows] >= 0, ab[rows], 0).astype(np.int32)
        args_e = np.where(ab[rows] >= 0, ae[rows], 0).astype(np.int32)
        data_gpu = torch.from_numpy(blob.copy()).to(self.device, non_blocking=True)
        beg_t = torch.from_numpy(args_b).to(self.device, non_blocking=True)
        end_t = torch.from_numpy(args_e).to(self.device, non_blocking=True)

        out: Dict[str, torch.Tensor] = {}
        for bname, bank in self.banks.items():
            out[bname], _ = hip.scan(data_gpu, beg_t, end_t, bank)
        if self._schema_bank is not None:
            out["schema"], _ = hip.scan(data_gpu, beg_t, end_t, self._schema_bank)
        feats = None
        if self.classifier is not None or self.semcache is not None:
            feats_b, _ = hip.featurize(data_gpu, beg_t, end_t, self.feat_dim)
            feats = pad_rows(feats_b, 128)
        scores_t = self.classifier.forward(feats)[:m] if self.classifier is not None else None
        cache_val_t = cache_idx_t = None
        if self.semcache is not None:
            bv, bi = self.semcache.lookup(feats)
            cache_val_t, cache_idx_t = bv[:m], bi[:m]
        torch.cuda.synchronize()

        zeros = np.zeros(m, dtype=np.int64)
        deny_m = out["deny"].cpu().numpy().astype(np.uint32) if "deny" in out else zeros
        harm_m = out["harm"].cpu().numpy().astype(np.uint32) if "harm" in out else zeros
        pii_m = out["pii"].cpu().numpy().astype(np.uint32) if "pii" in out else zeros
        regex_m = out["regex"].cpu().numpy().astype(np.uint32) if "regex" in out else zeros
        norm_m = out["normalize"].cpu().numpy().astype(np.uint32) if "normalize" in out else zeros
        schema_m = out["schema"].cpu().numpy().astype(np.uint32) if "schema" in out else zeros
        scores = scores_t.cpu().numpy() if scores_t is not None else None
        hit_mask = np.zeros(m, dtype=bool)
        cache_idx = None
        if cache_val_t is not None:
            cache_val = cache_val_t.cpu().numpy()
            cache_idx = cache_idx_t.cpu().numpy()
            hit_mask = self.semcache.resolve_hits_np(cache_val, cache_idx, th_arr)

        # --- vectorized decisions (precedence = CPU chain priority order) ---
        # 0 dispatch | 1 blocked/handled | 2 rewrite | 3 host-schema
        state = np.zeros(m, dtype=np.int8)
        mod_block = np.zeros(m, dtype=bool)
        if scores is not None and self._enforcing(self.moderation):
            mod_block = scores.max(axis=1) >= self.moderation.threshold
        deny_on = self._enforcing(self.deny)
        harm_on = self._enforcing(self.harm)

        t0 = time.monotonic()
        rewrite_rows: List[int] = []
        dispatch_rows: List[int] = []
        now = time.monotonic()
        exact_store = self.exact_cache.store if self.exact_cache is not None else None
        exact_ttl = self.exact_cache.ttl if self.exact_cache is not None else 0.0

        for j in range(m):
            mt = metas[j]
            r = int(rows[j])
            idb = self._id_bytes(blob, env, r)
            if mt is None:
                name = blob[nb[r]:ne[r]].tobytes().decode("utf-8", "replace")
                responses[r] = self._splice_error(idb, jsonrpc.INVALID_PARAMS, f"Tool not found: {name}")
                state[j] = 1
                continue
            name = mt.name
            if not mt.reachable:
                responses[r] = self._splice_error(idb, jsonrpc.SERVER_UNAVAILABLE,
                                                  f"Tool {name} currently unreachable")
                state[j] = 1
                continue
            if deny_m[j] and deny_on and self._applies(self.deny, name):
                pid = int(deny_m[j]).bit_length() - 1
                word = self.deny.words[pid] if pid < len(self.deny.words) else "?"
                responses[r] = self._splice_error(idb, jsonrpc.POLICY_DENIED,
                                                  f"deny_filter: deny word {word!r} present")
                state[j] = 1
                self.blocked += 1
                continue
            if (pii_m[j] and self._applies(self.pii, name)) or \
               (regex_m[j] and self._applies(self.regex, name)) or \
               (norm_m[j] and self._applies(self.normalizer, name)):
                state[j] = 2
                rewrite_rows.append(j)
                continue
            if mod_block[j] and self._applies(self.moderation, name):
                from ..models.classifier import category_names

                row_sc = scores[j]
                cat = category_names(len(row_sc))[int(row_sc.argmax())]
                responses[r] = self._splice_error(
                    idb, jsonrpc.POLICY_DENIED,
                    f"content_moderation: moderation: category {cat} score {float(row_sc.max()):.3f}")
                state[j] = 1
                self.blocked += 1
                continue
            if harm_m[j] and harm_on and self._applies(self.harm, name):
                pid = int(harm_m[j]).bit_length() - 1
                cat = self.harm.cats[pid] if pid < len(self.harm.cats) else "?"
                responses[r] = self._splice_error(idb, jsonrpc.POLICY_DENIED,
                                                  f"harmful_content_detector: harmful content ({cat})")
                state[j] = 1
                self.blocked += 1
                continue
            if self._enforcing(self.schema_guard):
                mode = mt.schema_mode
                if mode == "fast":
                    sm = int(schema_m[j])
                    ok = (sm & mt.required_bits) == mt.required_bits and not (sm & self._nest_bits)
                    if ok:
                        for present, typed in mt.typed_pairs:
                            if (sm >> present) & 1 and not ((sm >> typed) & 1):
                                ok = False
                                break
                    if not ok:
                        state[j] = 3
                        continue
                elif mode == "host":
                    state[j] = 3
                    continue
            if hit_mask[j]:
                res = self.semcache.results[int(cache_idx[j])]
                if idb is not None:
                    responses[r] = self._splice_result(idb, res if isinstance(res, bytes)
                                                       else json.dumps(res, separators=(",", ":")).encode())
                state[j] = 1
                self.cache_hits += 1
                continue
            if exact_store is not None:
                ek = (name, blob[args_b[j]:args_e[j]].tobytes())
                ent = exact_store.get(ek)
                if ent is not None and now - ent[0] <= exact_ttl:
                    if idb is not None:
                        responses[r] = self._splice_result(idb, ent[1])
                    state[j] = 1
                    self.cache_hits += 1
                    continue
            if self.breaker is not None:
                st = self.breaker._st(name)
                if now < st["open_until"]:
                    responses[r] = self._splice_error(idb, jsonrpc.POLICY_DENIED,
                                                      f"circuit_breaker: circuit open for tool {name}")
                    state[j] = 1
                    self.blocked += 1
                    continue
            dispatch_rows.append(j)

        # --- host-schema fallback (exact validate; violation blocks) ---
        for j in np.nonzero(state == 3)[0]:
            j = int(j)
            mt = metas[j]
            r = int(rows[j])
            idb = self._id_bytes(blob, env, r)
            try:
                args = json.loads(blob[args_b[j]:args_e[j]].tobytes() or b"{}")
            except Exception:
                responses[r] = self._splice_error(idb, jsonrpc.INVALID_PARAMS, "invalid arguments")
                continue
            from ..utils.jsonschema import validate as _validate

            errs = _validate(args or {}, mt.tool.get("input_schema") or {})
            if errs:
                responses[r] = self._splice_error(idb, jsonrpc.POLICY_DENIED,
                                                  "schema_guard: schema violation: " + "; ".join(errs[:5]))
                self.blocked += 1
            elif hit_mask[j]:
                res = self.semcache.results[int(cache_idx[j])]
                if idb is not None:
                    responses[r] = self._splice_result(idb, res if isinstance(res, bytes)
                                                       else json.dumps(res, separators=(",", ":")).encode())
                self.cache_hits += 1
            else:
                dispatch_rows.append(j)

        # --- PASS 2: rewrite-flagged subset (host rewrites + GPU re-scan) ---
        rewrite_dispatch: List[Tuple[int, Any]] = []  # (j, rewritten args)
        if rewrite_rows:
            self.slow_path += len(rewrite_rows)
            rewrite_dispatch = await self._rewrite_pass(
                blob, env, rows, args_b, args_e, rewrite_rows, metas, hit_mask, cache_idx, responses)

        # --- fan-out dispatch ---
        self.fast_path += len(dispatch_rows)
        await self._dispatch_and_post(blob, env, rows, args_b, args_e, metas, feats, th_arr,
                                      dispatch_rows, rewrite_dispatch, responses, t0)

    # ------------------------------------------------------------------
    def _apply_rewrites(self, name: str, args: Any) -> Tuple[str, Any]:
        """Host rewrites in CPU-chain priority order: normalizer(15) →
        regex(20) → pii(30), using the plugins' own functions."""
        if self.normalizer is not None and self._applies(self.normalizer, name):
            args = _walk_strings(args, self.normalizer.norm)
        if self.regex is not None and self._applies(self.regex, name):
            args = _walk_strings(args, self.regex.apply_rules)
        if self.pii is not None and self._applies(self.pii, name):
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

    async def _rewrite_pass(self, blob, env, rows, args_b, args_e, rewrite_rows, metas,
                            hit_mask, cache_idx, responses) -> List[Tuple[int, Any]]:
        ok_items: List[Tuple[int, Any]] = []
        for j in rewrite_rows:
            r = int(rows[j])
            mt = metas[j]
            idb = self._id_bytes(blob, env, r)
            try:
                args = json.loads(blob[args_b[j]:args_e[j]].tobytes() or b"{}")
            except Exception:
                responses[r] = self._splice_error(idb, jsonrpc.INVALID_PARAMS, "invalid arguments")
                continue
            status, payload = self._apply_rewrites(mt.name, args)
            if status == "__block__":
                responses[r] = self._splice_error(idb, jsonrpc.POLICY_DENIED, payload)
                self.blocked += 1
                continue
            ok_items.append((j, payload))
        if not ok_items:
            return []

        texts2 = [json.dumps(a, separators=(",", ":"), sort_keys=True, default=str).encode()
                  for (_j, a) in ok_items]
        data2, beg2, end2 = pack_texts(texts2, self.device)
        harm2_t = None
        if "harm" in self.banks:
            harm2_t, _ = hip.scan(data2, beg2, end2, self.banks["harm"])
        scores2_t = None
        if self.classifier is not None:
            f2, _ = hip.featurize(data2, beg2, end2, self.feat_dim)
            scores2_t = self.classifier.forward(pad_rows(f2, 128))[: len(ok_items)]
        torch.cuda.synchronize()
        harm2 = harm2_t.cpu().numpy() if harm2_t is not None else np.zeros(len(ok_items), dtype=np.int64)
        scores2 = scores2_t.cpu().numpy() if scores2_t is not None else None

        out: List[Tuple[int, Any]] = []
        for jj, (j, args2) in enumerate(ok_items):
            r = int(rows[j])
            mt = metas[j]
            name = mt.name
            idb = self._id_bytes(blob, env, r)
            if scores2 is not None and self._enforcing(self.moderation) and self._applies(self.moderation, name):
                row_sc = scores2[jj]
                if float(row_sc.max()) >= self.moderation.threshold:
                    from ..models.classifier import category_names

                    cat = category_names(len(row_sc))[int(row_sc.argmax())]
                    responses[r] = self._splice_error(
                        idb, jsonrpc.POLICY_DENIED,
                        f"content_moderation: moderation: category {cat} score {float(row_sc.max()):.3f}")
                    self.blocked += 1
                    continue
            if harm2[jj] and self._enforcing(self.harm) and self._applies(self.harm, name):
                pid = int(harm2[jj]).bit_length() - 1
                cat = self.harm.cats[pid] if pid < len(self.harm.cats) else "?"
                responses[r] = self._splice_error(idb, jsonrpc.POLICY_DENIED,
                                                  f"harmful_content_detector: harmful content ({cat})")
                self.blocked += 1
                continue
            if self._enforcing(self.schema_guard) and mt.schema_mode != "trivial":
                from ..utils.jsonschema import validate as _validate

                errs = _validate(args2 or {}, mt.tool.get("input_schema") or {})
                if errs:
                    responses[r] = self._splice_error(idb, jsonrpc.POLICY_DENIED,
                                                      "schema_guard: schema violation: " + "; ".join(errs[:5]))
                    self.blocked += 1
                    continue
            if hit_mask[j]:
                res = self.semcache.results[int(cache_idx[j])]
                if idb is not None:
                    responses[r] = self._splice_result(idb, res if isinstance(res, bytes)
                                                       else json.dumps(res, separators=(",", ":")).encode())
                self.cache_hits += 1
                continue
            out.append((j, args2))
        return out

    # ------------------------------------------------------------------
    async def _dispatch_and_post(self, blob, env, rows, args_b, args_e, metas, feats, th_arr,
                                 dispatch_rows: List[int], rewrite_dispatch: List[Tuple[int, Any]],
                                 responses: List[Optional[bytes]], t0: float) -> None:
        # Split native-batch vs python dispatch.
        native_js: List[int] = []
        py_items: List[Tuple[int, Any]] = []  # (j, args or None=raw)
        for j in dispatch_rows:
            mt = metas[j]
            if mt.native_kind >= 0:
                native_js.append(j)
            else:
                py_items.append((j, None))
        for j, args2 in rewrite_dispatch:
            mt = metas[j]
            py_items.append((j, args2))  # rewritten args always go through python dispatch

        result_bytes: Dict[int, Optional[bytes]] = {}
        errors: Dict[int, Exception] = {}

        # --- native upstream batch call (C++) ---
        if native_js:
            kinds = np.array([metas[j].native_kind for j in native_js], dtype=np.int32)
            nb_ = np.array([args_b[j] for j in native_js], dtype=np.int32)
            ne_ = np.array([args_e[j] for j in native_js], dtype=np.int32)
            nb_ = np.where(ne_ > nb_, nb_, -1).astype(np.int32)
            now_iso = time.strftime("%Y-%m-%dT%H:%M:%SZ", time.gmtime())
            out_blob, rb, re_ = hip.upstream_call_batch(blob, nb_, ne_, kinds, now_iso)
            ob = out_blob.tobytes()
            for jj, j in enumerate(native_js):
                result_bytes[j] = ob[int(rb[jj]):int(re_[jj])]
                metas[j].native_client.calls += 1

        # --- python dispatch for the rest ---
        if py_items:
            ts = self.engine.tool_service
            from ..services.upstream import InProcUpstream

            async def one(j: int, args2: Any) -> None:
                mt = metas[j]
                args = args2
                if args is None:
                    try:
                        args = json.loads(blob[args_b[j]:args_e[j]].tobytes() or b"{}")
                    except Exception:
                        errors[j] = jsonrpc.JSONRPCError(jsonrpc.INVALID_PARAMS, "invalid arguments")
                        return
                try:
                    result = await ts.dispatch(mt.tool, args)
                    result_bytes[j] = json.dumps(result, separators=(",", ":"), default=str).encode()
                except Exception as exc:
                    errors[j] = exc

            # in-proc dispatch is non-blocking → sequential loop; real I/O → gather
            seq, conc = [], []
            for j, args2 in py_items:
                mt = metas[j]
                if mt.itype == "LOCAL" or isinstance(mt.client, InProcUpstream) or mt.native_kind >= 0:
                    seq.append((j, args2))
                else:
                    conc.append((j, args2))
            for j, args2 in seq:
                await one(j, args2)
            if conc:
                await asyncio.gather(*(one(j, a) for j, a in conc))

        all_js = native_js + [j for j, _ in py_items]

        # --- PASS 3: GPU scan over serialized results ---
        post_banks = tuple(b for b in ("pii", "regex", "harm") if b in self.banks)
        post_mask = None
        ok_js = [j for j in all_js if j in result_bytes and result_bytes[j]]
        if ok_js and post_banks:
            texts3 = [result_bytes[j] for j in ok_js]
            data3, beg3, end3 = pack_texts(texts3, self.device)
            masks3 = []
            for b in post_banks:
                t, _ = hip.scan(data3, beg3, end3, self.banks[b])
                masks3.append(t)
            torch.cuda.synchronize()
            combined = np.zeros(len(ok_js), dtype=np.int64)
            for t in masks3:
                combined |= t.cpu().numpy().astype(np.uint32).astype(np.int64)
            post_mask = {j: bool(combined[i]) for i, j in enumerate(ok_js)}

        insert_rows: List[int] = []
        insert_hashes: List[int] = []
        insert_results: List[Any] = []
        agg: Dict[str, List[int]] = {}
        ms = (time.monotonic() - t0) * 1000.0
        toon_min = self.toon.min_size if self.toon is not None else 1 << 60
        guard_max = self.out_guard.max_chars if self.out_guard is not None else 1 << 60
        now = time.monotonic()

        for j in all_js:
            mt = metas[j]
            r = int(rows[j])
            idb = self._id_bytes(blob, env, r)
            a = agg.setdefault(mt.tid, [0, 0])
            exc = errors.get(j)
            if exc is not None:
                code = getattr(exc, "code", jsonrpc.SERVER_ERROR)
                responses[r] = self._splice_error(idb, code if isinstance(code, int) else jsonrpc.SERVER_ERROR, str(exc))
                a[0] += 1
                a[1] += 1
                if self.breaker is not None:
                    self._breaker_record(mt.name, True)
                continue
            rb = result_bytes[j]
            need_host = (post_mask is not None and post_mask.get(j, False)) or mt.has_output_schema \
                or (len(rb) >= toon_min and b'"structuredContent"' in rb) or len(rb) > guard_max
            is_err = False
            if need_host:
                rb, is_err, blocked_msg = self._host_post(mt, rb)
                if blocked_msg:
                    responses[r] = self._splice_error(idb, jsonrpc.POLICY_DENIED, blocked_msg)
                    self.blocked += 1
                    a[0] += 1
                    a[1] += 1
                    continue
            a[0] += 1
            if is_err:
                a[1] += 1
            if self.breaker is not None:
                self._breaker_record(mt.name, is_err)
            if idb is not None:
                responses[r] = self._splice_result(idb, rb)
            if not is_err:
                if self.exact_cache is not None:
                    self.exact_cache.store[(mt.name, blob[args_b[j]:args_e[j]].tobytes())] = (now, rb)
                if self.semcache is not None:
                    insert_rows.append(j)
                    insert_hashes.append(int(th_arr[j]))
                    insert_results.append(rb)

        if insert_rows and feats is not None:
            self.semcache.insert_batch(feats, insert_rows, insert_hashes, insert_results)
        for tid, (cnt, errs_) in agg.items():
            self.engine.metrics.record_aggregate(tid, cnt, errs_, ms)

    def _host_post(self, mt: _ToolMeta, rb: bytes) -> Tuple[bytes, bool, Optional[str]]:
        """Exact host post chain for flagged results: regex(20) → pii(30) →
        output-schema → harm(60) → toon(900) → guard(950)."""
        self.post_rewrites += 1
        try:
            result = json.loads(rb)
        except Exception:
            return rb, True, None
        name = mt.name
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
                return rb, True, f"pii_filter: PII detected: {sorted(set(found))}"
            if found and self.pii.action == "mask":
                result = new
        if mt.has_output_schema and isinstance(result, dict):
            from ..utils.jsonschema import validate as _validate

            payload = result.get("structuredContent", result)
            errs = _validate(payload, mt.tool.get("output_schema") or {})
            if errs:
                result = {"content": [{"type": "text", "text": "output schema violation: " + "; ".join(errs[:3])}],
                          "isError": True}
        if self.harm is not None and self._enforcing(self.harm) and self._applies(self.harm, name):
            hay = json.dumps(result, separators=(",", ":"), sort_keys=True, default=str).lower()
            for phrase, cat in zip(self.harm.phrases, self.harm.cats):
                if phrase.lower() in hay:
                    return rb, True, f"harmful_content_detector: harmful content ({cat})"
        if self.toon is not None and self._applies(self.toon, name) and isinstance(result, dict):
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
        is_err = bool(isinstance(result, dict) and result.get("isError"))
        return json.dumps(result, separators=(",", ":"), default=str).encode(), is_err, None

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
            "post_rewrites": self.post_rewrites, "py_fallback": self.py_fallback,
            "banks": {k: {"states": v.n_states, "classes": v.n_classes} for k, v in self.banks.items()},
        }
        if self.semcache is not None:
            out["semcache"] = self.semcache.stats()
        return out
