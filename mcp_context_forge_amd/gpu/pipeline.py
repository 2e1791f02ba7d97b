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

Raw-bytes-vs-decoded-text containment (tested invariant —
tests/test_parity_fuzz.py runs adversarial generators against the CPU
chain):

  * literal deny/harm words: a RAW match implies a DECODED match (words
    are alphanumeric; no word can span JSON structural separators), so
    raw-scan blocks are sound;
  * content that can decode DIFFERENTLY from its raw bytes (JSON escapes,
    non-ASCII) always trips the escape-trigger bank, which routes the row
    to the exact host path where deny (pre-rewrite, CPU order), rewrites,
    harm, moderation and schema run over decoded text;
  * schema-shape fast patterns cannot false-positive from key lookalikes
    inside string values (embedded quotes are escaped, which breaks the
    byte pattern) and any nesting defers to exact host validation;
  * the one intentional approximation: the moderation classifier featurizes
    raw bytes for rows with NO escape triggers — wire forms that differ
    from canonical text only by whitespace/key order can score differently
    near the threshold. Rows with triggers are re-scored over canonical
    text on the host path.
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
                 "has_output_schema", "original_name", "reachable", "host_chain", "a2a_fast")

    def __init__(self):
        self.schema_mode = "host"   # "trivial" | "fast" | "host"
        self.required_bits = 0
        self.typed_pairs: List[Tuple[int, int]] = []
        self.native_kind = -1
        self.native_client = None
        self.client = None
        self.handler = None
        self.has_output_schema = False
        self.host_chain = False     # per-tool plugin binding forces the CPU chain
        self.a2a_fast = False       # in-proc agent, hooks fully GPU-covered


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

        # plugins the fast path MODELS (as GPU banks, post-chain stages or
        # decision flags). Any OTHER tool-hooked plugin in the chain
        # (header_injector, an external-process plugin, a user plugin)
        # routes its tools to the exact host chain — correctness by
        # construction, never a silently skipped hook.
        self._modeled = {"deny_filter", "harmful_content_detector", "pii_filter", "regex_filter",
                         "argument_normalizer", "content_moderation", "response_cache_by_prompt",
                         "schema_guard", "toon_encoder", "output_length_guard",
                         "cached_tool_result", "circuit_breaker"}

        self.banks: Dict[str, hip.DeviceScanTables] = {}
        for name, p in (("deny", self.deny), ("harm", self.harm), ("pii", self.pii), ("regex", self.regex)):
            tables = p.scan_tables() if p is not None and hasattr(p, "scan_tables") else None
            if tables is not None:
                self.banks[name] = hip.DeviceScanTables(tables, device)
        # the escape/normalize trigger bank routes rows whose RAW bytes may
        # decode to different text (JSON escapes, non-ASCII) onto the exact
        # host path. It must exist whenever ANY content bank scans raw bytes
        # — not only when the normalizer plugin is enabled — otherwise an
        # escape-hidden deny/harm word (e.g. "forbidden") would slip
        # past the raw-byte scans with no host recheck (parity-fuzz finding)
        if any(p is not None for p in (self.normalizer, self.deny, self.harm,
                                       self.pii, self.regex, self.moderation)):
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
        # the HBM cache only exists when the plugin has an EXPLICIT per-tool
        # allowlist (empty by default, like the reference's cacheable_tools):
        # with no allowlisted tool the cosine sweep would be pure overhead
        # and a hit could never legally substitute a result anyway
        if self.semcache_plugin is not None and self.semcache_plugin.cacheable_tools:
            self.semcache = GpuSemanticCache(
                capacity=max(128, (s.gpu_semcache_capacity // 128) * 128),
                dim=self.feat_dim,
                threshold=self.semcache_plugin.threshold,
                ttl_s=self.semcache_plugin.ttl,
                device=device,
                sketch_dim=s.gpu_semcache_sketch_dim,
            )

        self.max_depth = s.max_json_depth
        self.max_string = s.max_string_length

        # native rewrite lane (rewrite.cpp): fixed-index PII maps for the
        # C matchers (the plugin's active list may be a config subset)
        from ..plugins.builtin import PII_PATTERNS as _PIIP

        self._pii_fixed: List[int] = []
        self._pii_active_mask = 0
        self._pii_names = [n for (n, _, _) in _PIIP]
        if self.pii is not None:
            fixed_index = {n: k for k, (n, _, _) in enumerate(_PIIP)}
            self._pii_fixed = [fixed_index[a[0]] for a in self.pii.active]
            self._pii_active_mask = sum(1 << i for i in self._pii_fixed)
        # deny word table for the native lane's pre-rewrite recheck
        # (lowercased when case-insensitive; non-ASCII words disable the
        # lane for deny-active tools — those rows stay on the Python path)
        self._deny_lane_ok = False
        self._deny_lane_blob = None
        self._deny_lane_off = None
        if self.deny is not None and self.deny.words:
            words = [w.lower() if self.deny.case_insensitive else w for w in self.deny.words]
            if all(w.isascii() for w in words):
                blob_d = b"".join(w.encode() for w in words)
                off_d = np.zeros(len(words) + 1, dtype=np.int32)
                for i, w in enumerate(words):
                    off_d[i + 1] = off_d[i] + len(w.encode())
                self._deny_lane_blob = np.frombuffer(blob_d, dtype=np.uint8).copy() \
                    if blob_d else np.zeros(1, dtype=np.uint8)
                self._deny_lane_off = off_d
                self._deny_lane_ok = True
        # harm phrase table for the native POST lane (_host_post's C path):
        # pre-lowercased, plugin order; non-ASCII phrases disable the lane
        # for harm-active tools (those rows stay on the Python path)
        self._harm_lane_ok = self.harm is None or not getattr(self.harm, "phrases", None)
        self._harm_lane_blob = None
        self._harm_lane_off = None
        if self.harm is not None and self.harm.phrases:
            ph = [p.lower() for p in self.harm.phrases]
            if all(p.isascii() for p in ph):
                blob_h = b"".join(p.encode() for p in ph)
                off_h = np.zeros(len(ph) + 1, dtype=np.int32)
                for i, p in enumerate(ph):
                    off_h[i + 1] = off_h[i] + len(p.encode())
                self._harm_lane_blob = np.frombuffer(blob_h, dtype=np.uint8).copy() \
                    if blob_h else np.zeros(1, dtype=np.uint8)
                self._harm_lane_off = off_h
                self._harm_lane_ok = True
        from ..ops.pybridge import get as _pb_get

        self._pb = _pb_get()  # C response-assembly loops (fails loudly if missing)
        # two batches may be in flight (engine splits big batches): GPU
        # sections (shared pinned arena + stream) serialize on this lock,
        # host/C++ sections interleave. A short GIL switch interval keeps
        # to_thread C calls from stalling behind long Python stretches.
        self._gpu_lock = asyncio.Lock()
        # pinned-host staging arena (bump-allocated per pass; reset after each sync)
        self._pin = torch.empty(16 << 20, dtype=torch.uint8, pin_memory=True)
        self._pin_off = 0

        # post-pass metadata bank: structuredContent presence → toon candidate
        self.banks["postmeta"] = hip.DeviceScanTables(
            dfa.compile_literals(['"structuredContent"'], case_insensitive=False), device)

        # fused multi-bank scans: pass 3's bank set is static; pass 1's
        # includes the per-registry schema bank (rebuilt in _rebuild_tool_meta)
        self._bankset3 = hip.ScanBankSet(
            [(n, self.banks[n]) for n in ("pii", "regex", "harm", "postmeta") if n in self.banks],
            device)
        self._bankset1: Optional[hip.ScanBankSet] = None
        self._graphs = None   # pass-1 GraphCache; rebuilt when _bankset1 changes
        self._graphs3 = None  # pass-3 GraphCache (static _bankset3)

        # native decision-plane stores + string tables (fastpath.cpp)
        self._slot_store = hip.store_new(self.semcache.capacity) if self.semcache is not None else 0
        self._exact_native = hip.cache_new(self.exact_cache.ttl) if self.exact_cache is not None else 0

        def _strtable(items):
            blob = b"".join(x.encode() for x in items)
            off = np.zeros(len(items) + 1, dtype=np.int32)
            for i, x in enumerate(items):
                off[i + 1] = off[i] + len(x.encode())
            return np.frombuffer(blob, dtype=np.uint8).copy() if blob else np.zeros(1, dtype=np.uint8), off

        from ..models.classifier import category_names

        self._deny_tab = _strtable(self.deny.words if self.deny else [])
        self._harm_tab = _strtable(self.harm.cats if self.harm else [])
        self._mod_tab = _strtable(category_names(self.moderation.model.classes) if self.moderation else [])

        # per-tool metadata + schema-shape bank (rebuilt on registry change)
        self._meta_gen = -1
        self._plugins_ver = -1
        self._toolmap = 0
        self._meta_list: List[_ToolMeta] = []
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
        self.post_c = 0       # flagged results fully handled by forge_post_rows
        self.post_c_punt = 0  # rows the C post lane punted back to Python
        self.py_fallback = 0
        self.host_bound = 0   # requests routed to the CPU chain by plugin bindings
        # optional per-stage wall-clock accounting (FORGE_PIPELINE_TIMING=1)
        import os as _os

        self.timing_enabled = bool(_os.environ.get("FORGE_PIPELINE_TIMING"))
        self.timing: Dict[str, float] = {}

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
            elif m.itype == "A2A":
                # A2A fast lane (BASELINE config 4): an in-proc agent whose
                # agent-hook plugins are ALL GPU-covered banks with no
                # conditions/bindings can skip the per-row Python hook chain
                # — the decide pass applied the same deny/harm/moderation
                # blocks over the same text (args dict incl. the message
                # wrapper), and pass 3 + the host post chain cover the
                # response-side pii/regex/harm hooks
                a2a = getattr(self.engine, "a2a_service", None)
                agent = self.engine.registry.find("a2a_agent", m.original_name)
                handler = a2a._local_handlers.get(m.original_name) if a2a else None
                if agent is not None and agent.get("enabled", True) and handler is not None \
                        and str(agent.get("endpoint_url", "")).startswith("inproc://"):
                    from ..plugins.framework import HookType as _HT

                    bank_set = {p.name for p in (self.deny, self.pii, self.regex, self.normalizer,
                                                 self.moderation, self.harm) if p is not None}
                    agent_hooked = {p.name for p in self.engine.plugins.plugins
                                    if p.mode != PluginMode.DISABLED and not p.conditions
                                    and (_HT.AGENT_PRE_INVOKE in p.hooks or _HT.AGENT_POST_INVOKE in p.hooks)}
                    conditioned = any(p.conditions and
                                      (_HT.AGENT_PRE_INVOKE in p.hooks or _HT.AGENT_POST_INVOKE in p.hooks)
                                      for p in self.engine.plugins.plugins if p.mode != PluginMode.DISABLED)
                    if agent_hooked <= bank_set and not conditioned \
                            and not self.engine.plugins.bindings_for_tool(m.original_name):
                        m.a2a_fast = True
                        m.handler = handler
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

        # pass-1 fused bank set (depends on the schema bank just built)
        b1 = [(n, self.banks[n]) for n in ("deny", "harm", "pii", "regex", "normalize")
              if n in self.banks]
        if self._schema_bank is not None:
            b1.append(("schema", self._schema_bank))
        self._bankset1 = hip.ScanBankSet(b1, self.device)
        self._graphs = None  # captured graphs bake the old bank pointers

        # ---- flat arrays for the native decision plane (fastpath.cpp) ----
        metas = list(self._tool_meta.values())
        self._meta_list = metas
        nt = len(metas)
        name_bytes = [m.name.encode() for m in metas]
        blob = b"".join(name_bytes)
        self._t_name_blob = np.frombuffer(blob, dtype=np.uint8).copy() if blob else np.zeros(1, dtype=np.uint8)
        self._t_name_beg = np.zeros(nt, dtype=np.int32)
        self._t_name_end = np.zeros(nt, dtype=np.int32)
        off = 0
        for i, nb in enumerate(name_bytes):
            self._t_name_beg[i] = off
            off += len(nb)
            self._t_name_end[i] = off
        if getattr(self, "_toolmap", 0):
            # defer freeing: a concurrent in-flight batch may still resolve
            # against the old map (freed after the next few rebuilds)
            g = getattr(self, "_toolmap_graveyard", None)
            if g is None:
                g = self._toolmap_graveyard = []
            g.append(self._toolmap)
            while len(g) > 8:
                hip.toolmap_free(g.pop(0))
        self._toolmap = hip.toolmap_new(self._t_name_blob, self._t_name_beg, self._t_name_end) if nt else 0
        self._t_required = np.array([m.required_bits for m in metas], dtype=np.uint32) if nt else np.zeros(1, dtype=np.uint32)
        typed = np.full(max(nt, 1), 0xFFFFFFFFFFFFFFFF, dtype=np.uint64)
        for i, m in enumerate(metas):
            packed = 0
            for k in range(4):
                pk = 0xFFFF if k >= len(m.typed_pairs) else ((m.typed_pairs[k][0] << 8) | m.typed_pairs[k][1])
                packed |= pk << (k * 16)
            typed[i] = packed
        self._t_typed = typed
        self._t_native_kind = np.array([m.native_kind for m in metas], dtype=np.int8) if nt else np.zeros(1, dtype=np.int8)
        self._t_outschema = np.array([m.has_output_schema for m in metas], dtype=bool) if nt else np.zeros(1, dtype=bool)
        self._t_thash = np.array([m.thash for m in metas], dtype=np.int64) if nt else np.zeros(1, dtype=np.int64)
        self._t_index = {m.name: i for i, m in enumerate(metas)}
        self._t_name_list = [m.name for m in metas]
        self._t_tid_list = [m.tid for m in metas]

        # per-tool plugin bindings: mode flips for the bank plugins map onto
        # the flag bits; config overrides or bindings on non-bank plugins
        # force the full CPU chain for that tool (correctness over speed)
        mgr = self.engine.plugins
        bank_names = {p.name for p in (self.deny, self.pii, self.regex, self.normalizer,
                                       self.moderation, self.harm, self.schema_guard) if p is not None}
        from ..plugins.framework import HookType as _HT

        extra_chain = [p for p in mgr.plugins
                       if p.mode != PluginMode.DISABLED and p.name not in self._modeled
                       and (_HT.TOOL_PRE_INVOKE in p.hooks or _HT.TOOL_POST_INVOKE in p.hooks)]
        flags = np.zeros(max(nt, 1), dtype=np.uint32)
        hostbound = np.zeros(max(nt, 1), dtype=bool)
        postlane = np.full(max(nt, 1), -1, dtype=np.int8)
        # rewrite/verdict lane per-tool activation (hoists the per-row
        # _active/_applies python calls out of the hot loops)
        rw_norm = np.zeros(max(nt, 1), dtype=bool)
        rw_regex = np.zeros(max(nt, 1), dtype=bool)
        rw_deny = np.zeros(max(nt, 1), dtype=bool)
        rw_pii = np.zeros(max(nt, 1), dtype=bool)
        rw_harm = np.zeros(max(nt, 1), dtype=bool)
        mod_app = np.zeros(max(nt, 1), dtype=bool)
        for i, m in enumerate(metas):
            bmap = mgr.bindings_for_tool(m.name)
            m.host_chain = any(b.get("config") or (pname not in bank_names)
                               for pname, b in bmap.items())
            # unmodeled tool-hooked plugins (external-process, custom) that
            # apply to this tool force the exact host chain for it
            if not m.host_chain and extra_chain:
                m.host_chain = any(self._applies(p, m.name) for p in extra_chain)
            hostbound[i] = m.host_chain
            f = 0
            if m.reachable:
                f |= hip.TF_REACHABLE
            if self._active(self.deny, m.name, block_class=True):
                f |= hip.TF_DENY
            if self._active(self.pii, m.name, block_class=False):
                f |= hip.TF_PII
            if self._active(self.regex, m.name, block_class=False):
                f |= hip.TF_REGEX
            # TF_NORM = "escape triggers route this tool to the host path":
            # needed when the normalizer applies OR any raw-byte content
            # bank is active (their exact semantics are over DECODED text)
            if self._active(self.normalizer, m.name, block_class=False) or \
               self._active(self.deny, m.name, block_class=True) or \
               self._active(self.harm, m.name, block_class=True) or \
               self._active(self.pii, m.name, block_class=False) or \
               self._active(self.moderation, m.name, block_class=True):
                f |= hip.TF_NORM
            if self._active(self.moderation, m.name, block_class=True):
                f |= hip.TF_MOD
            if self._active(self.harm, m.name, block_class=True):
                f |= hip.TF_HARM
            if self._active(self.schema_guard, m.name, block_class=True):
                if m.schema_mode == "fast":
                    f |= hip.TF_SCHEMA_FAST
                elif m.schema_mode == "host":
                    f |= hip.TF_SCHEMA_HOST
            if self.semcache is not None and self.semcache_plugin.cacheable(m.name):
                f |= hip.TF_CACHE
            if self.exact_cache is not None:
                f |= hip.TF_EXACT
            flags[i] = f
            # native POST lane eligibility (forge_post_rows): -1 = this
            # tool's flagged results must run the exact Python _host_post
            # (user regexes, output schemas); otherwise bit0 pii, bit1
            # harm, bit2 toon — mirroring _host_post's own gates
            pf = 0
            if m.host_chain or m.has_output_schema:
                pf = -1
            if pf >= 0 and self._active(self.regex, m.name, block_class=False):
                # regex rules run in C-lane rows ONLY when the pass-3 regex
                # bank scan proves identity (no hit): rows WITH a bank hit
                # punt (bit3). No bank at all (un-DFA-able rules) → no
                # identity proof → the whole tool stays on the Python path.
                pf = (pf | 8) if "regex" in self.banks else -1
            if pf >= 0 and self._active(self.pii, m.name, block_class=False):
                pf |= 1
            if pf >= 0 and self.harm is not None and self._enforcing(self.harm) and \
               self._applies(self.harm, m.name):
                pf = (pf | 2) if self._harm_lane_ok else -1
            if pf >= 0 and self.toon is not None and self._applies(self.toon, m.name):
                # guard soundness: with positive min_savings the TOON text is
                # strictly shorter than the JSON it replaces, so a row under
                # the guard bound stays under it; otherwise punt to Python
                if self.out_guard is not None and self.toon.min_savings <= 0:
                    pf = -1
                else:
                    pf |= 4
            postlane[i] = pf
            rw_norm[i] = self._active(self.normalizer, m.name, block_class=False)
            rw_regex[i] = self._active(self.regex, m.name, block_class=False)
            rw_deny[i] = self._active(self.deny, m.name, block_class=True)
            rw_pii[i] = self._active(self.pii, m.name, block_class=False)
            rw_harm[i] = bool(self.harm is not None and self._enforcing(self.harm) and
                              self._applies(self.harm, m.name))
            mod_app[i] = bool(self.moderation is not None and self._applies(self.moderation, m.name))
        self._t_flags = flags
        self._t_postlane = postlane
        self._t_rw_norm, self._t_rw_regex, self._t_rw_deny = rw_norm, rw_regex, rw_deny
        self._t_rw_pii, self._t_rw_harm, self._t_mod_app = rw_pii, rw_harm, mod_app
        # FAST-mode schema descriptors for the C rewrite lane (presence +
        # type checks only — richer schemas never compile to "fast"); a C
        # pass means the exact validator is skipped, a C fail reruns it
        # for the byte-exact error message
        TYPE_CODE = {"string": 1, "integer": 2, "boolean": 3, "array": 4,
                     "object": 5, "null": 6}
        skb = bytearray()
        sk_beg: List[int] = []
        sk_end: List[int] = []
        sk_ty: List[int] = []
        sk_rq: List[int] = []
        sk_range = np.full((max(nt, 1), 2), -1, dtype=np.int32)
        for i, m in enumerate(metas):
            if m.schema_mode != "fast":
                continue
            schema = m.tool.get("input_schema") or {}
            props = schema.get("properties") or {}
            required = set(schema.get("required") or [])
            lo = len(sk_beg)
            seen = set()
            for k, sub in props.items():
                ty = sub.get("type") if isinstance(sub, dict) else None
                kb = k.encode()
                sk_beg.append(len(skb))
                skb += kb
                sk_end.append(len(skb))
                sk_ty.append(TYPE_CODE.get(ty, 0))
                sk_rq.append(1 if k in required else 0)
                seen.add(k)
            for k in required:
                if k not in seen:
                    kb = k.encode()
                    sk_beg.append(len(skb))
                    skb += kb
                    sk_end.append(len(skb))
                    sk_ty.append(0)
                    sk_rq.append(1)
            sk_range[i] = (lo, len(sk_beg))
        if sk_beg:
            self._sk_tables = (
                np.frombuffer(bytes(skb), dtype=np.uint8).copy() if skb else np.zeros(1, dtype=np.uint8),
                np.asarray(sk_beg, dtype=np.int32), np.asarray(sk_end, dtype=np.int32),
                np.asarray(sk_ty, dtype=np.int8), np.asarray(sk_rq, dtype=np.uint8))
        else:
            self._sk_tables = None
        self._sk_range = sk_range
        # semcache insert allowlist (tool-level; lookups are gated by TF_CACHE)
        self._t_semallow = np.array(
            [bool(self.semcache is not None and self.semcache_plugin.cacheable(m.name)) for m in metas],
            dtype=bool) if nt else np.zeros(1, dtype=bool)
        self._t_hostbound = hostbound if hostbound.any() else None
        self._meta_gen = self.engine.registry.generation
        self._plugins_ver = getattr(self.engine.plugins, "version", 0)

    def _meta(self) -> Dict[str, _ToolMeta]:
        if self._meta_gen != self.engine.registry.generation or \
           self._plugins_ver != getattr(self.engine.plugins, "version", 0):
            self._rebuild_tool_meta()
        return self._tool_meta

    # ------------------------------------------------------------------
    def _enforcing(self, plugin) -> bool:
        return plugin is not None and plugin.mode in (PluginMode.ENFORCE, PluginMode.ENFORCE_IGNORE_ERROR)

    def _active(self, plugin, tool_name: str, block_class: bool) -> bool:
        """Per-tool effective activation of a GPU bank: a plugin binding may
        flip the mode for this tool (reference: tool_plugin_bindings).
        block_class banks (deny/harm/moderation/schema) only fire in enforce
        modes; rewrite banks (pii/regex/normalize) fire unless disabled."""
        if plugin is None:
            return False
        mode = self.engine.plugins.effective_mode(plugin, tool_name)
        if mode == PluginMode.DISABLED:
            return False
        if block_class and mode == PluginMode.PERMISSIVE:
            return False
        return self._applies(plugin, tool_name)

    def _applies(self, plugin, name: str) -> bool:
        if plugin is None:
            return False
        if not plugin.conditions:
            return True
        from ..plugins.framework import HookType, PluginContext

        return plugin.applies_to(PluginContext(hook=HookType.TOOL_PRE_INVOKE, name=name))

    # ------------------------------------------------------------------
    def _tic(self):
        return time.monotonic() if self.timing_enabled else 0.0

    def _toc(self, key: str, t0: float) -> None:
        if self.timing_enabled:
            self.timing[key] = self.timing.get(key, 0.0) + (time.monotonic() - t0)

    async def process_batch(self, raws: List[bytes], user: Optional[str] = None,
                            server_id: Optional[str] = None,
                            users: Optional[List[Optional[str]]] = None) -> List[Optional[bytes]]:
        self.batches += 1
        n = len(raws)
        self.requests += n
        responses: List[Optional[bytes]] = [None] * n

        def row_user(i: int) -> Optional[str]:
            return users[i] if users is not None else user

        t0 = self._tic()
        joined, offs_raw = self._pb.concat_with_offsets(raws)
        offsets = np.frombuffer(offs_raw, dtype=np.int64)
        blob = np.frombuffer(joined, dtype=np.uint8) if joined else np.zeros(1, dtype=np.uint8)
        env = hip.parse_envelopes(blob, offsets)
        self._toc("pack_envelope", t0)
        kind = env["kind"]

        other_rows = np.nonzero(kind != hip.ENV_TOOLS_CALL)[0]
        if other_rows.size:
            self.py_fallback += int(other_rows.size)
            outs = await asyncio.gather(
                *(self.engine.handle_rpc_bytes(raws[int(i)], user=row_user(int(i)), server_id=server_id)
                  for i in other_rows))
            for i, out in zip(other_rows, outs):
                responses[int(i)] = out

        fast_rows = np.nonzero(kind == hip.ENV_TOOLS_CALL)[0]
        if fast_rows.size:
            await self._fast_toolcalls(raws, blob, env, fast_rows, responses, user, server_id, users)
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
                              user: Optional[str], server_id: Optional[str],
                              users: Optional[List[Optional[str]]] = None) -> None:
        self._meta()  # refresh tool tables on registry/plugin change
        m = rows.shape[0]
        # per-row tenant hash: cache identities (semantic + exact) are
        # (tool, user)-scoped so results never cross users (63-bit so the
        # XOR with tool_hash stays in int64 domain)
        if users is not None:
            memo: Dict[Optional[str], int] = {}
            uh_all = np.empty(len(users), dtype=np.int64)
            for _i, _u in enumerate(users):
                _h = memo.get(_u)
                if _h is None:
                    _h = memo[_u] = tool_hash(_u or "")
                uh_all[_i] = _h
        else:
            uh_all = None
        _uh_scalar = tool_hash(user or "")
        uh = uh_all[rows] if uh_all is not None else np.full(m, _uh_scalar, dtype=np.int64)
        nb = np.ascontiguousarray(env["name_beg"][rows])
        ne = np.ascontiguousarray(env["name_end"][rows])
        id_b = np.ascontiguousarray(env["id_beg"][rows])
        id_e = np.ascontiguousarray(env["id_end"][rows])
        ab, ae = env["args_beg"], env["args_end"]
        args_b = np.ascontiguousarray(np.where(ab[rows] >= 0, ab[rows], 0).astype(np.int32))
        args_e = np.ascontiguousarray(np.where(ab[rows] >= 0, ae[rows], 0).astype(np.int32))

        t_r = self._tic()
        tool_idx = hip.toolmap_resolve(self._toolmap, blob, nb, ne) if self._toolmap \
            else np.full(m, -1, dtype=np.int32)
        self._toc("toolmap", t_r)

        # --- cross-rank routing: rows whose tool is owned by a peer rank
        # ride the RCCL bus as per-destination batches (BASELINE config 3:
        # federation shard fan-out); they overlap the local GPU pass and are
        # spliced into `responses` at the end ---
        fw_state = None
        eng = self.engine
        if eng.foreign_tools and eng.forward_rpc_batch is not None and (tool_idx < 0).any():
            groups: Dict[int, List[int]] = {}
            for j in np.nonzero(tool_idx < 0)[0]:
                j = int(j)
                name = blob[nb[j]:ne[j]].tobytes().decode("utf-8", "replace")
                dest = eng.foreign_tools.get(name)
                if dest is not None and dest != eng.rank:
                    groups.setdefault(dest, []).append(j)
            if groups:
                tasks, fw_rows = [], []
                fw_mask = np.zeros(m, dtype=bool)
                for dest, js in groups.items():
                    rws = [raws[int(rows[j])] for j in js]
                    if users is not None:
                        us = [users[int(rows[j])] for j in js]
                    else:
                        us = [user] * len(js) if user is not None else None
                    tasks.append(eng.forward_rpc_batch(dest, rws, us))
                    fw_rows.append([int(rows[j]) for j in js])
                    fw_mask[js] = True
                fw_state = (asyncio.gather(*tasks), fw_rows)
                keep = ~fw_mask
                rows = rows[keep]
                m = rows.shape[0]
                if m == 0:
                    await self._finish_forward(fw_state, responses)
                    return
                uh = np.ascontiguousarray(uh[keep])
                tool_idx = np.ascontiguousarray(tool_idx[keep])
                nb = np.ascontiguousarray(nb[keep])
                ne = np.ascontiguousarray(ne[keep])
                id_b = np.ascontiguousarray(id_b[keep])
                id_e = np.ascontiguousarray(id_e[keep])
                args_b = np.ascontiguousarray(args_b[keep])
                args_e = np.ascontiguousarray(args_e[keep])

        # tools with semantics-altering plugin bindings run the full CPU
        # chain (the binding-aware invoke_hook path); everything else stays
        # on the GPU fast path
        hostbound = getattr(self, "_t_hostbound", None)
        if hostbound is not None:
            hb_mask = np.zeros(m, dtype=bool)
            valid = tool_idx >= 0
            hb_mask[valid] = hostbound[tool_idx[valid]]
            if hb_mask.any():
                hb_rows = rows[hb_mask]
                self.host_bound += int(hb_rows.size)
                outs = await asyncio.gather(
                    *(self.engine.handle_rpc_bytes(
                        raws[int(i)],
                        user=(users[int(i)] if users is not None else user),
                        server_id=server_id)
                      for i in hb_rows))
                for i, out in zip(hb_rows, outs):
                    responses[int(i)] = out
                keep = ~hb_mask
                rows = rows[keep]
                m = rows.shape[0]
                if m == 0:
                    await self._finish_forward(fw_state, responses)
                    return
                uh = np.ascontiguousarray(uh[keep])
                tool_idx = np.ascontiguousarray(tool_idx[keep])
                nb = np.ascontiguousarray(nb[keep])
                ne = np.ascontiguousarray(ne[keep])
                id_b = np.ascontiguousarray(id_b[keep])
                id_e = np.ascontiguousarray(id_e[keep])
                args_b = np.ascontiguousarray(args_b[keep])
                args_e = np.ascontiguousarray(args_e[keep])

        # --- GPU pass 1 over raw argument spans ---
        t_g = self._tic()
        await self._gpu_lock.acquire()
        self._toc("gp1_lockwait", t_g)
        # hipGraph fast path (gpu/graphs.py): the whole pass — H2D stages,
        # fused scan, featurize, classifier, D2H into pinned — is ONE
        # hipGraphLaunch when the batch fits a bucket and the semcache is
        # off (the cache probe stays eager)
        gbucket = None
        if self.semcache is None:
            if self._graphs is None:
                from .graphs import Pass1Graphs
                self._graphs = Pass1Graphs(self)
            gbucket = self._graphs.get(m, int(blob.nbytes))
        feats = None
        scores_t = None
        sc_np = None
        cache_val_t = cache_idx_t = None
        feats_sk = None
        if gbucket is not None:
            t_st = self._tic()
            gbucket.stage(blob, args_b, args_e, m)
            self._toc("gp1_stage", t_st)
            self._toc("gp1_upload", t_g)
            t_l = self._tic()
            gbucket.replay()
            self._graphs.replays += 1
            self._toc("gp1_launch", t_l)
            t_s = self._tic()
            try:
                await self._await_gpu()
                mm = gbucket.read_multi(m)
                sc_np = gbucket.read_scores(m) if self.classifier is not None else None
            finally:
                self._gpu_lock.release()
            self._toc("gp1_sync", t_s)
        else:
            self._pin_reset()
            data_gpu = self._upload(blob)
            beg_t = self._upload(args_b)
            end_t = self._upload(args_e)
            self._toc("gp1_upload", t_g)
            t_l = self._tic()
            # fused multi-bank scan: all DFA banks in ONE launch (grid.y=bank;
            # fills the chip bank-parallel where per-bank launches ran ~128
            # waves each) writing one [n_banks, m] mask matrix — one D2H later
            out_multi = hip.scan_multi(data_gpu, beg_t, end_t, self._bankset1)
            if self.classifier is not None or self.semcache is not None:
                feats, _ = hip.featurize(data_gpu, beg_t, end_t, self.feat_dim, pad_to=128)
            scores_t = self.classifier.forward(feats)[:m] if self.classifier is not None else None
            if self.semcache is not None:
                bv, bi, feats_sk = self.semcache.lookup(feats)
                cache_val_t, cache_idx_t = bv[:m], bi[:m]
            self._toc("gp1_launch", t_l)
            t_s = self._tic()
            # off-loop sync: a blocking synchronize would stall the event loop
            # and starve the micro-batch collector under live HTTP load
            try:
                await self._await_gpu()
            finally:
                self._gpu_lock.release()
            self._toc("gp1_sync", t_s)
        self._toc("gpu_pass1", t_g)
        t_d = self._tic()

        if gbucket is None:
            mm = out_multi.cpu().numpy().view(np.uint32)  # one D2H for every bank
        if scores_t is not None:
            sc_np = scores_t.cpu().numpy()

        def mask(name):
            i = self._bankset1.index.get(name)
            return mm[i] if i is not None else np.zeros(m, dtype=np.uint32)

        deny_m, harm_m, pii_m = mask("deny"), mask("harm"), mask("pii")
        regex_m, norm_m, schema_m = mask("regex"), mask("normalize"), mask("schema")

        # deny/harm raw-byte hits are VERDICT CANDIDATES, not verdicts:
        # duplicate JSON keys can place the word in a value json.loads
        # discards (last-key-wins), so a raw match is NOT ⊆ decoded match.
        # Candidates route to the host lane, whose pre-rewrite deny recheck
        # and post-rewrite harm rescan run over decoded text — exact CPU
        # semantics. Benign traffic never pays this (no hits, no routing).
        cand = (deny_m != 0) | (harm_m != 0)
        if cand.any():
            norm_m = norm_m.copy()
            norm_m[cand] |= 1
            deny_m = np.where(cand, 0, deny_m).astype(np.uint32)
            harm_m = np.where(cand, 0, harm_m).astype(np.uint32)

        mod_block = np.zeros(m, dtype=np.uint8)
        mod_cat = np.zeros(m, dtype=np.int32)
        mod_score = np.zeros(m, dtype=np.float32)
        if sc_np is not None:
            mod_score = np.ascontiguousarray(sc_np.max(axis=1).astype(np.float32))
            mod_cat = np.ascontiguousarray(sc_np.argmax(axis=1).astype(np.int32))
            mod_block = (mod_score >= self.moderation.threshold).astype(np.uint8)

        # semcache identity = tool_hash XOR user_hash (tenant-scoped)
        th_arr = np.where(tool_idx >= 0, self._t_thash[np.clip(tool_idx, 0, None)] ^ uh, 0)
        hit = np.zeros(m, dtype=np.uint8)
        hit_slot = np.full(m, -1, dtype=np.int32)
        if cache_val_t is not None:
            cache_val = cache_val_t.cpu().numpy()
            hit_slot = np.ascontiguousarray(cache_idx_t.cpu().numpy().astype(np.int32))
            hit = self.semcache.resolve_hits_np(cache_val, hit_slot, th_arr).astype(np.uint8)

        # breaker overlay: flip TF_BREAKER_OPEN for tools with open circuits
        flags = self._t_flags
        if self.breaker is not None and self.breaker.state:
            now0 = time.monotonic()
            open_tis = [self._t_index[nm] for nm, st in self.breaker.state.items()
                        if now0 < st["open_until"] and nm in self._t_index]
            if open_tis:
                flags = flags.copy()
                flags[open_tis] |= hip.TF_BREAKER_OPEN

        t0 = time.monotonic()
        state, nk, reason, arena, rb, re_, n_arena = hip.decide(
            blob, id_b, id_e, args_b, args_e, tool_idx, nb, ne,
            deny_m, harm_m, pii_m, regex_m, norm_m, schema_m,
            mod_block, mod_cat, mod_score, hit, hit_slot, np.ascontiguousarray(uh),
            flags, self._t_required, self._t_typed,
            self._t_name_beg, self._t_name_end, self._t_name_blob, self._t_native_kind,
            self._nest_bits,
            self._deny_tab[0], self._deny_tab[1], self._harm_tab[0], self._harm_tab[1],
            self._mod_tab[0], self._mod_tab[1],
            self._slot_store, self._exact_native, time.monotonic())

        self._toc("decide", t_d)
        t_a = self._tic()
        answered = np.nonzero(state == hip.ST_ANSWERED)[0]
        if answered.size:
            # C-loop scatter (ops/csrc/pybridge.c): no arena copy, no
            # per-row Python slicing
            self._pb.scatter_slices(arena, np.ascontiguousarray(rb[answered]),
                                    np.ascontiguousarray(re_[answered]),
                                    np.ascontiguousarray(rows[answered].astype(np.int64)),
                                    responses)
        rc = np.bincount(reason[answered].astype(np.int64), minlength=9) if answered.size else np.zeros(9, int)
        self.blocked += int(rc[3] + rc[4] + rc[5] + rc[8])
        self.cache_hits += int(rc[6] + rc[7])

        # --- host-schema fallback rows (exact validation) ---
        extra_py: List[Tuple[int, Any]] = []
        for j in np.nonzero(state == hip.ST_HOST_SCHEMA)[0]:
            j = int(j)
            mt = self._meta_list[tool_idx[j]]
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
            elif hit[j]:
                res = self._cache_result_bytes(int(hit_slot[j]))
                if idb is not None and res is not None:
                    responses[r] = self._splice_result(idb, res)
                self.cache_hits += 1
            else:
                # native tools land in the nat2 C++ batch inside dispatch
                extra_py.append((j, args))

        # (measured: running the native batch in an executor thread to
        # overlap with the rewrite pass LOSES ~10% — GIL handoff churn across
        # every host stage — so it stays synchronous inside dispatch)
        native_js = [int(j) for j in np.nonzero(state == hip.ST_DISPATCH_NATIVE)[0]]

        # --- PASS 2: rewrite-flagged subset ---
        rewrite_js = [int(j) for j in np.nonzero(state == hip.ST_REWRITE)[0]]
        rewrite_finish = None
        if rewrite_js:
            self.slow_path += len(rewrite_js)
            t_rw = self._tic()
            rewrite_finish = await self._rewrite_pass(
                blob, env, rows, args_b, args_e, rewrite_js, tool_idx, hit, hit_slot, responses,
                pii_m=pii_m, regex_m=regex_m, norm_m=norm_m, deny_cand=cand)
            self._toc("rewrite_pass", t_rw)

        self._toc("answer_assign", t_a)
        py_items = [(int(j), None) for j in np.nonzero(state == hip.ST_DISPATCH_PY)[0]] + extra_py
        self.fast_path += len(native_js) + len(py_items)
        if native_js or py_items or rewrite_finish is not None:
            await self._dispatch_and_post(blob, env, rows, id_b, id_e, args_b, args_e,
                                          tool_idx, nk, feats, th_arr, native_js, py_items,
                                          responses, t0, feats_sk, uh,
                                          rewrite_finish=rewrite_finish)
        await self._finish_forward(fw_state, responses)

    async def _finish_forward(self, fw_state, responses: List[Optional[bytes]]) -> None:
        """Splice cross-rank (bus-forwarded) responses into the batch output."""
        if fw_state is None:
            return
        gathered, fw_rows = fw_state
        outs_groups = await gathered
        for grp_rows, outs in zip(fw_rows, outs_groups):
            for r, o in zip(grp_rows, outs):
                responses[r] = o if (o is None or isinstance(o, bytes)) else bytes(o)

    async def _a2a_fast_invoke(self, mt: _ToolMeta, args: Any) -> Dict[str, Any]:
        """A2A fast lane: handler call + the exact result shape of
        tool_service A2A dispatch wrapping invoke_agent, minus the per-row
        Python hook chain (GPU banks + pass 3 + host post cover it)."""
        import uuid as _uuid

        message = args.get("message") if isinstance(args, dict) else None
        if not isinstance(message, str):
            message = json.dumps(args or {}, default=str)
        t0 = time.monotonic()
        reply_text = await mt.handler(message, {})
        sc = {"agent": mt.original_name, "response": reply_text,
              "latency_ms": round((time.monotonic() - t0) * 1000, 2),
              "uaid": _uuid.uuid4().hex}
        return {"content": [{"type": "text", "text": reply_text}],
                "structuredContent": sc, "isError": False}

    def _side_stream(self, i: int) -> "torch.cuda.Stream":
        ss = getattr(self, "_streams", None)
        if ss is None:
            ss = self._streams = [torch.cuda.Stream() for _ in range(8)]
        return ss[i % len(ss)]

    def _pin_reset(self) -> None:
        self._pin_off = 0

    async def _await_gpu(self) -> None:
        """Await completion of the current stream WITHOUT
        hipDeviceSynchronize: a device-wide sync holds HIP runtime locks
        that serialize the OTHER in-flight batch's kernel launches
        (measured ~1ms/batch of launch-side stalls). An event record +
        non-blocking hipEventQuery poll yields the event loop instead."""
        ev = torch.cuda.Event()
        ev.record()
        if ev.query():
            return
        for _ in range(200000):
            await asyncio.sleep(0)
            if ev.query():
                return
        await asyncio.to_thread(ev.synchronize)

    def _upload(self, arr: np.ndarray) -> torch.Tensor:
        """Stage a host array through the pinned arena → async H2D. Safe to
        call repeatedly within one pass; reset after the pass's sync."""
        raw = arr.view(np.uint8).reshape(-1)
        n = raw.nbytes
        off = (self._pin_off + 255) & ~255
        if off + n > self._pin.numel():
            grow = max(self._pin.numel() * 2, off + n)
            self._pin = torch.empty(grow, dtype=torch.uint8, pin_memory=True)
            off = 0
        self._pin_off = off + n
        view = self._pin[off:off + n]
        view.numpy()[:] = raw
        # pow2-bucketed device alloc: per-batch byte counts vary by a few
        # KB, and exact-size allocs make the caching allocator split blocks
        # and fall into real hipMalloc on the hot path (measured ~1ms/batch
        # of launch-side stalls) — pow2 buckets recycle perfectly
        cap = 1 << max(16, (n - 1).bit_length()) if n else 1
        dev = torch.empty(cap, dtype=torch.uint8, device=self.device)[:n]
        dev.copy_(view, non_blocking=True)
        tdt = {np.dtype(np.uint8): torch.uint8, np.dtype(np.int32): torch.int32,
               np.dtype(np.int64): torch.int64, np.dtype(np.float32): torch.float32}[np.dtype(arr.dtype)]
        return dev.view(tdt).reshape(arr.shape)

    def _cache_result_bytes(self, slot: int) -> Optional[bytes]:
        if self._slot_store:
            res = hip.store_get(self._slot_store, slot)
            if res:
                return res
        res = self.semcache.results[slot] if self.semcache is not None else None
        if res is None:
            return None
        return res if isinstance(res, bytes) else json.dumps(res, separators=(",", ":")).encode()

    # ------------------------------------------------------------------
    def _apply_rewrites(self, name: str, args: Any, do_norm: bool = True,
                        do_regex: bool = True, do_pii: bool = True,
                        pii_bits: int = -1) -> Tuple[str, Any]:
        """Host rewrites in CPU-chain priority order: normalizer(15) →
        regex(20) → pii(30), using the plugins' own functions. The do_*
        flags come from the per-bank GPU masks: a bank that did not fire on
        this row means its plugin is an identity transform here (the DFA
        banks are conservative supersets of the host matchers), so it can
        be skipped without changing the outcome."""
        if do_norm and self._active(self.normalizer, name, block_class=False):
            args = _walk_strings(args, self.normalizer.norm)
        if do_regex and self._active(self.regex, name, block_class=False):
            args = _walk_strings(args, self.regex.apply_rules)
        if do_pii and self._active(self.pii, name, block_class=False):
            found: List[str] = []

            def fn(s: str) -> str:
                if pii_bits >= 0:
                    masked, f = self.pii.mask_text_subset(s, pii_bits)
                else:
                    masked, f = self.pii.mask_text(s)
                found.extend(f)
                return masked

            new_args = _walk_strings(args, fn)
            if found and self.pii.action == "block" and self._enforcing(self.pii):
                return ("__block__", f"pii_filter: PII detected: {sorted(set(found))}")
            if found and self.pii.action == "mask":
                args = new_args
        return ("__ok__", args)


    async def _rewrite_pass(self, blob, env, rows, args_b, args_e, rewrite_js, tool_idx,
                            hit, hit_slot, responses,
                            pii_m=None, regex_m=None, norm_m=None,
                            deny_cand=None) -> List[Tuple[int, Any]]:
        from ..plugins.builtin import _text_of

        ok_items: List[Tuple[int, Any]] = []
        scan_bytes: Dict[int, bytes] = {}  # native lane: sorted-keys scan form
        harm_c: Dict[int, int] = {}        # row → harm phrase idx (post-rewrite text)

        # --- native lane (rewrite.cpp): rows with no deny candidacy, no
        # user regexes, and provably-equivalent content (the C side punts
        # anything outside its envelope back here) ---
        py_js = rewrite_js
        sg_en = self._enforcing(self.schema_guard)
        schema_c: Dict[int, int] = {}
        t_sub = self._tic()
        if rewrite_js:
            nat_idx: List[int] = []
            flags_l: List[int] = []
            want_l: List[int] = []
            for j in rewrite_js:
                ti = tool_idx[j]
                if regex_m is not None and regex_m[j] and self._t_rw_regex[ti]:
                    continue  # user-configured Python regexes → Python lane
                deny_needed = self._t_rw_deny[ti]
                if deny_needed and not self._deny_lane_ok:
                    continue  # non-ASCII deny words → Python lane recheck
                harm_needed = self._t_rw_harm[ti]
                if harm_needed and not self._harm_lane_ok:
                    continue  # non-ASCII harm phrases → Python lane
                do_norm = (norm_m is None or bool(norm_m[j])) and self._t_rw_norm[ti]
                do_pii = (pii_m is None or bool(pii_m[j])) and self._t_rw_pii[ti]
                fl = (1 if do_norm else 0) | (2 if do_pii else 0) | (4 if deny_needed else 0) | \
                    (8 if harm_needed else 0)
                if sg_en and self._sk_tables is not None and self._sk_range[ti, 0] >= 0:
                    fl |= 16
                bits = int(pii_m[j]) if (pii_m is not None and do_pii) else -1
                if bits < 0:
                    want = self._pii_active_mask
                else:
                    want = 0
                    for k, fixed in enumerate(self._pii_fixed):
                        if (bits >> k) & 1:
                            want |= 1 << fixed
                nat_idx.append(j)
                flags_l.append(fl)
                want_l.append(want)
            self._toc("rw_elig", t_sub)
            t_sub = self._tic()
            if nat_idx:
                pii_mode = 0
                if self.pii is not None and self.pii.action == "block" and self._enforcing(self.pii):
                    pii_mode = 1
                elif self.pii is not None and self.pii.action not in ("mask",):
                    pii_mode = 2
                njs = np.asarray(nat_idx, dtype=np.int64)
                tis = tool_idx[njs]
                t_call = self._tic()
                st, found, deny_hit, harm_hit, schema_ok, rw_arena, rb, re_, sb, se = hip.rewrite_rows(
                    blob, np.ascontiguousarray(args_b[njs]), np.ascontiguousarray(args_e[njs]),
                    np.asarray(flags_l, dtype=np.uint8), np.asarray(want_l, dtype=np.uint32),
                    self._pii_active_mask, pii_mode,
                    bool(self.normalizer and self.normalizer.collapse_ws),
                    bool(self.normalizer and self.normalizer.strip),
                    deny_blob=self._deny_lane_blob, deny_off=self._deny_lane_off,
                    deny_ci=bool(self.deny and self.deny.case_insensitive),
                    harm_blob=self._harm_lane_blob, harm_off=self._harm_lane_off,
                    sk_tables=self._sk_tables,
                    sk_lo=np.ascontiguousarray(self._sk_range[tis, 0]) if self._sk_tables is not None else None,
                    sk_hi=np.ascontiguousarray(self._sk_range[tis, 1]) if self._sk_tables is not None else None)
                self._toc("rw_call", t_call)
                punted: List[int] = []
                done_m = st == hip.RW_DONE
                done_ks = np.nonzero(done_m)[0]
                if done_ks.size:
                    # C-loop slicing for the common path: both byte forms in
                    # two bridge calls instead of 2 python slices per row
                    wires = self._pb.slices_list(rw_arena, np.ascontiguousarray(rb[done_ks]),
                                                 np.ascontiguousarray(re_[done_ks]))
                    scans = self._pb.slices_list(rw_arena, np.ascontiguousarray(sb[done_ks]),
                                                 np.ascontiguousarray(se[done_ks]))
                    for q, k in enumerate(done_ks):
                        j = nat_idx[int(k)]
                        ok_items.append((j, wires[q]))
                        scan_bytes[j] = scans[q]
                        if harm_hit[k] >= 0:
                            harm_c[j] = int(harm_hit[k])
                        if schema_ok[k] != 2:
                            schema_c[j] = int(schema_ok[k])
                for k, j in enumerate(nat_idx):
                    if done_m[k]:
                        pass
                    elif st[k] == hip.RW_DENY:
                        r = int(rows[j])
                        idb = self._id_bytes(blob, env, r)
                        word = self.deny.words[int(deny_hit[k])]
                        responses[r] = self._splice_error(
                            idb, jsonrpc.POLICY_DENIED, f"deny_filter: deny word {word!r} present")
                        self.blocked += 1
                    elif st[k] == hip.RW_BLOCKED:
                        r = int(rows[j])
                        idb = self._id_bytes(blob, env, r)
                        names = sorted(self._pii_names[i] for i in range(len(self._pii_names))
                                       if (int(found[k]) >> i) & 1)
                        responses[r] = self._splice_error(
                            idb, jsonrpc.POLICY_DENIED, f"pii_filter: PII detected: {names}")
                        self.blocked += 1
                    else:
                        punted.append(j)
                nat_set = set(nat_idx)  # hoisted: inside the comprehension this
                # was rebuilt per element — O(n²), 31 ms/batch at 20% flagged
                py_js = [j for j in rewrite_js if j not in nat_set] + punted
                py_js.sort()
        self._toc("rw_c", t_sub)
        t_sub = self._tic()

        for j in py_js:
            mt = self._meta_list[tool_idx[j]]
            r = int(rows[j])
            idb = self._id_bytes(blob, env, r)
            try:
                args = json.loads(blob[args_b[j]:args_e[j]].tobytes() or b"{}")
            except Exception:
                responses[r] = self._splice_error(idb, jsonrpc.INVALID_PARAMS, "invalid arguments")
                continue
            # deny recheck over DECODED text, pre-rewrite (CPU chain order:
            # deny@10 before normalizer@15) — the raw-byte scan misses
            # escape-hidden words (forbidden), which is exactly what
            # routed this row here (parity-fuzz finding)
            if self.deny is not None and self._t_rw_deny[tool_idx[j]]:
                hay = _text_of(args)
                hay = hay.lower() if self.deny.case_insensitive else hay
                word = next((w for w in self.deny.words
                             if (w.lower() if self.deny.case_insensitive else w) in hay), None)
                if word is not None:
                    responses[r] = self._splice_error(
                        idb, jsonrpc.POLICY_DENIED, f"deny_filter: deny word {word!r} present")
                    self.blocked += 1
                    continue
            status, payload = self._apply_rewrites(
                mt.name, args,
                do_norm=bool(norm_m[j]) if norm_m is not None else True,
                do_regex=bool(regex_m[j]) if regex_m is not None else True,
                do_pii=bool(pii_m[j]) if pii_m is not None else True,
                pii_bits=int(pii_m[j]) if pii_m is not None else -1)
            if status == "__block__":
                responses[r] = self._splice_error(idb, jsonrpc.POLICY_DENIED, payload)
                self.blocked += 1
                continue
            # harmful_content@60 over the rewritten text (the CPU plugin's
            # own check; verdict ordering vs moderation happens in
            # _finish_rewrite exactly like the chain)
            if self.harm is not None and self._t_rw_harm[tool_idx[j]]:
                hay2 = _text_of(payload).lower()
                for w, ph in enumerate(self.harm.phrases):
                    if ph.lower() in hay2:
                        harm_c[j] = w
                        break
            ok_items.append((j, payload))
        self._toc("rw_pyloop", t_sub)
        if not ok_items:
            return None
        t_sub = self._tic()

        # scan texts are the sorted-keys form (_text_of): native-lane rows
        # have it precomputed; python-lane entries serialize here
        texts2 = [scan_bytes[j] if j in scan_bytes
                  else json.dumps(a, separators=(",", ":"), sort_keys=True, default=str).encode()
                  for (j, a) in ok_items]
        scores2_t = None
        if self.classifier is not None:
            data2, beg2, end2 = pack_texts(texts2, self.device)
            f2, _ = hip.featurize(data2, beg2, end2, self.feat_dim, pad_to=128)
            scores2_t = self.classifier.forward(f2)[: len(ok_items)]

        self._toc("rw_launch", t_sub)
        # the rescan kernels are IN FLIGHT — the sync happens inside the
        # returned closure, which the caller awaits AFTER the big native
        # upstream batch so the GPU rescan overlaps that C++ work
        async def finish() -> List[Tuple[int, Any]]:
            return await self._finish_rewrite(ok_items, harm_c, schema_c, scores2_t,
                                              blob, env, rows, tool_idx, hit, hit_slot, responses)

        return finish

    async def _finish_rewrite(self, ok_items, harm_c, schema_c, scores2_t,
                              blob, env, rows, tool_idx, hit, hit_slot,
                              responses: List[Optional[bytes]]) -> List[Tuple[int, Any]]:
        """Phase B of the rewrite pass: await the rescan verdicts of the
        rewritten texts (harm scan + moderation classifier) and emit the
        surviving (row, args) dispatch items."""
        t_sub = self._tic()
        if scores2_t is not None:
            await self._await_gpu()
        self._toc("rw_sync", t_sub)
        t_sub = self._tic()
        scores2 = scores2_t.cpu().numpy() if scores2_t is not None else None

        # hoisted gating: vectorized score reduction, per-tool applies from
        # the precomputed arrays, id bytes only for rows that need a splice
        mod_on = scores2 is not None and self._enforcing(self.moderation)
        sc_max = scores2.max(axis=1) if mod_on else None
        sc_arg = scores2.argmax(axis=1) if mod_on else None
        mod_thr = self.moderation.threshold if mod_on else 0.0
        sg_on = self._enforcing(self.schema_guard)
        if mod_on:
            from ..models.classifier import category_names
            cat_names = category_names(scores2.shape[1])
        if sg_on:
            from ..utils.jsonschema import validate as _validate
        mod_app, harm_cats = self._t_mod_app, (self.harm.cats if self.harm else [])

        out: List[Tuple[int, Any]] = []
        for jj, (j, args2) in enumerate(ok_items):
            ti = tool_idx[j]
            if mod_on and mod_app[ti] and float(sc_max[jj]) >= mod_thr:
                r = int(rows[j])
                cat = cat_names[int(sc_arg[jj])]
                responses[r] = self._splice_error(
                    self._id_bytes(blob, env, r), jsonrpc.POLICY_DENIED,
                    f"content_moderation: moderation: category {cat} score {float(sc_max[jj]):.3f}")
                self.blocked += 1
                continue
            hj = harm_c.get(j, -1)
            if hj >= 0:   # gating applied where harm_c was computed
                r = int(rows[j])
                cat = harm_cats[hj] if hj < len(harm_cats) else "?"
                responses[r] = self._splice_error(
                    self._id_bytes(blob, env, r), jsonrpc.POLICY_DENIED,
                    f"harmful_content_detector: harmful content ({cat})")
                self.blocked += 1
                continue
            if sg_on and schema_c.get(j, 2) == 1:
                pass  # C fast-schema check passed on the rewritten tree
            elif sg_on and self._meta_list[ti].schema_mode != "trivial":
                mt = self._meta_list[ti]
                if isinstance(args2, bytes):
                    args2 = json.loads(args2)
                errs = _validate(args2 or {}, mt.tool.get("input_schema") or {})
                if errs:
                    r = int(rows[j])
                    responses[r] = self._splice_error(
                        self._id_bytes(blob, env, r), jsonrpc.POLICY_DENIED,
                        "schema_guard: schema violation: " + "; ".join(errs[:5]))
                    self.blocked += 1
                    continue
            if hit[j]:
                r = int(rows[j])
                res = self._cache_result_bytes(int(hit_slot[j]))
                idb = self._id_bytes(blob, env, r)
                if idb is not None and res is not None:
                    responses[r] = self._splice_result(idb, res)
                self.cache_hits += 1
                continue
            out.append((j, args2))
        self._toc("rw_verdict", t_sub)
        return out

    # ------------------------------------------------------------------
    async def _dispatch_and_post(self, blob, env, rows, id_b, id_e, args_b, args_e,
                                 tool_idx, nk, feats, th_arr,
                                 native_js: List[int], py_items: List[Tuple[int, Any]],
                                 responses: List[Optional[bytes]], t0: float,
                                 feats_sk=None, uh=None, rewrite_finish=None) -> None:
        # --- native upstream batch (C++) ---
        t_u = self._tic()
        nat_blob = np.zeros(0, dtype=np.uint8)
        nat_beg = nat_end = np.zeros(0, dtype=np.int64)
        if native_js:
            njs = np.asarray(native_js, dtype=np.int64)
            kinds = np.ascontiguousarray(nk[njs].astype(np.int32))
            nb_ = np.ascontiguousarray(args_b[njs])
            ne_ = np.ascontiguousarray(args_e[njs])
            nb_ = np.where(ne_ > nb_, nb_, -1).astype(np.int32)
            now_iso = time.strftime("%Y-%m-%dT%H:%M:%SZ", time.gmtime())
            nat_blob, nat_beg, nat_end = hip.upstream_call_batch(blob, nb_, ne_, kinds, now_iso)
            # empty spans = C++ punted a non-canonical args span (escapes,
            # floats, whitespace formatting); answer those rows on the exact
            # Python path so results match the CPU reference byte-for-byte
            punted = np.nonzero((nat_end == nat_beg) & (nb_ >= 0))[0]
            if punted.size:
                pm = set(int(x) for x in punted)
                py_items = py_items + [(native_js[i], None) for i in pm]
                keep = np.ones(len(native_js), dtype=bool)
                keep[list(pm)] = False
                native_js = [j for i, j in enumerate(native_js) if i not in pm]
                nat_beg = np.ascontiguousarray(nat_beg[keep])
                nat_end = np.ascontiguousarray(nat_end[keep])

        # rewrite rescan verdicts: awaited here so the GPU rescan (launched
        # in _rewrite_pass) ran concurrently with the C++ native batch above
        if rewrite_finish is not None:
            t_rw2 = self._tic()
            fin_items = await rewrite_finish()
            py_items = py_items + fin_items
            self.fast_path += len(fin_items)
            self._toc("rewrite_pass", t_rw2)

        # --- python dispatch (non-native upstreams / rewritten args) ---
        py_results: List[Optional[bytes]] = []
        py_errors: Dict[int, Exception] = {}
        if py_items:
            ts = self.engine.tool_service
            from ..services.upstream import InProcUpstream

            async def one(idx: int, j: int, args2: Any) -> None:
                mt = self._meta_list[tool_idx[j]]
                args = args2
                if args is None:
                    try:
                        args = json.loads(blob[args_b[j]:args_e[j]].tobytes() or b"{}")
                    except Exception:
                        py_errors[idx] = jsonrpc.JSONRPCError(jsonrpc.INVALID_PARAMS, "invalid arguments")
                        return
                elif isinstance(args, bytes):  # native rewrite-lane output
                    args = json.loads(args)
                try:
                    if mt.a2a_fast:
                        result = await self._a2a_fast_invoke(mt, args)
                    else:
                        result = await ts.dispatch(mt.tool, args)
                    py_results[idx] = json.dumps(result, separators=(",", ":"), default=str).encode()
                except Exception as exc:
                    py_errors[idx] = exc

            py_results = [None] * len(py_items)
            seq, conc, nat2 = [], [], []
            for idx, (j, args2) in enumerate(py_items):
                mt = self._meta_list[tool_idx[j]]
                if mt.native_kind >= 0 and args2 is not None:
                    # rewritten args for a native tool: batch through the C++
                    # upstream instead of per-request Python dispatch (the
                    # native rewrite lane already produced canonical bytes)
                    nat2.append((idx,
                                 args2 if isinstance(args2, bytes)
                                 else json.dumps(args2, separators=(",", ":"), default=str).encode(),
                                 mt.native_kind))
                elif (mt.itype == "LOCAL" and not (mt.tool.get("annotations") or {}).get("io")) \
                        or mt.a2a_fast \
                        or isinstance(mt.client, InProcUpstream) or mt.native_kind >= 0:
                    # pure in-proc handlers: sequential await beats gather
                    # overhead; IO-backed LOCAL tools (annotations.io, e.g.
                    # gRPC-translated methods) need real concurrency
                    seq.append((idx, j, args2))
                else:
                    conc.append((idx, j, args2))
            if nat2:
                blob2 = b"".join(a for _, a, _ in nat2)
                offs2 = np.zeros(len(nat2) + 1, dtype=np.int32)
                np.cumsum(np.fromiter((len(a) for _, a, _ in nat2), dtype=np.int32,
                                      count=len(nat2)), out=offs2[1:])
                kinds2 = np.fromiter((k for _, _, k in nat2), dtype=np.int32, count=len(nat2))
                now_iso2 = time.strftime("%Y-%m-%dT%H:%M:%SZ", time.gmtime())
                ob2, ob2b, ob2e = hip.upstream_call_batch(
                    np.frombuffer(blob2, dtype=np.uint8) if blob2 else np.zeros(1, dtype=np.uint8),
                    np.ascontiguousarray(offs2[:-1]), np.ascontiguousarray(offs2[1:]),
                    kinds2, now_iso2)
                outs2 = self._pb.slices_list(ob2, ob2b, ob2e)
                for (idx, _a, _k), rbytes in zip(nat2, outs2):
                    if rbytes:
                        py_results[idx] = rbytes
                    else:
                        # canonical-gate punt on the redispatch too → exact path
                        seq.append((idx, py_items[idx][0], py_items[idx][1]))
            for idx, j, args2 in seq:
                await one(idx, j, args2)
            if conc:
                await asyncio.gather(*(one(idx, j, a) for idx, j, a in conc))

        self._toc("dispatch", t_u)
        t_p = self._tic()
        # --- assemble one result blob (native results + python results) ---
        all_js = list(native_js) + [j for j, _ in py_items]
        n_all = len(all_js)
        nat_used = int(nat_end[-1]) if native_js else 0
        py_blob = b"".join(r or b"" for r in py_results)
        res_blob = np.concatenate([nat_blob[:nat_used],
                                   np.frombuffer(py_blob, dtype=np.uint8)]) if py_blob else \
            np.ascontiguousarray(nat_blob[:nat_used])
        if res_blob.size == 0:
            res_blob = np.zeros(1, dtype=np.uint8)
        res_beg = np.zeros(n_all, dtype=np.int64)
        res_end = np.zeros(n_all, dtype=np.int64)
        res_beg[: len(native_js)] = nat_beg
        res_end[: len(native_js)] = nat_end
        off = nat_used
        err_rows = np.zeros(n_all, dtype=np.uint8)
        for idx in range(len(py_items)):
            k = len(native_js) + idx
            if py_results[idx] is not None:
                res_beg[k] = off
                off += len(py_results[idx])
                res_end[k] = off
            else:
                res_beg[k] = res_end[k] = 0
                err_rows[k] = 1

        # --- PASS 3: GPU scans over serialized results ---
        all_js_np = np.asarray(all_js, dtype=np.int64)
        post_flag = np.zeros(n_all, dtype=bool)
        toon_meta = np.zeros(n_all, dtype=bool)
        regex3 = np.zeros(n_all, dtype=bool)
        if n_all:
            await self._gpu_lock.acquire()
            g3 = None
            if self._graphs3 is None:
                from .graphs import Pass3Graphs
                self._graphs3 = Pass3Graphs(self)
            g3 = self._graphs3.get(n_all, int(res_blob.nbytes))
            if g3 is not None:
                g3.stage(res_blob, res_beg.astype(np.int32), res_end.astype(np.int32), n_all)
                g3.replay()
                self._graphs3.replays += 1
                try:
                    await self._await_gpu()
                    mm3 = g3.read_multi(n_all)
                finally:
                    self._gpu_lock.release()
            else:
                self._pin_reset()
                data3 = self._upload(res_blob)
                b3 = self._upload(res_beg.astype(np.int32))
                e3 = self._upload(res_end.astype(np.int32))
                out3 = hip.scan_multi(data3, b3, e3, self._bankset3)  # fused, one launch
                try:
                    await self._await_gpu()
                finally:
                    self._gpu_lock.release()
                mm3 = out3.cpu().numpy().view(np.uint32)
            for b, i in self._bankset3.index.items():
                h = mm3[i] != 0
                if b == "postmeta":
                    toon_meta = h
                else:
                    if b == "regex":
                        regex3 = h
                    post_flag |= h

        res_len = res_end - res_beg
        toon_min = self.toon.min_size if self.toon is not None else 1 << 60
        guard_max = self.out_guard.max_chars if self.out_guard is not None else 1 << 60
        outschema = self._t_outschema[np.clip(tool_idx[all_js_np], 0, None)] if n_all else np.zeros(0, bool)
        needs_host = post_flag | outschema | (toon_meta & (res_len >= toon_min)) | \
            (res_len > guard_max) | err_rows.astype(bool)

        self._toc("post_scan", t_p)
        t_f = self._tic()
        t_fc = self._tic()
        now = time.monotonic()
        if uh is None:
            uh = np.zeros(id_b.shape[0], dtype=np.int64)
        arena2, rb2, re2, is_err, cacheable = hip.finalize(
            blob, id_b, id_e, args_b, args_e, tool_idx, np.ascontiguousarray(uh),
            np.ascontiguousarray(all_js_np.astype(np.int32)), res_blob, res_beg, res_end,
            np.ascontiguousarray(needs_host.astype(np.uint8)),
            self._t_name_beg, self._t_name_end, self._t_name_blob, self._t_flags,
            self._exact_native, now, self.exact_cache.ttl if self.exact_cache else 0.0)
        if n_all:
            self._pb.scatter_slices(arena2, rb2, re2,
                                    np.ascontiguousarray(rows[all_js_np].astype(np.int64)),
                                    responses)
        self._toc("fin_splice", t_fc)
        t_fh = self._tic()

        # --- host post chain for flagged rows ---
        host_ks = np.nonzero(needs_host)[0]
        host_cached: List[Tuple[int, bytes]] = []  # (k, result bytes) for cache insert

        # native POST lane (rewrite.cpp forge_post_rows): pii → harm → toon
        # → serialize in C for eligible rows; punts fall through to the
        # exact Python _host_post below
        c_handled: set = set()
        if host_ks.size:
            pl = self._t_postlane
            nat_ks = [int(k) for k in host_ks
                      if not err_rows[k]
                      and pl[tool_idx[all_js[int(k)]]] >= 0
                      and not (pl[tool_idx[all_js[int(k)]]] & 8 and regex3[int(k)])
                      and res_len[int(k)] <= guard_max]
            if nat_ks:
                pii_mode = 0
                if self.pii is not None and self.pii.action == "block" and self._enforcing(self.pii):
                    pii_mode = 1
                elif self.pii is not None and self.pii.action not in ("mask",):
                    pii_mode = 2
                nk_np = np.asarray(nat_ks, dtype=np.int64)
                st_p, found_p, harm_p, iserr_p, p_arena, pb, pe = hip.post_rows(
                    res_blob,
                    np.ascontiguousarray(res_beg[nk_np].astype(np.int64)),
                    np.ascontiguousarray(res_end[nk_np].astype(np.int64)),
                    np.ascontiguousarray((self._t_postlane[tool_idx[all_js_np[nk_np]]] & 7).astype(np.uint8)),
                    self._pii_active_mask, pii_mode,
                    harm_blob=self._harm_lane_blob, harm_off=self._harm_lane_off,
                    toon_min_size=self.toon.min_size if self.toon is not None else (1 << 60),
                    toon_min_savings=self.toon.min_savings if self.toon is not None else 1.0)
                for q, k in enumerate(nat_ks):
                    j = all_js[k]
                    r = int(rows[j])
                    idb = self._id_bytes(blob, env, r)
                    if st_p[q] == hip.RW_DONE:
                        self.post_rewrites += 1
                        rbytes = p_arena[pb[q]:pe[q]].tobytes()
                        if idb is not None:
                            responses[r] = self._splice_result(idb, rbytes)
                        if iserr_p[q]:
                            is_err[k] = 1
                        else:
                            is_err[k] = 0
                            cacheable[k] = 1
                            host_cached.append((k, rbytes))
                        c_handled.add(k)
                    elif st_p[q] == hip.RW_BLOCKED:
                        self.post_rewrites += 1
                        names = sorted(self._pii_names[i2] for i2 in range(len(self._pii_names))
                                       if (int(found_p[q]) >> i2) & 1)
                        responses[r] = self._splice_error(
                            idb, jsonrpc.POLICY_DENIED, f"pii_filter: PII detected: {names}")
                        self.blocked += 1
                        is_err[k] = 1
                        c_handled.add(k)
                    elif st_p[q] == hip.RW_DENY:
                        self.post_rewrites += 1
                        cat = self.harm.cats[int(harm_p[q])]
                        responses[r] = self._splice_error(
                            idb, jsonrpc.POLICY_DENIED,
                            f"harmful_content_detector: harmful content ({cat})")
                        self.blocked += 1
                        is_err[k] = 1
                        c_handled.add(k)
                    # PUNT/BADJSON → python loop below
                self.post_c += len(c_handled)
                self.post_c_punt += len(nat_ks) - len(c_handled)

        for k in host_ks:
            k = int(k)
            if k in c_handled:
                continue
            j = all_js[k]
            mt = self._meta_list[tool_idx[j]]
            r = int(rows[j])
            idb = self._id_bytes(blob, env, r)
            if err_rows[k]:
                exc = py_errors.get(k - len(native_js))
                code = getattr(exc, "code", jsonrpc.SERVER_ERROR)
                responses[r] = self._splice_error(idb, code if isinstance(code, int) else jsonrpc.SERVER_ERROR,
                                                  str(exc))
                is_err[k] = 1
                continue
            rbytes = res_blob[res_beg[k]:res_end[k]].tobytes()
            rbytes, host_err, blocked_msg = self._host_post(mt, rbytes)
            if blocked_msg:
                responses[r] = self._splice_error(idb, jsonrpc.POLICY_DENIED, blocked_msg)
                self.blocked += 1
                is_err[k] = 1
                continue
            is_err[k] = 1 if host_err else 0
            if idb is not None:
                responses[r] = self._splice_result(idb, rbytes)
            if not host_err:
                cacheable[k] = 1
                host_cached.append((k, rbytes))

        self._toc("fin_hostpost", t_fh)
        t_fs = self._tic()
        # --- bookkeeping: semcache insert, breaker, metrics ---
        if self.semcache is not None and n_all:
            # only allowlisted tools are inserted (cacheable_tools gate);
            # that also gates lookups implicitly — a non-allowlisted tool's
            # (tool,user) identity never exists in the cache
            ins = np.nonzero((cacheable == 1) &
                             self._t_semallow[np.clip(tool_idx[all_js_np], 0, None)])[0]
            if ins.size:
                ins_js = all_js_np[ins]
                slots = self.semcache.assign_slots(int(ins.size))
                self.semcache.insert_features(feats, ins_js, slots, th_arr[ins_js], sketch=feats_sk)
                if self._slot_store:
                    hip.store_put_batch(self._slot_store, slots, res_blob,
                                        np.ascontiguousarray(res_beg[ins]),
                                        np.ascontiguousarray(res_end[ins]))
                    for (k, rbytes) in host_cached:  # host-modified results overwrite
                        pos = np.nonzero(ins == k)[0]
                        if pos.size:
                            hip.store_put(self._slot_store, int(slots[pos[0]]), rbytes)

        self._toc("fin_seminsert", t_fs)
        t_fm = self._tic()
        if n_all:
            ms = (time.monotonic() - t0) * 1000.0
            tis = tool_idx[all_js_np]
            nt = len(self._meta_list)
            counts_all = np.bincount(tis, minlength=nt)
            errs_all = np.bincount(tis, weights=is_err.astype(np.float64),
                                   minlength=nt).astype(np.int64)
            uniq = np.nonzero(counts_all)[0]
            counts = counts_all[uniq]
            errs = errs_all[uniq]
            if self.breaker is not None:
                # breaker bookkeeping only where it can change state: tools with
                # errors this batch, or tools that already have a window/state
                for k, ti in enumerate(uniq):
                    name = self._t_name_list[int(ti)]
                    if errs[k] or name in self.breaker.state:
                        self._breaker_record_bulk(name, int(counts[k]), int(errs[k]))
            self.engine.metrics.record_aggregate_many(
                [self._t_tid_list[int(ti)] for ti in uniq], counts, errs, ms)
        self._toc("fin_metrics", t_fm)
        self._toc("finalize", t_f)

    def _breaker_record_bulk(self, name: str, n: int, n_err: int) -> None:
        b = self.breaker
        st = b._st(name)
        st["results"].extend([True] * n_err + [False] * (n - n_err))
        if len(st["results"]) > b.window:
            st["results"] = st["results"][-b.window:]
        if len(st["results"]) >= b.window and (sum(st["results"]) / len(st["results"])) >= b.error_threshold:
            st["open_until"] = time.monotonic() + b.cooldown
            st["results"] = []

    def _host_post(self, mt: _ToolMeta, rb: bytes) -> Tuple[bytes, bool, Optional[str]]:
        """Exact host post chain for flagged results: regex(20) → pii(30) →
        output-schema → harm(60) → toon(900) → guard(950)."""
        self.post_rewrites += 1
        try:
            result = json.loads(rb)
        except Exception:
            return rb, True, None
        name = mt.name
        if self._active(self.regex, name, block_class=False):
            result = _walk_strings(result, self.regex.apply_rules)
        if self._active(self.pii, name, block_class=False):
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
            "timing_s": {k: round(v, 4) for k, v in self.timing.items()} if self.timing_enabled else None,
            "batches": self.batches, "requests": self.requests, "fast_path": self.fast_path,
            "slow_path": self.slow_path, "blocked": self.blocked, "cache_hits": self.cache_hits,
            "host_bound": self.host_bound,
            "post_rewrites": self.post_rewrites, "post_c": self.post_c,
            "post_c_punt": self.post_c_punt, "py_fallback": self.py_fallback,
            "banks": {k: {"states": v.n_states, "classes": v.n_classes} for k, v in self.banks.items()},
        }
        if self.semcache is not None:
            out["semcache"] = self.semcache.stats()
        return out
