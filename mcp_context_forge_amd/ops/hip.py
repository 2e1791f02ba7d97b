"""ctypes bindings for the in-tree HIP kernel library (libforge_hip.so).

All entry points take torch tensors already resident on the device and
launch on torch's *current* HIP stream — no extra synchronization, composes
with torch ops and HIP graphs. On a GPU box the library is REQUIRED: any
failure raises, never a silent eager fallback (per the build contract).
"""

from __future__ import annotations

import ctypes
from functools import lru_cache
from typing import List, Optional, Tuple

import numpy as np
import torch

from .dfa import ScanTables


class ForgeHipError(RuntimeError):
    pass


@lru_cache(maxsize=1)
def _load() -> ctypes.CDLL:
    from .build import LIB, build

    if not LIB.exists():
        build()
    lib = ctypes.CDLL(str(LIB))
    protos = {
        "forge_scan": [ctypes.c_void_p] * 3 + [ctypes.c_int] + [ctypes.c_void_p] * 3 + [ctypes.c_int] * 2 + [ctypes.c_void_p] * 3,
        "forge_scan_multi": [ctypes.c_void_p] * 3 + [ctypes.c_int, ctypes.c_void_p,
                                                    ctypes.c_int, ctypes.c_int] + [ctypes.c_void_p] * 2,
        "forge_featurize": [ctypes.c_void_p] * 3 + [ctypes.c_int, ctypes.c_int] + [ctypes.c_void_p] * 3,
        "forge_json_guard": [ctypes.c_void_p] * 3 + [ctypes.c_int] * 3 + [ctypes.c_void_p] * 3,
        "forge_gemm_bt": [ctypes.c_void_p] * 4 + [ctypes.c_int] * 5 + [ctypes.c_void_p],
        "forge_gemm_bt_v2": [ctypes.c_void_p] * 4 + [ctypes.c_int] * 5 + [ctypes.c_void_p],
        "forge_gemv_head": [ctypes.c_void_p] * 4 + [ctypes.c_int] * 4 + [ctypes.c_void_p],
        "forge_rows_argmax_merge": [ctypes.c_void_p, ctypes.c_int, ctypes.c_int, ctypes.c_int, ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p],
        "forge_rows_scatter_bf16": [ctypes.c_void_p] * 3 + [ctypes.c_int] * 2 + [ctypes.c_void_p],
        "forge_rows_gather_scatter_bf16": [ctypes.c_void_p] * 5 + [ctypes.c_int] * 2 + [ctypes.c_void_p],
        "forge_synchronize": [ctypes.c_void_p],
        "forge_parse_envelopes": [ctypes.c_void_p, ctypes.c_void_p, ctypes.c_int] + [ctypes.c_void_p] * 7,
        "forge_upstream_call_batch": [ctypes.c_void_p] * 4 + [ctypes.c_int, ctypes.c_char_p, ctypes.c_void_p, ctypes.c_int64, ctypes.c_void_p, ctypes.c_void_p],
        "forge_rewrite_rows": [ctypes.c_void_p] * 3 + [ctypes.c_int] + [ctypes.c_void_p] * 2 +
                              [ctypes.c_uint32] + [ctypes.c_int] * 3 +
                              [ctypes.c_void_p, ctypes.c_void_p, ctypes.c_int, ctypes.c_int] +
                              [ctypes.c_void_p] * 3 +
                              [ctypes.c_void_p, ctypes.c_int64] + [ctypes.c_void_p] * 4 +
                              [ctypes.c_void_p, ctypes.c_void_p, ctypes.c_int, ctypes.c_void_p] +
                              [ctypes.c_void_p] * 8,
        "forge_post_rows": [ctypes.c_void_p] * 3 + [ctypes.c_int, ctypes.c_void_p] +
                           [ctypes.c_uint32, ctypes.c_int] +
                           [ctypes.c_void_p, ctypes.c_void_p, ctypes.c_int] +
                           [ctypes.c_int64, ctypes.c_double] +
                           [ctypes.c_void_p] * 4 +
                           [ctypes.c_void_p, ctypes.c_int64] + [ctypes.c_void_p] * 2,
    }
    for name, argtypes in protos.items():
        fn = getattr(lib, name)
        fn.argtypes = argtypes
        fn.restype = ctypes.c_int64 if name in ("forge_upstream_call_batch",
                                                "forge_rewrite_rows",
                                                "forge_post_rows") else ctypes.c_int
    return lib


def available() -> bool:
    try:
        return torch.cuda.is_available() and _load() is not None
    except Exception:
        return False


def _stream() -> ctypes.c_void_p:
    return ctypes.c_void_p(torch.cuda.current_stream().cuda_stream)


def _check(name: str, rc: int) -> None:
    if rc != 0:
        raise ForgeHipError(f"{name} failed with code {rc}")


def _ptr(t: Optional[torch.Tensor]) -> ctypes.c_void_p:
    return ctypes.c_void_p(t.data_ptr()) if t is not None else ctypes.c_void_p(0)


class DeviceScanTables:
    """ScanTables uploaded to device memory."""

    def __init__(self, tables: ScanTables, device: str = "cuda"):
        self.meta = tables
        # torch lacks uint16/uint32 storage in places — bit-identical views as int16/int32
        self.next = torch.from_numpy(tables.next.view(np.int16).copy()).to(device)
        self.klass = torch.from_numpy(tables.klass.copy()).to(device)
        self.accept = torch.from_numpy(tables.accept.view(np.int32).copy()).to(device)
        self.n_states = tables.n_states
        self.n_classes = tables.n_classes


def scan(data: torch.Tensor, beg: torch.Tensor, end: torch.Tensor, tables: DeviceScanTables,
         first_end: bool = False) -> Tuple[torch.Tensor, Optional[torch.Tensor]]:
    """data uint8 [total], beg/end int32 [B] spans on device → mask int32 [B]."""
    batch = beg.numel()
    out_mask = torch.zeros(batch, dtype=torch.int32, device=data.device)
    out_end = torch.full((batch,), -1, dtype=torch.int32, device=data.device) if first_end else None
    _check("forge_scan", _load().forge_scan(
        _ptr(data), _ptr(beg), _ptr(end), batch,
        _ptr(tables.next), _ptr(tables.klass), _ptr(tables.accept),
        tables.n_states, tables.n_classes,
        _ptr(out_mask), _ptr(out_end), _stream()))
    return out_mask, out_end


class ScanBankSet:
    """Device descriptor table for the fused multi-bank scan
    (scan_multi_kernel): all banks in ONE launch, one [n_banks, B] output.
    Built once per bank combination; holds references to the bank tensors
    so their device memory outlives the descriptor pointers."""

    LDS_LIMIT = 64 * 1024

    def __init__(self, banks: List[Tuple[str, "DeviceScanTables"]], device: str = "cuda"):
        self.names = [n for n, _ in banks]
        self.banks = [b for _, b in banks]
        self.n = len(self.banks)
        self.index = {n: i for i, n in enumerate(self.names)}
        self.lds_bytes = 0
        recs = np.zeros((max(self.n, 1), 6), dtype=np.int64)
        for i, b in enumerate(self.banks):
            tbytes = b.n_states * b.n_classes * 2 + 256 + b.n_states * 4
            use_lds = 1 if tbytes <= self.LDS_LIMIT else 0
            if use_lds:
                self.lds_bytes = max(self.lds_bytes, tbytes)
            recs[i, 0] = b.next.data_ptr()
            recs[i, 1] = b.klass.data_ptr()
            recs[i, 2] = b.accept.data_ptr()
            recs[i, 3] = (b.n_states & 0xFFFFFFFF) | (b.n_classes << 32)
            recs[i, 4] = use_lds
        self.descs = torch.from_numpy(recs).to(device)


def scan_multi(data: torch.Tensor, beg: torch.Tensor, end: torch.Tensor,
               bankset: ScanBankSet, out: Optional[torch.Tensor] = None) -> torch.Tensor:
    """→ int32 [n_banks, B] mask matrix (bit-identical to per-bank scan).
    `out` (preallocated [n_banks, B]) makes the call graph-capture safe."""
    batch = beg.numel()
    if out is None:
        out = torch.zeros((bankset.n, batch), dtype=torch.int32, device=data.device)
    else:
        out.zero_()
    if bankset.n == 0 or batch == 0:
        return out
    _check("forge_scan_multi", _load().forge_scan_multi(
        _ptr(data), _ptr(beg), _ptr(end), batch,
        _ptr(bankset.descs), bankset.n, bankset.lds_bytes,
        _ptr(out), _stream()))
    return out


def featurize(data: torch.Tensor, beg: torch.Tensor, end: torch.Tensor, dim: int,
              want_f32: bool = False, pad_to: int = 0,
              out: Optional[torch.Tensor] = None) -> Tuple[torch.Tensor, Optional[torch.Tensor]]:
    """pad_to > 0: allocate (and zero) dim0 rounded up to that multiple and
    return the FULL padded tensor (kernel fills the first `batch` rows) —
    stable alloc shapes for the caching allocator and MFMA-ready M.
    `out` (preallocated bf16 [batch, dim]) makes the call capture-safe."""
    batch = beg.numel()
    if out is not None:
        _check("forge_featurize", _load().forge_featurize(
            _ptr(data), _ptr(beg), _ptr(end), batch, dim, _ptr(out), _ptr(None), _stream()))
        return out, None
    rows = ((batch + pad_to - 1) // pad_to) * pad_to if pad_to else batch
    out_bf16 = torch.empty((rows, dim), dtype=torch.bfloat16, device=data.device)
    out_f32 = torch.empty((rows, dim), dtype=torch.float32, device=data.device) if want_f32 else None
    if rows > batch:
        out_bf16[batch:].zero_()
        if out_f32 is not None:
            out_f32[batch:].zero_()
    _check("forge_featurize", _load().forge_featurize(
        _ptr(data), _ptr(beg), _ptr(end), batch, dim, _ptr(out_bf16), _ptr(out_f32), _stream()))
    return out_bf16, out_f32


def json_guard(data: torch.Tensor, beg: torch.Tensor, end: torch.Tensor, max_depth: int = 64,
               max_string: int = 1 << 20) -> Tuple[torch.Tensor, torch.Tensor]:
    batch = beg.numel()
    status = torch.zeros(batch, dtype=torch.int32, device=data.device)
    depth = torch.zeros(batch, dtype=torch.int32, device=data.device)
    _check("forge_json_guard", _load().forge_json_guard(
        _ptr(data), _ptr(beg), _ptr(end), batch, max_depth, max_string, _ptr(status), _ptr(depth), _stream()))
    return status, depth


ACT_NONE, ACT_GELU, ACT_SIGMOID = 0, 1, 2


def gemm_bt(a: torch.Tensor, bt: torch.Tensor, bias: Optional[torch.Tensor] = None,
            act: int = ACT_NONE, out_bf16: bool = False,
            out: Optional[torch.Tensor] = None) -> torch.Tensor:
    """C[M,N] = act(A[M,K] @ BT[N,K]^T + bias). M%128==0, N%128==0, K%64==0."""
    assert a.dtype == torch.bfloat16 and bt.dtype == torch.bfloat16
    assert a.is_contiguous() and bt.is_contiguous()
    m, k = a.shape
    n, k2 = bt.shape
    assert k == k2, (a.shape, bt.shape)
    if out is None:
        out = torch.empty((m, n), dtype=torch.bfloat16 if out_bf16 else torch.float32, device=a.device)
    # v2 (256² tile, 4-phase ring, counted vmcnt) when shapes allow; v1 fallback
    import os

    if m % 256 == 0 and n % 256 == 0 and k % 32 == 0 and not os.environ.get("FORGE_GEMM_V1"):
        rc = _load().forge_gemm_bt_v2(_ptr(a), _ptr(bt), _ptr(bias), _ptr(out), m, n, k,
                                      act, 1 if out_bf16 else 0, _stream())
        _check("forge_gemm_bt_v2", rc)
        return out
    rc = _load().forge_gemm_bt(_ptr(a), _ptr(bt), _ptr(bias), _ptr(out), m, n, k,
                               act, 1 if out_bf16 else 0, _stream())
    _check("forge_gemm_bt", rc)
    return out


def gemv_head(a: torch.Tensor, wt: torch.Tensor, bias: Optional[torch.Tensor] = None,
              act: int = ACT_NONE, out: Optional[torch.Tensor] = None) -> torch.Tensor:
    m, k = a.shape
    c, k2 = wt.shape
    assert k == k2 and c <= 32
    if out is None:
        out = torch.empty((m, c), dtype=torch.float32, device=a.device)
    _check("forge_gemv_head", _load().forge_gemv_head(_ptr(a), _ptr(wt), _ptr(bias), _ptr(out), m, c, k, act, _stream()))
    return out


def rows_argmax_merge(scores: torch.Tensor, best_val: torch.Tensor, best_idx: torch.Tensor,
                      idx_base: int = 0, valid: Optional[torch.Tensor] = None) -> None:
    m, nc = scores.shape
    _check("forge_rows_argmax_merge", _load().forge_rows_argmax_merge(
        _ptr(scores), m, nc, idx_base, _ptr(valid), _ptr(best_val), _ptr(best_idx), _stream()))


def rows_gather_scatter_bf16(src: torch.Tensor, src_rows: torch.Tensor, dst_slots: torch.Tensor,
                             dst: torch.Tensor, valid: Optional[torch.Tensor] = None) -> None:
    """dst[dst_slots[i]] = src[src_rows[i]]; valid[dst_slots[i]] = 1."""
    r = src_rows.numel()
    d = src.shape[1]
    _check("forge_rows_gather_scatter_bf16", _load().forge_rows_gather_scatter_bf16(
        _ptr(src), _ptr(src_rows), _ptr(dst_slots), _ptr(dst), _ptr(valid), r, d, _stream()))


def verify_dot(feats: torch.Tensor, keys: torch.Tensor, idx: torch.Tensor) -> torch.Tensor:
    """Exact bf16 dot of feats[r] with keys[idx[r]] (-1 → -1e30). fp32 [M]."""
    m, d = feats.shape
    out = torch.empty(m, dtype=torch.float32, device=feats.device)
    _check("forge_verify_dot", _load().forge_verify_dot(
        _ptr(feats), _ptr(keys), _ptr(idx), _ptr(out), m, d, _stream()))
    return out


def rows_scatter_bf16(src: torch.Tensor, slots: torch.Tensor, dst: torch.Tensor) -> None:
    r, d = src.shape
    _check("forge_rows_scatter_bf16", _load().forge_rows_scatter_bf16(_ptr(src), _ptr(slots), _ptr(dst), r, d, _stream()))


# ---------------------------------------------------------------------------
# Host-side native fast path (CPU functions in the same library)
# ---------------------------------------------------------------------------

ENV_NEEDS_PY, ENV_PARSE_ERR, ENV_OTHER, ENV_TOOLS_CALL = -2, -1, 0, 1


def parse_envelopes(blob: np.ndarray, offsets: np.ndarray) -> dict:
    """C++ JSON-RPC envelope scan over a packed request blob (host memory).

    blob: uint8 [total]; offsets: int64 [n+1]. Returns numpy arrays:
    kind, id_beg/id_end, name_beg/name_end, args_beg/args_end (int32 [n]).
    """
    n = offsets.shape[0] - 1
    kind = np.empty(n, dtype=np.int32)
    id_b = np.empty(n, dtype=np.int32)
    id_e = np.empty(n, dtype=np.int32)
    nm_b = np.empty(n, dtype=np.int32)
    nm_e = np.empty(n, dtype=np.int32)
    ar_b = np.empty(n, dtype=np.int32)
    ar_e = np.empty(n, dtype=np.int32)

    def p(a):
        return ctypes.c_void_p(a.ctypes.data)

    _check("forge_parse_envelopes", _load().forge_parse_envelopes(
        p(blob), p(offsets), n, p(kind), p(id_b), p(id_e), p(nm_b), p(nm_e), p(ar_b), p(ar_e)))
    return {"kind": kind, "id_beg": id_b, "id_end": id_e,
            "name_beg": nm_b, "name_end": nm_e, "args_beg": ar_b, "args_end": ar_e}


def upstream_call_batch(blob: np.ndarray, args_beg: np.ndarray, args_end: np.ndarray,
                        kinds: np.ndarray, now_iso: str) -> Tuple[np.ndarray, np.ndarray, np.ndarray]:
    """Native in-proc upstream batch call. Returns (out_blob u8, res_beg i64, res_end i64)."""
    n = kinds.shape[0]
    res_b = np.empty(n, dtype=np.int64)
    res_e = np.empty(n, dtype=np.int64)

    def p(a):
        return ctypes.c_void_p(a.ctypes.data)

    lib = _load()
    spans_total = int((args_end[kinds >= 0] - np.maximum(args_beg[kinds >= 0], 0)).clip(min=0).sum()) if n else 0
    cap = spans_total * 3 + n * 256 + 1024
    out = np.empty(cap, dtype=np.uint8)
    need = lib.forge_upstream_call_batch(p(blob), p(args_beg), p(args_end), p(kinds), n,
                                         now_iso.encode(), p(out), cap, p(res_b), p(res_e))
    if need > cap:  # rare: resize and redo
        out = np.empty(int(need) + 1024, dtype=np.uint8)
        lib.forge_upstream_call_batch(p(blob), p(args_beg), p(args_end), p(kinds), n,
                                      now_iso.encode(), p(out), out.shape[0], p(res_b), p(res_e))
    return out, res_b, res_e


# ---------------------------------------------------------------------------
# Native decision plane (fastpath.cpp)
# ---------------------------------------------------------------------------

def _np_ptr(a: Optional[np.ndarray]) -> ctypes.c_void_p:
    return ctypes.c_void_p(a.ctypes.data) if a is not None else ctypes.c_void_p(0)


@lru_cache(maxsize=1)
def _fastpath_protos() -> ctypes.CDLL:
    lib = _load()
    lib.forge_store_new.argtypes = [ctypes.c_int]
    lib.forge_store_new.restype = ctypes.c_void_p
    lib.forge_store_put.argtypes = [ctypes.c_void_p, ctypes.c_int, ctypes.c_void_p, ctypes.c_int64]
    lib.forge_store_put.restype = None
    lib.forge_store_put_batch.argtypes = [ctypes.c_void_p, ctypes.c_void_p, ctypes.c_int,
                                          ctypes.c_void_p, ctypes.c_void_p, ctypes.c_void_p]
    lib.forge_store_put_batch.restype = None
    lib.forge_store_free.argtypes = [ctypes.c_void_p]
    lib.forge_store_free.restype = None
    lib.forge_cache_new.argtypes = [ctypes.c_double]
    lib.forge_cache_new.restype = ctypes.c_void_p
    lib.forge_cache_free.argtypes = [ctypes.c_void_p]
    lib.forge_cache_free.restype = None
    lib.forge_toolmap_new.argtypes = [ctypes.c_void_p] * 3 + [ctypes.c_int]
    lib.forge_toolmap_new.restype = ctypes.c_void_p
    lib.forge_toolmap_free.argtypes = [ctypes.c_void_p]
    lib.forge_toolmap_free.restype = None
    lib.forge_toolmap_resolve.argtypes = [ctypes.c_void_p] * 4 + [ctypes.c_int, ctypes.c_void_p]
    lib.forge_toolmap_resolve.restype = None
    lib.forge_decide.argtypes = [ctypes.c_void_p] * 20 + [ctypes.c_int] + [ctypes.c_void_p] * 7 + \
        [ctypes.c_uint32] + \
        [ctypes.c_void_p, ctypes.c_void_p, ctypes.c_int] * 3 + \
        [ctypes.c_void_p, ctypes.c_void_p, ctypes.c_double] + \
        [ctypes.c_void_p] * 4 + [ctypes.c_int64] + [ctypes.c_void_p] * 2
    lib.forge_store_get.argtypes = [ctypes.c_void_p, ctypes.c_int, ctypes.c_void_p, ctypes.c_int64]
    lib.forge_store_get.restype = ctypes.c_int64
    lib.forge_decide.restype = ctypes.c_int64
    lib.forge_finalize.argtypes = [ctypes.c_void_p] * 7 + [ctypes.c_int] + [ctypes.c_void_p, ctypes.c_int] + \
        [ctypes.c_void_p] * 4 + [ctypes.c_void_p] * 4 + \
        [ctypes.c_void_p, ctypes.c_double, ctypes.c_double] + \
        [ctypes.c_void_p, ctypes.c_int64] + [ctypes.c_void_p] * 4
    lib.forge_finalize.restype = ctypes.c_int64
    return lib


def store_new(capacity: int) -> int:
    return _fastpath_protos().forge_store_new(capacity)


def store_put(store: int, slot: int, data: bytes) -> None:
    _fastpath_protos().forge_store_put(ctypes.c_void_p(store), slot, data, len(data))


def store_put_batch(store: int, slots: np.ndarray, blob: np.ndarray,
                    beg: np.ndarray, end: np.ndarray) -> None:
    _fastpath_protos().forge_store_put_batch(ctypes.c_void_p(store), _np_ptr(slots), slots.shape[0],
                                             _np_ptr(blob), _np_ptr(beg), _np_ptr(end))


def store_free(store: int) -> None:
    _fastpath_protos().forge_store_free(ctypes.c_void_p(store))


def cache_new(ttl: float) -> int:
    return _fastpath_protos().forge_cache_new(ttl)


def cache_free(cache: int) -> None:
    _fastpath_protos().forge_cache_free(ctypes.c_void_p(cache))


def toolmap_new(blob: np.ndarray, beg: np.ndarray, end: np.ndarray) -> int:
    return _fastpath_protos().forge_toolmap_new(_np_ptr(blob), _np_ptr(beg), _np_ptr(end), beg.shape[0])


def toolmap_free(tm: int) -> None:
    _fastpath_protos().forge_toolmap_free(ctypes.c_void_p(tm))


def toolmap_resolve(tm: int, blob: np.ndarray, name_beg: np.ndarray, name_end: np.ndarray) -> np.ndarray:
    m = name_beg.shape[0]
    out = np.empty(m, dtype=np.int32)
    _fastpath_protos().forge_toolmap_resolve(ctypes.c_void_p(tm), _np_ptr(blob),
                                             _np_ptr(name_beg), _np_ptr(name_end), m, _np_ptr(out))
    return out


def decide(blob, id_beg, id_end, args_beg, args_end, tool_idx, name_beg, name_end,
           deny_m, harm_m, pii_m, regex_m, norm_m, schema_m,
           mod_block, mod_cat, mod_score, hit, hit_slot, user_hash,
           tool_flags, tool_required_bits, tool_typed_pairs,
           tname_beg, tname_end, tname_blob, tool_native_kind, nest_bits,
           deny_words, deny_off, harm_cats, harm_off, mod_cats, mod_off,
           slot_store: int, exact_cache: int, now: float):
    """→ (state i32[m], native_kind i32[m], arena bytes np, resp_beg i64, resp_end i64)."""
    m = id_beg.shape[0]
    lib = _fastpath_protos()
    state = np.empty(m, dtype=np.int32)
    nk = np.empty(m, dtype=np.int32)
    reason = np.empty(m, dtype=np.int8)
    resp_beg = np.empty(m, dtype=np.int64)
    resp_end = np.empty(m, dtype=np.int64)
    cap = m * 96 + 4096
    while True:
        arena = np.empty(cap, dtype=np.uint8)
        n = lib.forge_decide(
            _np_ptr(blob), _np_ptr(id_beg), _np_ptr(id_end), _np_ptr(args_beg), _np_ptr(args_end),
            _np_ptr(tool_idx), _np_ptr(name_beg), _np_ptr(name_end),
            _np_ptr(deny_m), _np_ptr(harm_m), _np_ptr(pii_m), _np_ptr(regex_m), _np_ptr(norm_m), _np_ptr(schema_m),
            _np_ptr(mod_block), _np_ptr(mod_cat), _np_ptr(mod_score), _np_ptr(hit), _np_ptr(hit_slot),
            _np_ptr(user_hash),
            m,
            _np_ptr(tool_flags), _np_ptr(tool_required_bits), _np_ptr(tool_typed_pairs),
            _np_ptr(tname_beg), _np_ptr(tname_end), _np_ptr(tname_blob), _np_ptr(tool_native_kind),
            ctypes.c_uint32(nest_bits),
            _np_ptr(deny_words), _np_ptr(deny_off), deny_off.shape[0] - 1,
            _np_ptr(harm_cats), _np_ptr(harm_off), harm_off.shape[0] - 1,
            _np_ptr(mod_cats), _np_ptr(mod_off), mod_off.shape[0] - 1,
            ctypes.c_void_p(slot_store), ctypes.c_void_p(exact_cache), now,
            _np_ptr(state), _np_ptr(nk), _np_ptr(reason), _np_ptr(arena), cap, _np_ptr(resp_beg), _np_ptr(resp_end))
        if n >= 0:
            return state, nk, reason, arena, resp_beg, resp_end, int(n)
        cap = -int(n) + 4096


def finalize(blob, id_beg, id_end, args_beg, args_end, tool_idx, user_hash,
             rows, res_blob, res_beg, res_end, needs_host,
             tname_beg, tname_end, tname_blob, tool_flags,
             exact_cache: int, now: float, exact_ttl: float):
    """→ (arena np, resp_beg i64, resp_end i64, is_error u8, cacheable u8)."""
    lib = _fastpath_protos()
    n_rows = rows.shape[0]
    resp_beg2 = np.empty(n_rows, dtype=np.int64)
    resp_end2 = np.empty(n_rows, dtype=np.int64)
    is_err = np.empty(n_rows, dtype=np.uint8)
    cacheable = np.empty(n_rows, dtype=np.uint8)
    cap = int(res_blob.shape[0]) + n_rows * 64 + 4096
    while True:
        arena = np.empty(cap, dtype=np.uint8)
        n = lib.forge_finalize(
            _np_ptr(blob), _np_ptr(id_beg), _np_ptr(id_end), _np_ptr(args_beg), _np_ptr(args_end),
            _np_ptr(tool_idx), _np_ptr(user_hash), id_beg.shape[0],
            _np_ptr(rows), n_rows,
            _np_ptr(res_blob), _np_ptr(res_beg), _np_ptr(res_end), _np_ptr(needs_host),
            _np_ptr(tname_beg), _np_ptr(tname_end), _np_ptr(tname_blob), _np_ptr(tool_flags),
            ctypes.c_void_p(exact_cache), now, exact_ttl,
            _np_ptr(arena), cap, _np_ptr(resp_beg2), _np_ptr(resp_end2),
            _np_ptr(is_err), _np_ptr(cacheable))
        if n >= 0:
            return arena, resp_beg2, resp_end2, is_err, cacheable
        cap = -int(n) + 4096


RW_DONE, RW_PUNT, RW_BLOCKED, RW_BADJSON, RW_DENY = 0, 1, 2, 3, 4


def rewrite_rows(blob: np.ndarray, args_beg: np.ndarray, args_end: np.ndarray,
                 do_flags: np.ndarray, pii_want: np.ndarray,
                 pii_active_mask: int, pii_mode: int,
                 norm_collapse: bool, norm_strip: bool,
                 deny_blob: Optional[np.ndarray] = None,
                 deny_off: Optional[np.ndarray] = None,
                 deny_ci: bool = True,
                 harm_blob: Optional[np.ndarray] = None,
                 harm_off: Optional[np.ndarray] = None,
                 sk_tables: Optional[tuple] = None,
                 sk_lo: Optional[np.ndarray] = None,
                 sk_hi: Optional[np.ndarray] = None):
    """Native rewrite pass (rewrite.cpp): normalizer + PII over the flagged
    rows that fit the provable-equivalence envelope; everything else gets
    RW_PUNT and takes the Python path.
    → (status i32[n], found u32[n], deny_hit i32[n], harm_hit i32[n],
       schema_ok u8[n] (1 pass / 0 fail / 2 unchecked),
       arena u8, out_beg/out_end i64[n] (dispatch form, wire key order),
       scan_beg/scan_end i64[n] (sorted-keys scan form))."""
    n = args_beg.shape[0]
    lib = _load()
    status = np.empty(n, dtype=np.int32)
    found = np.empty(n, dtype=np.uint32)
    out_beg = np.empty(n, dtype=np.int64)
    out_end = np.empty(n, dtype=np.int64)
    scan_beg = np.empty(n, dtype=np.int64)
    scan_end = np.empty(n, dtype=np.int64)
    deny_hit = np.empty(n, dtype=np.int32)
    harm_hit = np.empty(n, dtype=np.int32)
    schema_ok = np.full(n, 2, dtype=np.uint8)
    n_deny = (deny_off.shape[0] - 1) if deny_off is not None else 0
    n_harm = (harm_off.shape[0] - 1) if harm_off is not None else 0
    sk_blob = sk_beg = sk_end = sk_type = sk_req = None
    if sk_tables is not None:
        sk_blob, sk_beg, sk_end, sk_type, sk_req = sk_tables
    cap = int((args_end - args_beg).sum()) * 2 + n * 32 + 4096
    while True:
        arena = np.empty(cap, dtype=np.uint8)
        rc = lib.forge_rewrite_rows(
            _np_ptr(blob), _np_ptr(args_beg), _np_ptr(args_end), n,
            _np_ptr(do_flags), _np_ptr(pii_want),
            ctypes.c_uint32(pii_active_mask), pii_mode,
            1 if norm_collapse else 0, 1 if norm_strip else 0,
            _np_ptr(deny_blob), _np_ptr(deny_off), n_deny, 1 if deny_ci else 0,
            _np_ptr(status), _np_ptr(found), _np_ptr(deny_hit),
            _np_ptr(arena), cap, _np_ptr(out_beg), _np_ptr(out_end),
            _np_ptr(scan_beg), _np_ptr(scan_end),
            _np_ptr(harm_blob), _np_ptr(harm_off), n_harm, _np_ptr(harm_hit),
            _np_ptr(sk_blob), _np_ptr(sk_beg), _np_ptr(sk_end),
            _np_ptr(sk_type), _np_ptr(sk_req),
            _np_ptr(sk_lo), _np_ptr(sk_hi), _np_ptr(schema_ok))
        if rc >= 0:
            return status, found, deny_hit, harm_hit, schema_ok, arena, out_beg, out_end, scan_beg, scan_end
        cap = -int(rc) + 4096


def post_rows(res_blob: np.ndarray, res_beg: np.ndarray, res_end: np.ndarray,
              do_flags: np.ndarray, pii_active_mask: int, pii_mode: int,
              harm_blob: Optional[np.ndarray] = None,
              harm_off: Optional[np.ndarray] = None,
              toon_min_size: int = 1 << 60, toon_min_savings: float = 1.0):
    """Native result post chain (rewrite.cpp forge_post_rows): the
    _host_post hot path — pii → harm → toon → wire serialization — for
    eligible flagged results; RW_PUNT rows take the Python path.
    → (status i32[n], found u32[n], harm_hit i32[n], is_err u8[n],
       arena u8, out_beg/out_end i64[n])."""
    n = res_beg.shape[0]
    lib = _load()
    status = np.empty(n, dtype=np.int32)
    found = np.empty(n, dtype=np.uint32)
    harm_hit = np.empty(n, dtype=np.int32)
    is_err = np.empty(n, dtype=np.uint8)
    out_beg = np.empty(n, dtype=np.int64)
    out_end = np.empty(n, dtype=np.int64)
    n_harm = (harm_off.shape[0] - 1) if harm_off is not None else 0
    cap = int((res_end - res_beg).sum()) * 2 + n * 64 + 4096
    while True:
        arena = np.empty(cap, dtype=np.uint8)
        rc = lib.forge_post_rows(
            _np_ptr(res_blob), _np_ptr(res_beg), _np_ptr(res_end), n,
            _np_ptr(do_flags),
            ctypes.c_uint32(pii_active_mask), pii_mode,
            _np_ptr(harm_blob), _np_ptr(harm_off), n_harm,
            ctypes.c_int64(toon_min_size), ctypes.c_double(toon_min_savings),
            _np_ptr(status), _np_ptr(found), _np_ptr(harm_hit), _np_ptr(is_err),
            _np_ptr(arena), cap, _np_ptr(out_beg), _np_ptr(out_end))
        if rc >= 0:
            return status, found, harm_hit, is_err, arena, out_beg, out_end
        cap = -int(rc) + 4096


TF_REACHABLE, TF_DENY, TF_PII, TF_REGEX, TF_NORM, TF_MOD, TF_HARM = (1 << i for i in range(7))
TF_SCHEMA_FAST, TF_SCHEMA_HOST, TF_CACHE, TF_EXACT, TF_BREAKER_OPEN = (1 << i for i in range(7, 12))
ST_DISPATCH_NATIVE, ST_ANSWERED, ST_REWRITE, ST_HOST_SCHEMA, ST_DISPATCH_PY = 0, 1, 2, 3, 4


def store_get(store: int, slot: int) -> Optional[bytes]:
    lib = _fastpath_protos()
    buf = ctypes.create_string_buffer(4096)
    n = lib.forge_store_get(ctypes.c_void_p(store), slot, buf, 4096)
    if n < 0 or n == 0:
        return None
    if n > 4096:
        buf = ctypes.create_string_buffer(int(n))
        n = lib.forge_store_get(ctypes.c_void_p(store), slot, buf, int(n))
    return buf.raw[:int(n)]
