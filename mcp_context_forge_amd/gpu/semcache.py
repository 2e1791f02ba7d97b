"""HBM-resident semantic response cache.

Reference analog: plugins/response_cache_by_prompt (count-vector cosine in
Python dicts) + the Redis registry caches. Per BASELINE.json the cache keys
live in device HBM as a bf16 matrix [capacity, dim]; lookup is a chunked
MFMA GEMM (feats @ keys^T) folded by the rows_argmax_merge kernel — no
Redis, no host round-trip per entry. Results (host objects) stay on host
indexed by slot.
"""

from __future__ import annotations

import time
from typing import Any, List, Optional, Tuple

import numpy as np
import torch

from ..ops import hip


def tool_hash(name: str) -> int:
    h = 1469598103934665603
    for b in name.encode():
        h = ((h ^ b) * 1099511628211) & ((1 << 63) - 1)
    return h


class GpuSemanticCache:
    def __init__(self, capacity: int = 65536, dim: int = 4096, threshold: float = 0.92,
                 ttl_s: float = 600.0, device: str = "cuda", chunk: int = 4096,
                 sketch_dim: int = 256):
        assert capacity % 128 == 0 and chunk % 128 == 0 and dim % 64 == 0
        self.capacity = capacity
        self.dim = dim
        self.threshold = threshold
        self.ttl_s = ttl_s
        self.chunk = min(chunk, capacity)
        self.device = device
        self.keys = torch.zeros((capacity, dim), dtype=torch.bfloat16, device=device)
        self.valid = torch.zeros(capacity, dtype=torch.uint8, device=device)
        # two-stage search: rank with a random-projection sketch sweep
        # (16x less FLOP than the full sweep at dim=4096/sketch=256), then
        # verify the top candidate with an exact full-dim dot (verify_dot
        # kernel) — the threshold compare always sees the TRUE cosine.
        self.sketch_dim = sketch_dim if (sketch_dim and sketch_dim % 64 == 0
                                         and sketch_dim < dim) else 0
        if self.sketch_dim:
            g = torch.Generator().manual_seed(0x5EED)
            proj = (torch.randint(0, 2, (self.sketch_dim, dim), generator=g).float() * 2 - 1)
            self.proj_t = (proj / dim ** 0.5).to(torch.bfloat16).to(device)  # [sk, dim]
            self.keys_sk = torch.zeros((capacity, self.sketch_dim),
                                       dtype=torch.bfloat16, device=device)
        else:
            self.proj_t = None
            self.keys_sk = None

        # host-side metadata mirrors (slot-indexed)
        self.tool_hashes = np.zeros(capacity, dtype=np.int64)
        self.timestamps = np.zeros(capacity, dtype=np.float64)
        self.results: List[Any] = [None] * capacity
        self.write_ptr = 0
        self.size = 0
        self.hits = 0
        self.misses = 0

    def lookup(self, feats_bf16: torch.Tensor) -> Tuple[torch.Tensor, torch.Tensor, Optional[torch.Tensor]]:
        """feats [Bpad, dim] bf16 → (best_val fp32 [Bpad], best_idx i32 [Bpad]) on device.

        Chunked sweep over the key matrix; scores for padded/invalid slots are
        suppressed by the validity mask inside the merge kernel.
        """
        bpad = feats_bf16.shape[0]
        best_val = torch.full((bpad,), -1e30, dtype=torch.float32, device=self.device)
        best_idx = torch.full((bpad,), -1, dtype=torch.int32, device=self.device)
        sk = None
        if self.sketch_dim:
            # sketches returned to the caller, which passes them back into
            # insert_features (no shared slot — batches may be in flight
            # concurrently)
            sk = hip.gemm_bt(feats_bf16, self.proj_t, out_bf16=True)
        if self.size == 0:
            return best_val, best_idx, sk
        active = min(self.size, self.capacity)
        active_pad = ((active + self.chunk - 1) // self.chunk) * self.chunk
        if self.sketch_dim:
            # stage 1: approximate ranking over the sketch matrix
            for c0 in range(0, active_pad, self.chunk):
                nc = min(self.chunk, self.capacity - c0)
                scores = hip.gemm_bt(sk, self.keys_sk[c0:c0 + nc])
                hip.rows_argmax_merge(scores, best_val, best_idx, idx_base=c0,
                                      valid=self.valid[c0:c0 + nc])
            # stage 2: exact full-dim dot of each row's top candidate
            best_val = hip.verify_dot(feats_bf16, self.keys, best_idx)
            return best_val, best_idx, sk
        for c0 in range(0, active_pad, self.chunk):
            nc = min(self.chunk, self.capacity - c0)
            scores = hip.gemm_bt(feats_bf16, self.keys[c0:c0 + nc])
            hip.rows_argmax_merge(scores, best_val, best_idx, idx_base=c0, valid=self.valid[c0:c0 + nc])
        return best_val, best_idx, sk

    def resolve_hits_np(self, best_val: np.ndarray, best_idx: np.ndarray,
                        tool_hashes: np.ndarray) -> np.ndarray:
        """Vectorized hit confirmation → bool mask. Caller fetches
        self.results[best_idx[i]] for hits."""
        n = best_val.shape[0]
        if self.size == 0:
            self.misses += n
            return np.zeros(n, dtype=bool)
        idx = best_idx.astype(np.int64)
        safe = np.clip(idx, 0, self.capacity - 1)
        now = time.monotonic()
        mask = (idx >= 0) & (best_val >= self.threshold) \
            & (self.tool_hashes[safe] == tool_hashes) \
            & ((now - self.timestamps[safe]) <= self.ttl_s)
        nh = int(mask.sum())
        self.hits += nh
        self.misses += n - nh
        return mask

    def resolve_hits(self, best_val: np.ndarray, best_idx: np.ndarray,
                     tool_hashes: np.ndarray) -> List[Optional[Any]]:
        """Host-side hit confirmation: threshold + tool identity + TTL
        (reference semantics: _find_best threshold lookup :163)."""
        now = time.monotonic()
        out: List[Optional[Any]] = []
        for val, idx, th in zip(best_val, best_idx, tool_hashes):
            if idx < 0 or val < self.threshold:
                out.append(None)
                self.misses += 1
                continue
            slot = int(idx)
            if self.tool_hashes[slot] != th or (now - self.timestamps[slot]) > self.ttl_s:
                out.append(None)
                self.misses += 1
                continue
            self.hits += 1
            out.append(self.results[slot])
        return out

    def assign_slots(self, n: int) -> np.ndarray:
        """Ring-allocate n slots (eviction = overwrite)."""
        slots = (np.arange(n, dtype=np.int64) + self.write_ptr) % self.capacity
        self.write_ptr = int((self.write_ptr + n) % self.capacity)
        self.size = min(self.size + n, self.capacity)
        return slots.astype(np.int32)

    def insert_features(self, feats_bf16: torch.Tensor, rows: np.ndarray, slots: np.ndarray,
                        tool_hashes: np.ndarray, sketch: Optional[torch.Tensor] = None) -> None:
        """Fused gather+scatter of feature rows into pre-assigned slots
        (one kernel; results are stored separately — C++ slot store)."""
        rows_t = torch.from_numpy(rows.astype(np.int32)).to(self.device, non_blocking=True)
        slot_t = torch.from_numpy(slots.astype(np.int32)).to(self.device, non_blocking=True)
        hip.rows_gather_scatter_bf16(feats_bf16, rows_t, slot_t, self.keys, self.valid)
        if self.sketch_dim:
            sk = sketch
            if sk is None or sk.shape[0] != feats_bf16.shape[0]:
                sk = hip.gemm_bt(feats_bf16, self.proj_t, out_bf16=True)
            hip.rows_gather_scatter_bf16(sk, rows_t, slot_t, self.keys_sk, self.valid)
        now = time.monotonic()
        self.tool_hashes[slots] = tool_hashes
        self.timestamps[slots] = now

    def insert_batch(self, feats_bf16: torch.Tensor, rows: List[int],
                     tool_hashes: List[int], results: List[Any]) -> None:
        """Scatter feature rows into cache slots (ring eviction)."""
        if not rows:
            return
        n = len(rows)
        slots = self.assign_slots(n)
        self.insert_features(feats_bf16, np.asarray(rows), slots, np.asarray(tool_hashes, dtype=np.int64))
        for s, res in zip(slots, results):
            self.results[int(s)] = res

    def stats(self) -> dict:
        return {"size": self.size, "capacity": self.capacity, "hits": self.hits, "misses": self.misses,
                "hbm_bytes": self.keys.numel() * 2}
