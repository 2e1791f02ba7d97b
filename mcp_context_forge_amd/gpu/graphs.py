"""hipGraph capture of pass 1 (scan_multi + featurize + MFMA classifier).

At HTTP batch sizes every batch pays the CPU-side cost of the H2D stages
plus 4-6 kernel launches; capturing the whole sequence as one hipGraph
turns that into a single `hipGraphLaunch` (round-3 worklist item 2).
Batches are bucketed by row count (1k/2k/4k/8k) with rows padded as
empty spans (beg == end == 0: the scan masks to 0, featurize emits the
zero vector, classifier outputs on pad rows are sliced away) and the
byte blob staged through a fixed pinned arena, so every replay sees
identical shapes and pointers. Outputs are copied INTO pinned host
buffers inside the graph — after the event sync the host reads them
with no further D2H.

Correctness containment: the graph executes the exact same kernels with
the same tensors as the eager path; the GPU parity fuzz exercises the
pipeline with graphs on. FORGE_PASS1_GRAPH=0 disables capture (eager
fallback) for A/B checks. Graphs bake the bank-table device pointers,
so the cache is dropped whenever the tool registry rebuilds
_bankset1 (pipeline._rebuild_tool_meta)."""

from __future__ import annotations

import logging
import os
from typing import Dict, Optional

import numpy as np
import torch

from ..ops import hip

logger = logging.getLogger(__name__)

ROW_BUCKETS = (1024, 2048, 4096, 8192)
BYTES_PER_ROW = 384  # pinned blob capacity per bucket row (HTTP payloads ~150 B)


class _Bucket:
    def __init__(self, pipeline, rows: int, bankset, with_classifier: bool,
                 bytes_per_row: int):
        self.rows = rows
        self.byte_cap = rows * bytes_per_row
        self.bankset = bankset
        dev = pipeline.device
        nb = bankset.n
        dim = pipeline.feat_dim
        clf = pipeline.classifier if with_classifier else None

        self.pin_blob = torch.empty(self.byte_cap, dtype=torch.uint8, pin_memory=True)
        self.pin_beg = torch.empty(rows, dtype=torch.int32, pin_memory=True)
        self.pin_end = torch.empty(rows, dtype=torch.int32, pin_memory=True)
        self.d_blob = torch.empty(self.byte_cap, dtype=torch.uint8, device=dev)
        self.d_beg = torch.empty(rows, dtype=torch.int32, device=dev)
        self.d_end = torch.empty(rows, dtype=torch.int32, device=dev)
        self.out_multi = torch.zeros((max(nb, 1), rows), dtype=torch.int32, device=dev)
        self.pin_multi = torch.empty((max(nb, 1), rows), dtype=torch.int32, pin_memory=True)
        self.feats = None
        self.h = None
        self.scores = None
        self.pin_scores = None
        if clf is not None:
            self.feats = torch.empty((rows, dim), dtype=torch.bfloat16, device=dev)
            self.h = torch.empty((rows, clf.hidden), dtype=torch.bfloat16, device=dev)
            self.scores = torch.empty((rows, clf.classes), dtype=torch.float32, device=dev)
            self.pin_scores = torch.empty((rows, clf.classes), dtype=torch.float32,
                                          pin_memory=True)
        self._pipeline = pipeline
        # warm up twice on a side stream (first-launch code-object load and
        # allocator state must settle before capture), then capture
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        self.pin_beg.zero_()
        self.pin_end.zero_()
        with torch.cuda.stream(s):
            for _ in range(2):
                self._launch()
        torch.cuda.current_stream().wait_stream(s)
        torch.cuda.synchronize()
        self.graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.graph):
            self._launch()

    def _launch(self) -> None:
        p = self._pipeline
        self.d_blob.copy_(self.pin_blob, non_blocking=True)
        self.d_beg.copy_(self.pin_beg, non_blocking=True)
        self.d_end.copy_(self.pin_end, non_blocking=True)
        hip.scan_multi(self.d_blob, self.d_beg, self.d_end, self.bankset, out=self.out_multi)
        if self.feats is not None:
            hip.featurize(self.d_blob, self.d_beg, self.d_end, p.feat_dim, out=self.feats)
            p.classifier.forward_into(self.feats, self.h, self.scores)
            self.pin_scores.copy_(self.scores, non_blocking=True)
        self.pin_multi.copy_(self.out_multi, non_blocking=True)

    def stage(self, blob: np.ndarray, args_b: np.ndarray, args_e: np.ndarray, m: int) -> None:
        nbytes = blob.nbytes
        self.pin_blob[:nbytes].numpy()[:] = blob.view(np.uint8).reshape(-1)
        pb = self.pin_beg.numpy()
        pe = self.pin_end.numpy()
        pb[:m] = args_b
        pe[:m] = args_e
        if m < self.rows:
            pb[m:] = 0
            pe[m:] = 0

    def replay(self) -> None:
        self.graph.replay()

    def read_multi(self, m: int) -> np.ndarray:
        # copy: the pinned buffer is overwritten by the next replay
        return np.array(self.pin_multi.numpy()[:, :m]).view(np.uint32)

    def read_scores(self, m: int) -> Optional[np.ndarray]:
        if self.pin_scores is None:
            return None
        return np.array(self.pin_scores.numpy()[:m])


class GraphCache:
    """Lazy per-bucket hipGraph cache for one (bank set, shape) family."""

    def __init__(self, pipeline, bankset, with_classifier: bool,
                 bytes_per_row: int = BYTES_PER_ROW):
        self._pipeline = pipeline
        self._bankset = bankset
        self._with_classifier = with_classifier
        self._bytes_per_row = bytes_per_row
        self._buckets: Dict[int, _Bucket] = {}
        self.disabled = os.environ.get("FORGE_PASS1_GRAPH", "1") == "0"
        self.replays = 0
        self.captures = 0

    def get(self, m: int, nbytes: int) -> Optional[_Bucket]:
        if self.disabled:
            return None
        for rows in ROW_BUCKETS:
            if m <= rows:
                if nbytes > rows * self._bytes_per_row:
                    continue  # oversized payloads → next bucket or eager
                b = self._buckets.get(rows)
                if b is None:
                    try:
                        b = _Bucket(self._pipeline, rows, self._bankset,
                                    self._with_classifier, self._bytes_per_row)
                        self.captures += 1
                    except Exception:
                        logger.exception("pass graph capture failed; eager fallback")
                        self.disabled = True
                        return None
                    self._buckets[rows] = b
                return b
        return None


def Pass1Graphs(pipeline):
    return GraphCache(pipeline, pipeline._bankset1,
                      with_classifier=pipeline.classifier is not None)


def Pass3Graphs(pipeline):
    # results are larger than requests (content + structuredContent echo)
    return GraphCache(pipeline, pipeline._bankset3, with_classifier=False,
                      bytes_per_row=640)
