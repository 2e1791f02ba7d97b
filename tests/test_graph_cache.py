"""Bucket-selection logic of the hipGraph cache (gpu/graphs.py) — pure
CPU: capture itself is exercised by the GPU tier
(test_gpu_parity.py::test_pass1_graph_matches_eager)."""

import numpy as np
import pytest

from mcp_context_forge_amd.gpu.graphs import BYTES_PER_ROW, ROW_BUCKETS, GraphCache


class _FakeBucket:
    def __init__(self, pipeline, rows, bankset, with_classifier, bytes_per_row):
        self.rows = rows
        self.byte_cap = rows * bytes_per_row


@pytest.fixture()
def cache(monkeypatch):
    import mcp_context_forge_amd.gpu.graphs as g

    monkeypatch.setattr(g, "_Bucket", _FakeBucket)
    return GraphCache(pipeline=object(), bankset=object(), with_classifier=False)


def test_bucket_rounding(cache):
    assert cache.get(1, 100).rows == 1024
    assert cache.get(1024, 100).rows == 1024
    assert cache.get(1025, 100).rows == 2048
    assert cache.get(8192, 100).rows == 8192
    assert cache.get(8193, 100) is None  # beyond the largest bucket → eager


def test_byte_overflow_escalates_bucket(cache):
    # a small row count with a huge blob skips to a bucket whose byte cap fits
    big = 1024 * BYTES_PER_ROW + 1
    b = cache.get(512, big)
    assert b is not None and b.rows == 2048
    # larger than every bucket's cap → eager
    assert cache.get(512, ROW_BUCKETS[-1] * BYTES_PER_ROW + 1) is None


def test_buckets_are_cached(cache):
    a = cache.get(100, 10)
    b = cache.get(900, 10)
    assert a is b
    assert cache.captures == 1


def test_disabled_flag(monkeypatch):
    import mcp_context_forge_amd.gpu.graphs as g

    monkeypatch.setattr(g, "_Bucket", _FakeBucket)
    monkeypatch.setenv("FORGE_PASS1_GRAPH", "0")
    c = GraphCache(pipeline=object(), bankset=object(), with_classifier=False)
    assert c.get(100, 10) is None


def test_capture_failure_falls_back(monkeypatch):
    import mcp_context_forge_amd.gpu.graphs as g

    class Boom:
        def __init__(self, *a, **k):
            raise RuntimeError("capture failed")

    monkeypatch.setattr(g, "_Bucket", Boom)
    c = GraphCache(pipeline=object(), bankset=object(), with_classifier=False)
    assert c.get(100, 10) is None
    assert c.disabled  # permanently eager after a failed capture
