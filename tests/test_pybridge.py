"""Direct-unit tier for the forge_pybridge CPython extension
(ops/csrc/pybridge.c) — the C loops every batch's response assembly goes
through. Exercised indirectly by every pipeline test; pinned here at the
function level (empty inputs, zero-length spans, interleaving)."""

import numpy as np
import pytest

from mcp_context_forge_amd.ops.pybridge import get


@pytest.fixture(scope="module")
def pb():
    return get()


def test_concat_with_offsets_roundtrip(pb):
    parts = [b"alpha", b"", b"bc", b"\x00\xffbin"]
    joined, offs_raw = pb.concat_with_offsets(parts)
    offs = np.frombuffer(offs_raw, dtype=np.int64)
    assert joined == b"alphabc\x00\xffbin"
    assert list(offs) == [0, 5, 5, 7, 12]
    for i, p in enumerate(parts):
        assert joined[offs[i]:offs[i + 1]] == p
    j2, o2 = pb.concat_with_offsets([])
    assert j2 == b"" and list(np.frombuffer(o2, dtype=np.int64)) == [0]


def test_scatter_slices_into_responses(pb):
    arena = np.frombuffer(b"AAABBCCCC", dtype=np.uint8)
    beg = np.asarray([0, 3, 5], dtype=np.int64)
    end = np.asarray([3, 5, 9], dtype=np.int64)
    rows = np.asarray([2, 0, 4], dtype=np.int64)
    responses = [None] * 5
    pb.scatter_slices(arena, beg, end, rows, responses)
    assert responses == [b"BB", None, b"AAA", None, b"CCCC"]


def test_slices_list_zero_length_and_order(pb):
    arena = np.frombuffer(b"xyz123", dtype=np.uint8)
    beg = np.asarray([3, 0, 2, 2], dtype=np.int64)
    end = np.asarray([6, 2, 2, 3], dtype=np.int64)
    out = pb.slices_list(arena, beg, end)
    assert out == [b"123", b"xy", b"", b"z"]


def test_frame_pack_unpack_roundtrip(pb):
    """Edge owner↔worker frame: [u32 payload][u32 n] then per-entry
    [u64 id][u32 len][bytes]; None rides as the 0xFFFFFFFF sentinel."""
    ids = np.asarray([7, 0, 2**40, 3], dtype=np.int64)
    msgs = [b"hello", b"", b"\x00" * 17, b"tail"]
    frame = pb.pack_frame(ids.tobytes(), msgs)
    # strip the [u32 payload][u32 n] header before unpacking
    payload = bytes(frame)[8:]
    ids_raw, back = pb.unpack_frame(payload, len(msgs))
    assert list(np.frombuffer(ids_raw, dtype=np.int64)) == list(ids)
    assert back == msgs
    # None encodes as the 0xFFFFFFFF no-body sentinel (owner->worker only)
    f2 = bytes(pb.pack_frame(np.asarray([1], dtype=np.int64).tobytes(), [None]))
    assert f2[8 + 8:8 + 12] == b"\xff\xff\xff\xff" and len(f2) == 8 + 12
