"""Property-based tier (hypothesis): structured random inputs beyond the
hand-seeded fuzz corpora.

1. TOON encode→decode losslessness over arbitrary JSON-able trees whose
   strings avoid the decoder's documented ambiguities (leading/trailing
   whitespace is quoted away by the encoder; the decoder's CSV split is
   exact for quoted fields).
2. The C rewrite lane's canonicalization (rewrite.cpp) against
   json.dumps over arbitrary ASCII/int trees — byte equality for both
   the wire-order and sorted forms.
"""

import json
import string

import numpy as np
from hypothesis import given, settings
from hypothesis import strategies as st

from mcp_context_forge_amd.ops import hip
from mcp_context_forge_amd.plugins import toon

# -------------------------------------------------------- strategies

# strings whose repr survives TOON's line-oriented format: no newlines
# (the encoder emits them raw inside quoted scalars where the decoder
# would split lines — a documented format limitation, like YAML's)
_toon_text = st.text(alphabet=string.ascii_letters + string.digits + " _.@+-,:#$%&*()[]{}'\"\\/",
                     max_size=24).filter(lambda s: s.strip() == s)

_toon_scalar = st.one_of(st.none(), st.booleans(), st.integers(-10**12, 10**12), _toon_text)


def _toon_tree(depth=3):
    if depth == 0:
        return _toon_scalar
    sub = _toon_tree(depth - 1)
    key = st.text(alphabet=string.ascii_lowercase + string.digits + "_", min_size=1, max_size=8)
    return st.one_of(
        _toon_scalar,
        st.lists(sub, max_size=4),
        st.dictionaries(key, sub, max_size=4),
    )


@settings(max_examples=300, deadline=None)
@given(_toon_tree())
def test_toon_roundtrip_lossless(value):
    text = toon.encode(value)
    back = toon.decode(text)
    assert back == value, (value, text, back)


# ------------------------------------------- C canonicalization parity

_ascii_text = st.text(alphabet=[chr(c) for c in range(0x20, 0x7F)] + ["\t", "\n", "\r"],
                      max_size=30)
_c_scalar = st.one_of(st.none(), st.booleans(), st.integers(-10**30, 10**30), _ascii_text)


def _c_tree(depth=3):
    if depth == 0:
        return _c_scalar
    sub = _c_tree(depth - 1)
    return st.one_of(
        _c_scalar,
        st.lists(sub, max_size=4),
        st.dictionaries(_ascii_text.filter(lambda s: len(s) <= 10), sub, max_size=4),
    )


@settings(max_examples=300, deadline=None)
@given(_c_tree(), st.booleans())
def test_c_canonicalization_matches_json_dumps(value, pretty):
    raw = json.dumps(value, indent=2 if pretty else None,
                     sort_keys=False, ensure_ascii=True).encode()
    status, found, dh, hh, sok, arena, ob, oe, sb, se = hip.rewrite_rows(
        np.frombuffer(raw, dtype=np.uint8).copy() if raw else np.zeros(1, dtype=np.uint8),
        np.asarray([0], dtype=np.int32), np.asarray([len(raw)], dtype=np.int32),
        np.asarray([0], dtype=np.uint8), np.asarray([0], dtype=np.uint32),
        0, 0, True, True)
    assert status[0] == hip.RW_DONE, (raw, status[0])
    wire = json.dumps(value, separators=(",", ":")).encode()
    srt = json.dumps(value, separators=(",", ":"), sort_keys=True).encode()
    assert arena[ob[0]:oe[0]].tobytes() == wire, raw
    assert arena[sb[0]:se[0]].tobytes() == srt, raw
