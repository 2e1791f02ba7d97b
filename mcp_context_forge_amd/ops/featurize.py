"""Hashed count-vector featurization — CPU reference.

This is the exact semantic the HIP kernel `ops/csrc/featurize.hip`
implements; tests compare the two. Reference analog: the semantic cache's
`_vectorize` (plugins/response_cache_by_prompt/response_cache_by_prompt.py:55
hashed-token count vectors) and the moderation classifiers' text features.

Tokenization: maximal runs of [A-Za-z0-9_] bytes; ASCII-lowercased.
Hash: FNV-1a 32-bit over the token bytes; bucket = hash & (dim-1)
(dim must be a power of two so CPU/GPU agree without modulo).
"""

from __future__ import annotations

from typing import List, Union

import numpy as np

FNV_OFFSET = 0x811C9DC5
FNV_PRIME = 0x01000193
MASK32 = 0xFFFFFFFF


def fnv1a(data: bytes) -> int:
    h = FNV_OFFSET
    for b in data:
        h = ((h ^ b) * FNV_PRIME) & MASK32
    return h


def _is_word(b: int) -> bool:
    return (0x30 <= b <= 0x39) or (0x41 <= b <= 0x5A) or (0x61 <= b <= 0x7A) or b == 0x5F


def _lower(b: int) -> int:
    return b + 0x20 if 0x41 <= b <= 0x5A else b


def token_buckets(text: Union[str, bytes], dim: int) -> List[int]:
    """Bucket index per token, in order."""
    assert dim & (dim - 1) == 0, "dim must be a power of two"
    data = text.encode("utf-8", "replace") if isinstance(text, str) else text
    out: List[int] = []
    h = FNV_OFFSET
    in_tok = False
    for b in data:
        if _is_word(b):
            h = ((h ^ _lower(b)) * FNV_PRIME) & MASK32
            in_tok = True
        else:
            if in_tok:
                out.append(h & (dim - 1))
            h = FNV_OFFSET
            in_tok = False
    if in_tok:
        out.append(h & (dim - 1))
    return out


def featurize(text: Union[str, bytes], dim: int, normalize: bool = True, dtype=np.float32) -> np.ndarray:
    """Hashed count vector (optionally L2-normalized)."""
    vec = np.zeros(dim, dtype=np.float32)
    for b in token_buckets(text, dim):
        vec[b] += 1.0
    if normalize:
        n = float(np.linalg.norm(vec))
        if n > 0:
            vec /= n
    return vec.astype(dtype)


def featurize_batch(texts: List[Union[str, bytes]], dim: int, normalize: bool = True) -> np.ndarray:
    out = np.zeros((len(texts), dim), dtype=np.float32)
    for i, t in enumerate(texts):
        out[i] = featurize(t, dim, normalize)
    return out
