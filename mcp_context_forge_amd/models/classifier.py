"""Moderation / harm classifiers over hashed text features.

Reference analog: plugins/content_moderation (external-API classifiers) and
plugins/harmful_content_detector (keyword heuristics). Per BASELINE.json the
MI355X build runs these as **local bf16 MFMA matmuls with random-init
weights** over hashed count-vector features: feature[B,D] @ W1[D,H] → GELU →
@ W2[H,C] → sigmoid per-category score.

The torch module here is the fp32 CPU reference; the GPU pipeline runs the
same weights through the hand-written MFMA GEMM kernel (ops/csrc/gemm_bf16.hip)
and numerics tests compare the two.
"""

from __future__ import annotations

from typing import List, Optional

import torch
import torch.nn as nn

MODERATION_CATEGORIES = [
    "hate", "violence", "sexual", "self_harm", "harassment", "illegal", "profanity", "spam",
]


class HashedTextClassifier(nn.Module):
    """2-layer MLP over hashed count vectors. Deterministic init per seed."""

    def __init__(self, dim: int = 4096, hidden: int = 1024, classes: int = len(MODERATION_CATEGORIES), seed: int = 1234):
        super().__init__()
        self.dim, self.hidden, self.classes = dim, hidden, classes
        g = torch.Generator().manual_seed(seed)
        self.w1 = nn.Parameter(torch.randn(dim, hidden, generator=g) * (dim ** -0.5))
        self.b1 = nn.Parameter(torch.zeros(hidden))
        self.w2 = nn.Parameter(torch.randn(hidden, classes, generator=g) * (hidden ** -0.5))
        self.b2 = nn.Parameter(torch.zeros(classes))

    def forward(self, feats: torch.Tensor) -> torch.Tensor:
        """feats [B, D] float → category probabilities [B, C]."""
        h = torch.nn.functional.gelu(feats @ self.w1 + self.b1, approximate="tanh")
        return torch.sigmoid(h @ self.w2 + self.b2)

    @torch.no_grad()
    def score(self, feats: torch.Tensor) -> torch.Tensor:
        return self.forward(feats)


def category_names(classes: int) -> List[str]:
    if classes <= len(MODERATION_CATEGORIES):
        return MODERATION_CATEGORIES[:classes]
    return MODERATION_CATEGORIES + [f"cat{i}" for i in range(len(MODERATION_CATEGORIES), classes)]
