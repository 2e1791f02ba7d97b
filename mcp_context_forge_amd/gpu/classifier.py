"""Device-resident bf16 classifier running on the hand-written MFMA kernels.

Weights mirror models/classifier.HashedTextClassifier (the fp32 CPU oracle):
  h = gelu(feats @ w1 + b1); p = sigmoid(h @ w2 + b2)
W1 is stored transposed [H, D] (the GEMM consumes B^T), W2 transposed [C, H]
for the skinny gemv head. Numerics: bf16 inputs/weights, fp32 accumulate.
"""

from __future__ import annotations

import torch

from ..models.classifier import HashedTextClassifier
from ..ops import hip


class GpuClassifier:
    def __init__(self, model: HashedTextClassifier, device: str = "cuda"):
        self.dim = model.dim
        self.hidden = model.hidden
        self.classes = model.classes
        assert self.dim % 64 == 0 and self.hidden % 128 == 0, "MFMA tile constraints"
        with torch.no_grad():
            self.w1t = model.w1.t().contiguous().to(device=device, dtype=torch.bfloat16)  # [H, D]
            self.b1 = model.b1.to(device=device, dtype=torch.float32).contiguous()
            self.w2t = model.w2.t().contiguous().to(device=device, dtype=torch.bfloat16)  # [C, H]
            self.b2 = model.b2.to(device=device, dtype=torch.float32).contiguous()

    def forward(self, feats_bf16: torch.Tensor) -> torch.Tensor:
        """feats [Bpad, D] bf16 (Bpad%128==0) → probabilities fp32 [Bpad, C]."""
        h = hip.gemm_bt(feats_bf16, self.w1t, self.b1, act=hip.ACT_GELU, out_bf16=True)
        return hip.gemv_head(h, self.w2t, self.b2, act=hip.ACT_SIGMOID)

    def forward_into(self, feats_bf16: torch.Tensor, h_buf: torch.Tensor,
                     scores_buf: torch.Tensor) -> None:
        """Capture-safe forward: writes into preallocated buffers."""
        hip.gemm_bt(feats_bf16, self.w1t, self.b1, act=hip.ACT_GELU, out_bf16=True, out=h_buf)
        hip.gemv_head(h_buf, self.w2t, self.b2, act=hip.ACT_SIGMOID, out=scores_buf)
